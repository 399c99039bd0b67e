"""Training CLI — drop-in for the reference's open_diloco/train_fsdp.py.

Same flag surface (Config/HvConfig/CkptConfig incl. dotted flags, see
config.py) and the same loop structure (train_fsdp.py:177-516): grad-accum
microbatches -> clip -> optimizer.step(scaler=) -> scheduler -> metrics ->
checkpoint.  MI355X mapping (one process per GPU over RCCL):

  - ``--hv`` enabled: every torch.distributed rank IS one DiLoCo worker
    (1 GPU per worker, BASELINE.json configs[4]); launch ONE torchrun with
    --nproc-per-node = number of workers.  ``total_batch_size`` is the
    per-worker batch (matching the reference's per-worker torchrun worlds).
    The cross-worker pseudo-gradient all-reduce happens inside
    DiLoCoOptimizer every ``hv.local_steps``.
  - ``--hv`` disabled: plain data parallelism; ranks average gradients every
    real step (the reference's FSDP NO_SHARD behaviour, train_fsdp.py:239-245
    + no_sync gating :377) via one flat RCCL all-reduce;
    ``total_batch_size`` is global and divided across ranks
    (train_fsdp.py:186-190).

Precision (train_fsdp.py:226-228): fp16-mixed (default) = f16 compute +
GradScaler; bf16-mixed = bf16 compute; 32-true = fp32 (CPU path).  Parameters
are always fp32 masters; on GPU the model's hand-written HIP kernels compute
in the chosen dtype.
"""

from __future__ import annotations

import datetime
import os
import time
from functools import partial

import torch
import torch.distributed as dist

from opendiloco_amd.ckpt import (
    CKPT_PREFIX,
    check_checkpoint_path_access,
    delete_old_checkpoints,
    get_diloco_rank_dir_name,
    get_resume_info,
    load_checkpoint,
    save_checkpoint,
)
from opendiloco_amd.config import Config, parse_argv
from opendiloco_amd.data import FakeTokenizedDataLoader
from opendiloco_amd.diloco import DiLoCoOptimizer
from opendiloco_amd.model import LlamaForCausalLM
from opendiloco_amd.optim import FusedAdamW, clip_grad_norm_flat_
from opendiloco_amd.schedule import get_cosine_schedule_with_warmup
from opendiloco_amd.utils_compat import (
    DummyLogger,
    WandbLogger,
    get_compression_kwargs,
    get_sharding_strategy,
    register_metrics_hooks,
)

TIMEOUT_NCCL_MINUTES = int(os.environ.get("TIMEOUT_NCCL_MINUTES", 120))
TARGET_LAYER_ACTIVATIONS = ["self_attn", "lm_head"]
TEST_VOCAB_SIZE = 1024  # reference train_fsdp.py:66


def ddp_setup():
    backend = "nccl" if torch.cuda.is_available() else "gloo"
    dist.init_process_group(backend=backend,
                            timeout=datetime.timedelta(minutes=TIMEOUT_NCCL_MINUTES))
    if torch.cuda.is_available():
        torch.cuda.set_device(int(os.environ.get("LOCAL_RANK", 0)))


def log(message):
    print(f"[rank {os.environ.get('LOCAL_RANK', 0)}] {message}", flush=True)


def _device() -> torch.device:
    if torch.cuda.is_available():
        return torch.device("cuda", int(os.environ.get("LOCAL_RANK", 0)))
    return torch.device("cpu")


def _compute_dtype(precision: str, device: torch.device) -> torch.dtype:
    if device.type == "cpu":
        return torch.float32
    if precision == "bf16-mixed":
        return torch.bfloat16
    if precision == "fp16-mixed":
        return torch.float16
    raise NotImplementedError(
        "precision=32-true on GPU is not supported by the MI355X build "
        "(the hot-path kernels compute in bf16/f16 with fp32 accumulate; "
        "use bf16-mixed, or run 32-true on CPU)")


def train(config: Config):
    get_sharding_strategy(config.sharding_strategy)  # validate flag
    local_rank = int(os.environ.get("LOCAL_RANK", 0))
    world_size = dist.get_world_size() if dist.is_initialized() else 1
    rank = dist.get_rank() if dist.is_initialized() else 0
    device = _device()

    hv = config.hv is not None
    if hv:
        # each rank is one DiLoCo worker; keep the per-worker batch semantics
        worker_batch = config.total_batch_size
    else:
        assert config.total_batch_size % world_size == 0
        worker_batch = config.total_batch_size // world_size
    assert worker_batch % config.per_device_train_batch_size == 0
    gradient_accumulation_steps = worker_batch // config.per_device_train_batch_size

    resume_from_ckpt, resume_path = get_resume_info(config.ckpt)

    metric_logger = None
    if rank == 0:
        logger_cls = WandbLogger if config.metric_logger_type == "wandb" else DummyLogger
        metric_logger = logger_cls(project=config.project, config=config.model_dump(),
                                   resume=resume_from_ckpt)
    if hv:
        log("diloco enabled (single-node RCCL backend; every rank is a worker)")
    if local_rank == 0:
        check_checkpoint_path_access(config.ckpt.path, rank,
                                     config.hv.world_rank if hv else None)

    # ---- data ----
    if not config.fake_data:
        raise NotImplementedError(
            "offline build: only --fake_data is supported (the reference streams "
            "C4 from the network, train_fsdp.py:136-159)")
    train_dataloader = FakeTokenizedDataLoader(
        seq_len=config.seq_length, vocab_size=TEST_VOCAB_SIZE,
        batch_size=config.per_device_train_batch_size,
        seed=config.data_seed, rank=rank)

    # ---- model ----
    if device.type == "cuda":
        from opendiloco_amd.gemm_tuning import enable_tuned_gemms

        enable_tuned_gemms()
    model = LlamaForCausalLM.from_pretrained(config.path_model)
    model = model.to(device)
    model.compute_dtype = _compute_dtype(config.precision, device)
    model.train()

    scaler_enabled = config.precision == "fp16-mixed" and device.type == "cuda"
    scaler = torch.amp.GradScaler(device.type if device.type == "cuda" else "cpu",
                                  enabled=scaler_enabled)

    # ---- optimizers (reference hyperparams, train_fsdp.py:250-253) ----
    inner_optimizer = partial(torch.optim.AdamW, lr=config.lr, weight_decay=0.1,
                              betas=(0.9, 0.95))

    def scheduler_fn(opt):
        return get_cosine_schedule_with_warmup(opt, num_warmup_steps=config.warmup_steps,
                                               num_training_steps=config.total_steps)

    if hv:
        outer_optimizer = partial(torch.optim.SGD, lr=config.hv.outer_lr, momentum=0.9,
                                  nesterov=True)
        diloco_args = dict(
            dht=None,
            run_id="llama",
            batch_size=worker_batch,
            num_inner_steps=config.hv.local_steps,
            outer_optimizer=outer_optimizer,
            inner_optimizer=inner_optimizer,
            params=model.parameters(),
            scheduler=None,
            all_reduce_strategy=config.hv.all_reduce_strategy,
            timeout_waiting_for_peers=config.hv.timeout_waiting_for_peers,
        )
        diloco_args.update(get_compression_kwargs(config.hv.hivemind_compression))
        if config.hv.matchmaking_time is not None:
            diloco_args["matchmaking_time"] = config.hv.matchmaking_time
        optimizer = DiLoCoOptimizer(**diloco_args)
        scheduler = scheduler_fn(optimizer.inner_optimizer)
        inner_opt_for_ckpt = optimizer.inner_optimizer
        outer_opt_for_ckpt = optimizer.state_averager.optimizer
    else:
        optimizer = FusedAdamW(list(model.parameters()), lr=config.lr, weight_decay=0.1,
                               betas=(0.9, 0.95))
        scheduler = scheduler_fn(optimizer)
        inner_opt_for_ckpt = optimizer
        outer_opt_for_ckpt = None

    # ---- resume ----
    start_step = 0
    if resume_from_ckpt:
        ckpt_path = resume_path
        if hv:
            ckpt_path = os.path.join(resume_path, get_diloco_rank_dir_name(rank))
        last_loss = load_checkpoint(
            checkpoint_path=ckpt_path, model=model, optimizer=inner_opt_for_ckpt,
            scheduler=scheduler, outer_optimizer=outer_opt_for_ckpt, scaler=scaler,
            data_loader=train_dataloader, rank=rank)
        if hv:
            optimizer.sync_outer_from_local()  # reference double-load semantics
        start_step = scheduler.last_epoch
        log(f"Resumed from checkpoint at step {start_step} with loss {last_loss}")

    if hv and not config.hv.skip_load_from_peers:
        optimizer.load_state_from_peers()

    current_time = time.time()
    log(f"starting from step {start_step}")

    loss_batch = torch.zeros((), device=device)
    log_activations: dict = {}
    flat = optimizer.flat

    step = start_step * gradient_accumulation_steps - 1
    max_num_peers = 0
    # high-priority main stream (same as bench.py): the side-stream dW
    # GEMMs (ops.py DK_DW_ASYNC) otherwise starve small main-stream kernels
    # at workgroup-dispatch arbitration.  DK_MAIN_PRIO=0 disables.
    main_stream = (torch.cuda.Stream(priority=-1)
                   if device.type == "cuda" and os.environ.get("DK_MAIN_PRIO", "1") != "0"
                   else None)
    import contextlib

    def _main_stream_ctx():
        return torch.cuda.stream(main_stream) if main_stream else contextlib.nullcontext()

    for batch in train_dataloader:
        step += 1
        real_step = (step + 1) // gradient_accumulation_steps
        is_accumulating = bool((step + 1) % gradient_accumulation_steps)

        logging_activations_steps = (config.log_activations_steps is not None
                                     and real_step % config.log_activations_steps == 0)
        if logging_activations_steps:
            handles = register_metrics_hooks(model, TARGET_LAYER_ACTIVATIONS,
                                             log_activations, gradient_accumulation_steps)

        for key in batch:
            batch[key] = batch[key].to(device)

        with _main_stream_ctx():
            outputs = model(**batch)
            loss = outputs.loss / gradient_accumulation_steps
            loss_batch += loss.detach()
            scaler.scale(loss).backward()
        if main_stream:
            torch.cuda.current_stream().wait_stream(main_stream)

        if logging_activations_steps:
            for handle in handles:
                handle.remove()

        if not is_accumulating:
            if not hv and world_size > 1:
                # FSDP NO_SHARD grad averaging (reference train_fsdp.py:239-245,377):
                # one flat all-reduce per real step
                if dist.get_backend() == "gloo":
                    dist.all_reduce(flat.flat_grad, op=dist.ReduceOp.SUM)
                    flat.flat_grad.div_(world_size)
                else:
                    dist.all_reduce(flat.flat_grad, op=dist.ReduceOp.AVG)

            scaler.unscale_(optimizer=inner_opt_for_ckpt)
            if device.type == "cuda":
                clip_grad_norm_flat_(flat.flat_grad, 1.0)  # train_fsdp.py:395
            else:
                torch.nn.utils.clip_grad_norm_(model.parameters(), 1.0)

            if hv:
                try:
                    optimizer.step(scaler=scaler)
                except RuntimeError as e:
                    # a dead peer surfaces as a failed collective on the fixed
                    # RCCL/gloo world; map it to the reference's rank-drop
                    # error shape (train_fsdp.py:452-457).  The reference can
                    # keep training without the lost peer (elastic DHT swarm);
                    # a fixed communicator cannot — fail fast either way,
                    # with the reference's message when fail_rank_drop is set.
                    num_peers = max(optimizer.tracker.global_progress.num_peers - 1, 0)
                    optimizer.tracker.global_progress.num_peers = num_peers
                    log(f"Lost a diloco worker, num_peers: {num_peers}, "
                        f"galaxy_size: {config.hv.galaxy_size}")
                    if config.hv.fail_rank_drop:
                        raise ValueError(
                            f"Lost a diloco worker, num_peers: {num_peers}, "
                            f"galaxy_size: {config.hv.galaxy_size}") from e
                    raise
            else:
                scaler.step(optimizer)
            scaler.update()
            scheduler.step()
            optimizer.zero_grad()

            # (reference broadcasts params across the worker's own GPUs every
            # H steps, train_fsdp.py:410-413; a worker is exactly 1 GPU here)

            if rank == 0:
                total_samples = real_step * config.total_batch_size
                effective_step = real_step
                if hv:
                    # reference uses config.hv.galaxy_size as-is
                    # (train_fsdp.py:418-423; "not robust to off/on ramping")
                    effective_step = real_step * config.hv.galaxy_size
                    total_samples = real_step * config.total_batch_size * config.hv.galaxy_size
                metrics = {
                    "Loss": loss_batch.item(),
                    "step": real_step,
                    "lr": [g["lr"] for g in optimizer.param_groups][0],
                    "Perplexity": torch.exp(loss_batch).item(),
                    "effective_step": effective_step,
                    "total_samples": total_samples,
                    "time_taken": time.time() - current_time,
                    "tokens_per_second": config.seq_length * config.total_batch_size
                                         / (time.time() - current_time),
                }
                if hv:
                    metrics["outer_lr"] = optimizer.state_averager.optimizer.param_groups[0]["lr"]
                    num_peers = optimizer.tracker.global_progress.num_peers
                    max_num_peers = max(max_num_peers, num_peers)
                    metrics["num_peers"] = num_peers
                    if num_peers < max_num_peers and config.hv.fail_rank_drop:
                        # reference train_fsdp.py:455-457 message shape
                        raise ValueError(
                            f"Lost a diloco worker, num_peers: {num_peers}, "
                            f"galaxy_size: {config.hv.galaxy_size}")
                if logging_activations_steps:
                    metrics.update({k: float(v) for k, v in log_activations.items()})
                    log_activations = {}
                current_time = time.time()
                metric_logger.log(metrics)
                if not hv:
                    log(f"step: {real_step}, loss: {loss_batch.item()}, lr "
                        f"{[g['lr'] for g in optimizer.param_groups][0]}")

            if config.ckpt.interval is not None and real_step % config.ckpt.interval == 0:
                log(f"saving at step {real_step}, step {step + 1}")
                ckpt_path = os.path.join(config.ckpt.path, f"{CKPT_PREFIX}_{int(real_step)}")
                if hv:
                    ckpt_path = os.path.join(ckpt_path, get_diloco_rank_dir_name(rank))
                    with optimizer.tracker.pause_updates():
                        save_checkpoint(
                            checkpoint_path=ckpt_path, model=model,
                            optimizer=optimizer.inner_optimizer, scheduler=scheduler,
                            outer_optimizer=optimizer.state_averager.optimizer,
                            loss=loss_batch.item(), scaler=scaler,
                            data_loader=train_dataloader, save_global_state=True, rank=rank)
                else:
                    # model/optimizer states are replicated across ranks after
                    # the grad all-reduce: only rank 0 writes model_optim.pt
                    # (concurrent same-path writes could corrupt it); every
                    # rank still writes its own dataloader state file
                    save_checkpoint(
                        checkpoint_path=ckpt_path, model=model, optimizer=optimizer,
                        scheduler=scheduler, loss=loss_batch.item(), scaler=scaler,
                        data_loader=train_dataloader, save_global_state=(rank == 0),
                        save_model_optim=(rank == 0), rank=rank)
                    if world_size > 1:
                        dist.barrier()
                if local_rank == 0 and config.ckpt.topk is not None:
                    deleted = delete_old_checkpoints(config.ckpt.path, config.ckpt.topk)
                    if deleted:
                        log(f"Deleted old checkpoints: {deleted}")

            loss_batch = torch.zeros((), device=device)

            # test-only fault injection for the straggler harness
            # (tests/test_train_cli.py::test_rank_drop_fails_fast): simulate a
            # worker dying without a clean process-group shutdown
            die = os.environ.get("DILOCO_TEST_DIE_RANK_STEP")
            if die:
                die_rank, die_step = (int(x) for x in die.split(":"))
                if rank == die_rank and real_step >= die_step:
                    log(f"test fault injection: rank {rank} exiting at step {real_step}")
                    os._exit(0)

            if config.max_steps is not None and real_step >= config.max_steps:
                break

    log("Training completed.")
    if rank == 0:
        metric_logger.finish()


def main():
    ddp_setup()
    config = Config(**parse_argv())
    train(config)
    dist.destroy_process_group()


if __name__ == "__main__":
    main()
