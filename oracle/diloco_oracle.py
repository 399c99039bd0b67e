"""CPU oracle: restatement of the reference's pure-torch DiLoCo loop.

Follows open_diloco/train_diloco_torch.py (PrimeIntellect-ai/OpenDiloco @
2024-10-08) line by line, with these mechanical substitutions only:

  - the N workers run as N model replicas inside ONE process; the
    cross-worker ``dist.all_reduce(tensor=param.grad, op=ReduceOp.AVG)``
    (train_diloco_torch.py:345) becomes an explicit mean over the replicas'
    pseudo-gradients (identical arithmetic for ReduceOp.AVG);
  - wandb/eval/activation-logging side paths (:284-307, :329-332, :355-407)
    and checkpointing are dropped (the file itself cannot be imported here:
    stale import at :24, and hivemind/cyclopts/wandb are absent);
  - data is the seeded fake-token stream (utils.py:163-167 semantics:
    ``randint(3, vocab)`` ids, all-ones attention mask, labels = input_ids
    via the mlm=False LM collator, train_fsdp.py:161), generator seed
    ``seed + 1337 * worker_rank`` — the same scheme the product CLI uses so
    traces are directly comparable.

The model is ``transformers.LlamaForCausalLM`` — the exact module the
reference calls (train_fsdp.py:171-174, train_diloco_torch.py:183).

This file is TEST INFRASTRUCTURE (see oracle/__init__.py): only tests/,
__graft_entry__.smoke() and bench.py's cpu_baseline leg may use it.
"""

from __future__ import annotations

import copy
import hashlib
from dataclasses import dataclass, field

import torch


@dataclass
class OracleConfig:
    model_path: str = ""                      # HF dir with config.json (+ optional safetensors)
    n_workers: int = 1
    local_steps: int = 5                      # H (train_diloco_torch.py:159 `local_steps`)
    batch_size: int = 16                      # per-worker batch per real step (:143)
    per_device_train_batch_size: int = 8      # micro-batch (:144)
    seq_length: int = 128                     # (:145)
    vocab_size: int = 1024                    # fake-data vocab (train_fsdp.py:66 TEST_VOCAB_SIZE)
    lr: float = 4e-4                          # inner AdamW lr (:154)
    outer_lr: float = 0.7                     # outer SGD lr (:162)
    warmup_steps: int = 1000                  # (:149)
    total_steps: int = 88_000                 # (:150)
    max_steps: int = 10                       # how many real steps to run
    seed: int = 42
    max_grad_norm: float = 1.0                # clip (:323)
    record_param_hash_every: int = 0          # 0 = only at end
    fresh_init_seed: int | None = None        # if set, random-init instead of from_pretrained


def _fake_batch(gen: torch.Generator, bs: int, seq_len: int, vocab: int) -> dict[str, torch.Tensor]:
    # FakeTokenizedDataset semantics (open_diloco/utils.py:163-167) + mlm=False collation
    ids = torch.randint(3, vocab, (bs, seq_len), generator=gen, dtype=torch.int64)
    return {"input_ids": ids, "attention_mask": torch.ones_like(ids), "labels": ids.clone()}


def _round_flatten(a: torch.Tensor, max_size: int = 1000) -> str:
    # restates open_diloco/utils.py:70-77
    bounds = int(max_size**0.5)
    a2 = a.reshape(a.shape[0], -1) if a.dim() > 1 else a.reshape(1, -1)
    return ",".join(f"{float(i):.4f}" for i, _ in zip(a2[:bounds, :bounds].flatten(), range(max_size)))


def hash_tensor_content(a: torch.Tensor, max_size: int = 1000) -> str:
    # restates open_diloco/utils.py:79-80
    return hashlib.md5(_round_flatten(a.detach(), max_size=max_size).encode("utf-8")).hexdigest()


def model_digest(model: torch.nn.Module) -> str:
    h = hashlib.md5()
    for name, p in sorted(model.named_parameters()):
        h.update(name.encode())
        h.update(hash_tensor_content(p).encode())
    return h.hexdigest()


def get_cosine_schedule_with_warmup(optimizer, num_warmup_steps: int, num_training_steps: int):
    """Restates transformers.get_cosine_schedule_with_warmup (used by the
    reference at train_diloco_torch.py:189-193), num_cycles=0.5.  Written
    independently of the product's opendiloco_amd/schedule.py on purpose —
    a CPU test checks both against transformers' own implementation."""
    import math

    def lr_lambda(current_step: int) -> float:
        if current_step < num_warmup_steps:
            return float(current_step) / float(max(1, num_warmup_steps))
        progress = float(current_step - num_warmup_steps) / float(max(1, num_training_steps - num_warmup_steps))
        return max(0.0, 0.5 * (1.0 + math.cos(math.pi * 1.0 * progress)))

    return torch.optim.lr_scheduler.LambdaLR(optimizer, lr_lambda, -1)


def make_reference_model(cfg: OracleConfig):
    """The reference's model: transformers LlamaForCausalLM on CPU, fp32."""
    from transformers import LlamaConfig, LlamaForCausalLM

    if cfg.fresh_init_seed is not None:
        torch.manual_seed(cfg.fresh_init_seed)
        lcfg = LlamaConfig.from_pretrained(cfg.model_path)
        lcfg.use_cache = False
        model = LlamaForCausalLM(lcfg)
    else:
        model = LlamaForCausalLM.from_pretrained(cfg.model_path)
        model.config.use_cache = False
    return model.float()


def run_diloco_oracle(cfg: OracleConfig, model_factory=None, progress=None) -> dict:
    """Run the restated DiLoCo loop; return per-step records and digests.

    Returns {"records": [{"step", "lr", "losses": [per-worker]} ...],
             "final_digest": [per-worker md5], "outer_steps": int}
    """
    torch.manual_seed(cfg.seed)
    assert cfg.batch_size % cfg.per_device_train_batch_size == 0
    grad_acc = cfg.batch_size // cfg.per_device_train_batch_size  # :172

    base_model = model_factory(cfg) if model_factory is not None else make_reference_model(cfg)
    base_model = base_model.train()

    workers = []
    for r in range(cfg.n_workers):
        # dist.broadcast(param, src=0) at :253-255 == identical replicas
        model = copy.deepcopy(base_model)
        inner = torch.optim.AdamW(model.parameters(), lr=cfg.lr, weight_decay=0.1, betas=(0.9, 0.95))  # :186
        outer = torch.optim.SGD(model.parameters(), lr=cfg.outer_lr, momentum=0.9, nesterov=True)  # :187
        sched = get_cosine_schedule_with_warmup(inner, cfg.warmup_steps, cfg.total_steps)  # :189-193
        # get_offloaded_param (:132-135, :257): snapshot of theta_outer
        offloaded = [p.data.detach().clone() for g in outer.param_groups for p in g["params"]]
        gen = torch.Generator().manual_seed(cfg.seed + 1337 * r)
        workers.append(dict(model=model, inner=inner, outer=outer, sched=sched,
                            offloaded=offloaded, gen=gen, loss_batch=torch.zeros(())))
    del base_model

    records = []
    outer_steps_done = 0

    for real_step in range(1, cfg.max_steps + 1):
        for w in workers:
            w["loss_batch"] = torch.zeros(())
            for _micro in range(grad_acc):  # inner loop :272-318 (fp32, no autocast/scaler)
                batch = _fake_batch(w["gen"], cfg.per_device_train_batch_size, cfg.seq_length, cfg.vocab_size)
                out = w["model"](**batch)
                loss = out.loss / grad_acc  # :314
                w["loss_batch"] = w["loss_batch"] + loss.detach()  # :316
                loss.backward()  # :318
            torch.nn.utils.clip_grad_norm_(w["model"].parameters(), cfg.max_grad_norm)  # :323
            w["inner"].step()  # :325
            w["sched"].step()  # :327
            w["inner"].zero_grad()  # :334

        if real_step % cfg.local_steps == 0:  # outer block :336-353
            # per worker: pseudo-grad = theta_offloaded - theta_local (:342-344)
            all_grads = []
            for w in workers:
                main_param = [p for g in w["inner"].param_groups for p in g["params"]]  # :340
                grads = [off.data - p.data for off, p in zip(w["offloaded"], main_param)]
                all_grads.append(grads)
            # dist.all_reduce(AVG) (:345) == mean over workers, applied to every worker
            n = float(cfg.n_workers)
            mean_grads = [sum(g[i] for g in all_grads) / n for i in range(len(all_grads[0]))]
            for w in workers:
                main_param = [p for g in w["inner"].param_groups for p in g["params"]]
                for p, off, mg in zip(main_param, w["offloaded"], mean_grads):
                    p.grad = mg.clone()
                    p.data = off.data.clone()  # :346 (restore theta_outer)
                w["outer"].step()  # :349
                w["outer"].zero_grad()  # :351
                w["offloaded"] = [p.data.detach().clone() for g in w["outer"].param_groups
                                  for p in g["params"]]  # :353
            outer_steps_done += 1

        rec = {
            "step": real_step,
            "lr": [g["lr"] for g in workers[0]["inner"].param_groups][0],  # :371 (logged after sched.step)
            "losses": [float(w["loss_batch"]) for w in workers],
        }
        if cfg.record_param_hash_every and real_step % cfg.record_param_hash_every == 0:
            rec["digests"] = [model_digest(w["model"]) for w in workers]
        records.append(rec)
        if progress is not None:
            progress(rec)

    return {
        "records": records,
        "final_digest": [model_digest(w["model"]) for w in workers],
        "outer_steps": outer_steps_done,
    }
