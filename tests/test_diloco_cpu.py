"""CPU tests of DiLoCoOptimizer: API surface, state round-trip (mirrors
reference tests/test_diloco_hivemind.py::test_load_and_save_state), and
2-process gloo all-reduce semantics (mirrors
test_allreduce_dilco_grad_averager, with a real numeric pin instead of the
reference's non-NaN check)."""

import copy
import os
from functools import partial

import pytest
import torch
import torch.multiprocessing as mp

from opendiloco_amd.diloco import AllReduceStrategy, DiLoCoOptimizer


def _make_opt(seed=0, H=2, lr=0.1):
    torch.manual_seed(seed)
    model = torch.nn.Linear(5, 1)
    opt = DiLoCoOptimizer(
        dht=None, run_id="test", batch_size=32, num_inner_steps=H,
        outer_optimizer=partial(torch.optim.SGD, lr=0.7, momentum=0.9, nesterov=True),
        inner_optimizer=partial(torch.optim.AdamW, lr=lr, weight_decay=0.1, betas=(0.9, 0.95)),
        params=model.parameters(),
    )
    return model, opt


def _train_steps(model, opt, n, seed=10):
    torch.manual_seed(seed)
    losses = []
    for _ in range(n):
        x = torch.randn(32, 5)
        y = x @ torch.arange(5, dtype=torch.float32).unsqueeze(1)
        loss = torch.nn.functional.mse_loss(model(x), y)
        opt.zero_grad()
        loss.backward()
        opt.step()
        losses.append(loss.item())
    return losses


def test_step_and_state_dict_roundtrip():
    """reference test_load_and_save_state (test_diloco_hivemind.py:99-151):
    2 steps of AdamW-inner/SGD-outer on a 5->1 linear regression, exact
    state_dict round-trip incl. inner and outer optimizer states."""
    model, opt = _make_opt(H=2)
    _train_steps(model, opt, 2)
    assert opt.local_epoch == 1  # outer fired after H=2 steps
    sd = copy.deepcopy(opt.state_dict())
    assert set(sd.keys()) == {"state_dict_outer", "state_dict_inner"}
    assert sd["state_dict_outer"]["state"]["local_epoch"] == 1

    model2, opt2 = _make_opt(seed=99, H=2)
    with torch.no_grad():
        for p, q in zip(model.parameters(), model2.parameters()):
            q.copy_(p)
    opt2.load_state_dict(copy.deepcopy(sd))
    assert opt2.local_epoch == 1
    assert torch.equal(opt.inner_optimizer.flat_m, opt2.inner_optimizer.flat_m)
    assert torch.equal(opt.inner_optimizer.flat_v, opt2.inner_optimizer.flat_v)
    assert torch.allclose(opt.state_averager.optimizer.momentum_buf,
                          opt2.state_averager.optimizer.momentum_buf)
    # identical continuation
    l1 = _train_steps(model, opt, 2, seed=20)
    l2 = _train_steps(model2, opt2, 2, seed=20)
    assert l1 == l2


def test_pseudo_grad_math():
    model, opt = _make_opt(H=100)  # outer never fires
    _train_steps(model, opt, 1)
    ga = opt.diloco_grad_averager
    ga.compute_and_load_pseudo_grad_into_averager()
    expect = opt.state_averager.flat_outer - opt.flat.flat_param
    assert torch.equal(ga.pseudo_grad, expect)
    assert not torch.isnan(ga.pseudo_grad).any()
    assert ga.pseudo_grad.abs().sum() > 0


def test_api_surface_matches_reference():
    """Attributes/methods the reference CLI + tests use
    (SURVEY.md §8b inventory)."""
    model, opt = _make_opt()
    assert opt.param_groups is opt.inner_optimizer.param_groups
    assert hasattr(opt.state_averager, "optimizer")
    assert hasattr(opt.state_averager.optimizer, "param_groups")
    assert opt.tracker.global_progress.num_peers == 1
    with opt.tracker.pause_updates():
        pass
    assert opt.local_epoch == 0
    opt.load_state_from_peers()  # no-op at world 1
    opt.update_main_param_after_outer_step()
    assert opt.all_reduce_strategy == AllReduceStrategy.WAIT_FOR_ALL


def test_rejects_reference_forbidden_kwargs():
    with pytest.raises(KeyError):
        _, _ = torch.nn.Linear(2, 1), DiLoCoOptimizer(
            batch_size=1, num_inner_steps=1,
            outer_optimizer=partial(torch.optim.SGD, lr=0.7),
            inner_optimizer=partial(torch.optim.AdamW, lr=1e-3),
            params=torch.nn.Linear(2, 1).parameters(),
            optimizer="nope")
    with pytest.raises(ValueError):
        DiLoCoOptimizer(
            batch_size=1, num_inner_steps=1,
            outer_optimizer=partial(torch.optim.SGD, lr=0.7),
            inner_optimizer=partial(torch.optim.AdamW, lr=1e-3),
            params=torch.nn.Linear(2, 1).parameters(),
            all_reduce_strategy=AllReduceStrategy.NO_WAIT,
            timeout_waiting_for_peers=10.0)


# ---------- 2-process gloo: the cross-worker average ----------

def _worker(rank, world, port, outdir, compression=None):
    os.environ.update(dict(RANK=str(rank), WORLD_SIZE=str(world), LOCAL_RANK=str(rank),
                           MASTER_ADDR="127.0.0.1", MASTER_PORT=str(port)))
    import torch.distributed as dist

    dist.init_process_group("gloo", rank=rank, world_size=world)
    try:
        if compression is None:
            model, opt = _make_opt(seed=0, H=2)  # same init on both ranks
        else:
            torch.manual_seed(0)
            model = torch.nn.Linear(5, 1)
            opt = DiLoCoOptimizer(
                dht=None, run_id="test", batch_size=32, num_inner_steps=2,
                outer_optimizer=partial(torch.optim.SGD, lr=0.7, momentum=0.9,
                                        nesterov=True),
                inner_optimizer=partial(torch.optim.AdamW, lr=0.1, weight_decay=0.1,
                                        betas=(0.9, 0.95)),
                params=model.parameters(), grad_compression=compression)
            want = {"fp16": torch.float16, "uniform8bit": torch.uint8}[compression]
            assert opt.diloco_grad_averager.comm_dtype == want
        _train_steps(model, opt, 2, seed=50 + rank)  # different data per rank
        # after the outer round every worker must hold identical params
        torch.save({"flat": opt.flat.flat_param.clone(), "epoch": opt.local_epoch},
                   os.path.join(outdir, f"out_{rank}.pt"))
    finally:
        dist.destroy_process_group()


def test_two_worker_average_converges_to_same_params(tmp_path):
    import socket

    with socket.socket() as s:
        s.bind(("", 0))
        port = s.getsockname()[1]
    ctx = mp.get_context("spawn")
    procs = [ctx.Process(target=_worker, args=(r, 2, port, str(tmp_path))) for r in range(2)]
    for p in procs:
        p.start()
    for p in procs:
        p.join(180)
        assert p.exitcode == 0
    r0 = torch.load(tmp_path / "out_0.pt", weights_only=False)
    r1 = torch.load(tmp_path / "out_1.pt", weights_only=False)
    assert r0["epoch"] == 1 and r1["epoch"] == 1
    assert torch.allclose(r0["flat"], r1["flat"], atol=1e-7)


def test_two_worker_fp16_compression(tmp_path):
    """`--hv.hivemind_compression fp16` twin (reference Float16Compression,
    utils.py:83-121): the pseudo-gradient all-reduce payload is fp16; both
    workers must still land on identical finite params."""
    import socket

    with socket.socket() as s:
        s.bind(("", 0))
        port = s.getsockname()[1]
    ctx = mp.get_context("spawn")
    procs = [ctx.Process(target=_worker, args=(r, 2, port, str(tmp_path), "fp16"))
             for r in range(2)]
    for p in procs:
        p.start()
    for p in procs:
        p.join(180)
        assert p.exitcode == 0
    r0 = torch.load(tmp_path / "out_0.pt", weights_only=False)
    r1 = torch.load(tmp_path / "out_1.pt", weights_only=False)
    assert r0["epoch"] == 1 and r1["epoch"] == 1
    assert torch.equal(r0["flat"], r1["flat"])
    assert torch.isfinite(r0["flat"]).all()


def test_two_worker_uniform8bit_compression(tmp_path):
    """`--hv.hivemind_compression uniform8bit` twin (reference
    Uniform8BitQuantization via utils.py:103-107, restated on the flat
    payload): each peer's pseudo-gradient travels as uint8 (256 uniform
    levels over mean +- 3 sigma) + (offset, scale); peers de-quantize and
    mean in fp32.  Both workers must land on identical finite params
    (the reference's own pin for 8-bit codecs is only non-NaN averages,
    test_diloco_hivemind.py:90-93)."""
    import socket

    with socket.socket() as s:
        s.bind(("", 0))
        port = s.getsockname()[1]
    ctx = mp.get_context("spawn")
    procs = [ctx.Process(target=_worker, args=(r, 2, port, str(tmp_path), "uniform8bit"))
             for r in range(2)]
    for p in procs:
        p.start()
    for p in procs:
        p.join(180)
        assert p.exitcode == 0
    r0 = torch.load(tmp_path / "out_0.pt", weights_only=False)
    r1 = torch.load(tmp_path / "out_1.pt", weights_only=False)
    assert r0["epoch"] == 1 and r1["epoch"] == 1
    assert torch.equal(r0["flat"], r1["flat"])
    assert torch.isfinite(r0["flat"]).all()


def test_uniform8bit_quantization_roundtrip():
    """Codec-level pin: quantize->dequantize of a payload is within one
    quantization step (6 sigma / 255) of the original for values inside the
    +-3 sigma range, and the 1-worker path leaves the buffer untouched."""
    from opendiloco_amd.diloco import DiLoCoGradAverager

    torch.manual_seed(0)
    buf = torch.randn(10_000)
    offset = buf.mean()
    scale = DiLoCoGradAverager.RANGE_IN_SIGMAS * buf.std() / 255.0
    q = torch.clamp(torch.round((buf - offset) / scale) + 128.0, 0.0, 255.0).to(torch.uint8)
    deq = q.to(torch.float32).sub_(128.0).mul_(scale).add_(offset)
    inside = (buf - offset).abs() <= 127.0 * scale
    assert inside.float().mean() > 0.99
    assert (deq[inside] - buf[inside]).abs().max() <= scale * 0.5 + 1e-7


# ---------- 2-process gloo vs oracle golden: BOTH workers pinned ----------

def _golden_worker(rank, world, port, outdir, fixture, golden_path):
    os.environ.update(dict(RANK=str(rank), WORLD_SIZE=str(world), LOCAL_RANK=str(rank),
                           MASTER_ADDR="127.0.0.1", MASTER_PORT=str(port)))
    import json

    import torch.distributed as dist

    from opendiloco_amd.data import FakeTokenizedDataLoader
    from opendiloco_amd.model import LlamaForCausalLM
    from opendiloco_amd.schedule import get_cosine_schedule_with_warmup

    dist.init_process_group("gloo", rank=rank, world_size=world)
    try:
        with open(golden_path) as f:
            cfg = json.load(f)["config"]
        model = LlamaForCausalLM.from_pretrained(fixture).train()
        opt = DiLoCoOptimizer(
            batch_size=cfg["batch_size"], num_inner_steps=cfg["local_steps"],
            outer_optimizer=partial(torch.optim.SGD, lr=cfg["outer_lr"], momentum=0.9,
                                    nesterov=True),
            inner_optimizer=partial(torch.optim.AdamW, lr=cfg["lr"], weight_decay=0.1,
                                    betas=(0.9, 0.95)),
            params=model.parameters())
        sched = get_cosine_schedule_with_warmup(opt.inner_optimizer, cfg["warmup_steps"],
                                                cfg["total_steps"])
        loader = iter(FakeTokenizedDataLoader(cfg["seq_length"], cfg["vocab_size"],
                                              cfg["per_device_train_batch_size"],
                                              cfg["seed"], rank))
        grad_acc = cfg["batch_size"] // cfg["per_device_train_batch_size"]
        losses = []
        for _ in range(cfg["max_steps"]):
            acc = 0.0
            for _ in range(grad_acc):
                batch = next(loader)
                loss = model(**batch).loss / grad_acc
                acc += loss.item()
                loss.backward()
            torch.nn.utils.clip_grad_norm_(model.parameters(), 1.0)
            opt.step()
            sched.step()
            opt.zero_grad()
            losses.append(acc)
        torch.save({"losses": losses, "epoch": opt.local_epoch},
                   os.path.join(outdir, f"golden_out_{rank}.pt"))
    finally:
        dist.destroy_process_group()


def test_two_worker_matches_golden_both_ranks(tmp_path, fixture_2m, golden_dir):
    """Closes the 'worker 0 only' gap: the product stack on 2 gloo ranks must
    reproduce the oracle golden llama2m_w2_h3 per-step losses for BOTH
    workers (the CLI/DummyLogger harness only ever sees rank 0's trace)."""
    import json
    import socket

    golden_path = os.path.join(golden_dir, "llama2m_w2_h3.json")
    with socket.socket() as s:
        s.bind(("", 0))
        port = s.getsockname()[1]
    ctx = mp.get_context("spawn")
    procs = [ctx.Process(target=_golden_worker,
                         args=(r, 2, port, str(tmp_path), fixture_2m, golden_path))
             for r in range(2)]
    for p in procs:
        p.start()
    for p in procs:
        p.join(600)
        assert p.exitcode == 0
    with open(golden_path) as f:
        golden = json.load(f)
    outs = [torch.load(tmp_path / f"golden_out_{r}.pt", weights_only=False) for r in range(2)]
    assert outs[0]["epoch"] == golden["outer_steps"]
    for i, rec in enumerate(golden["records"]):
        for r in range(2):
            got = outs[r]["losses"][i]
            want = rec["losses"][r]
            assert abs(got - want) < 1e-3, f"step {rec['step']} worker {r}: {got} vs {want}"
