"""Flat parameter space + fused optimizers for the DiLoCo hot path.

The reference runs torch AdamW per-parameter (hivemind_diloco.py:546-550,
train_diloco_torch.py:186,325) and torch SGD-Nesterov on offloaded CPU
parameters (train_diloco_torch.py:187,349).  Here all parameters live as
views into ONE flat fp32 buffer (288 GB HBM3E: keep everything resident),
grads accumulate into a flat fp32 buffer, and the whole AdamW step is one
HBM-bound HIP kernel launch (dk_fused_adamw).  The outer Nesterov step +
pseudo-gradient + copy-back are one fused kernel as well (dk_outer_nesterov),
and the cross-worker all-reduce is a single flat RCCL call instead of the
reference's per-parameter tensors (train_diloco_torch.py:345).

On CPU the same classes run the identical torch op sequence (mul_/lerp_/
addcmul_/addcdiv_ — the exact single-tensor torch.optim.AdamW order) so the
gloo multi-process tests and the reference CPU path stay bit-faithful.
"""

from __future__ import annotations

import math

import torch


class FlatSpace:
    """Re-homes a parameter list into one flat fp32 buffer (+ flat grads)."""

    def __init__(self, params: list[torch.nn.Parameter]):
        assert len(params) > 0
        self.device = params[0].device
        assert all(p.dtype == torch.float32 for p in params), "master params must be fp32"
        self.params = params
        self.numels = [p.numel() for p in params]
        self.shapes = [p.shape for p in params]
        self.n = sum(self.numels)
        self.flat_param = torch.empty(self.n, dtype=torch.float32, device=self.device)
        self.flat_grad = torch.zeros(self.n, dtype=torch.float32, device=self.device)
        off = 0
        for p, n in zip(params, self.numels):
            self.flat_param[off:off + n].copy_(p.data.reshape(-1))
            p.data = self.flat_param[off:off + n].view(p.shape)
            p.grad = self.flat_grad[off:off + n].view(p.shape)
            off += n

    def grad_views(self):
        off = 0
        for n, s in zip(self.numels, self.shapes):
            yield self.flat_grad[off:off + n].view(s)
            off += n

    def zero_grad(self):
        self.flat_grad.zero_()

    def relink(self):
        """Re-point p.data/p.grad at the flat buffers (after an external
        load_state_dict replaced them)."""
        off = 0
        for p, n, s in zip(self.params, self.numels, self.shapes):
            if p.data.data_ptr() != self.flat_param[off:off + n].data_ptr():
                self.flat_param[off:off + n].copy_(p.data.reshape(-1))
                p.data = self.flat_param[off:off + n].view(s)
            p.grad = self.flat_grad[off:off + n].view(s)
            off += n


class FusedAdamW(torch.optim.Optimizer):
    """AdamW with flat fp32 state; one HIP kernel per step on GPU.

    Matches torch.optim.AdamW single-tensor math (the reference's inner
    optimizer, hyperparams at train_fsdp.py:250: lr cfg, wd=0.1,
    betas=(0.9,0.95)).  state_dict() is per-parameter like torch's, with
    exp_avg/exp_avg_sq exposed as views into the flat buffers.
    """

    def __init__(self, params, lr=1e-3, betas=(0.9, 0.999), eps=1e-8, weight_decay=1e-2,
                 flat: FlatSpace | None = None):
        params = list(params)
        defaults = dict(lr=lr, betas=betas, eps=eps, weight_decay=weight_decay)
        super().__init__(params, defaults)
        assert len(self.param_groups) == 1, "single param group supported"
        self.flat = flat if flat is not None else FlatSpace(self.param_groups[0]["params"])
        self.flat_m = torch.zeros(self.flat.n, dtype=torch.float32, device=self.flat.device)
        self.flat_v = torch.zeros(self.flat.n, dtype=torch.float32, device=self.flat.device)
        self._step_count_t = 0
        self._link_state()

    def _link_state(self):
        off = 0
        for p, n, s in zip(self.flat.params, self.flat.numels, self.flat.shapes):
            self.state[p] = {
                "step": torch.tensor(float(self._step_count_t)),
                "exp_avg": self.flat_m[off:off + n].view(s),
                "exp_avg_sq": self.flat_v[off:off + n].view(s),
            }
            off += n

    def zero_grad(self, set_to_none: bool = True):  # noqa: ARG002
        # grads are views into the flat buffer; never drop them
        self.flat.zero_grad()

    @torch.no_grad()
    def step(self, closure=None):
        loss = None
        if closure is not None:
            with torch.enable_grad():
                loss = closure()
        g = self.param_groups[0]
        lr, (b1, b2), eps, wd = g["lr"], g["betas"], g["eps"], g["weight_decay"]
        self._step_count_t += 1
        t = self._step_count_t
        for st in self.state.values():
            st["step"] += 1
        from opendiloco_amd.model import bump_weights_version

        bump_weights_version()  # invalidate cached bf16 weight copies
        if self.flat.device.type == "cuda":
            from opendiloco_amd.ops import _ext

            _ext().fused_adamw(self.flat.flat_param, self.flat.flat_grad, self.flat_m,
                               self.flat_v, lr, b1, b2, eps, wd, t)
        else:
            # exact torch.optim.AdamW single-tensor op order
            p, gr, m, v = self.flat.flat_param, self.flat.flat_grad, self.flat_m, self.flat_v
            p.mul_(1 - lr * wd)
            m.lerp_(gr, 1 - b1)
            v.mul_(b2).addcmul_(gr, gr, value=1 - b2)
            bc1 = 1 - b1 ** t
            bc2_sqrt = math.sqrt(1 - b2 ** t)
            denom = (v.sqrt() / bc2_sqrt).add_(eps)
            p.addcdiv_(m, denom, value=-(lr / bc1))
        return loss

    def load_state_dict(self, state_dict):
        super().load_state_dict(state_dict)
        # torch replaced our view tensors; copy values back into flat storage
        off = 0
        max_step = 0.0
        for p, n, s in zip(self.flat.params, self.flat.numels, self.flat.shapes):
            st = self.state[p]
            self.flat_m[off:off + n].copy_(st["exp_avg"].reshape(-1))
            self.flat_v[off:off + n].copy_(st["exp_avg_sq"].reshape(-1))
            max_step = max(max_step, float(st["step"]))
            off += n
        self._step_count_t = int(max_step)
        self._link_state()
        self.flat.relink()


class FlatSGDNesterov:
    """Outer optimizer state container (torch.optim.SGD-compatible surface)
    over the flat theta_outer buffer; the actual update on GPU is the fused
    dk_outer_nesterov kernel (pseudo-grad + momentum + copy-back in one pass).

    Mirrors torch SGD(lr, momentum=0.9, nesterov=True) math exactly
    (reference train_fsdp.py:253, train_diloco_torch.py:187):
        buf = g                 (first step)
        buf = mu*buf + g        (later)
        p  -= lr * (g + mu*buf)
    """

    def __init__(self, flat_outer: torch.Tensor, lr: float, momentum: float = 0.9,
                 nesterov: bool = True, weight_decay: float = 0.0, dampening: float = 0.0):
        assert nesterov and weight_decay == 0.0 and dampening == 0.0, \
            "reference outer optimizer is SGD(momentum, nesterov) only"
        self.flat_outer = flat_outer
        self.momentum_buf: torch.Tensor | None = None
        self.lr = lr
        self.momentum = momentum
        self.param_groups = [{
            "params": [flat_outer], "lr": lr, "momentum": momentum, "dampening": 0.0,
            "weight_decay": 0.0, "nesterov": True, "maximize": False, "foreach": None,
            "differentiable": False, "fused": None,
        }]

    def step_fused(self, flat_local: torch.Tensor, g_avg: torch.Tensor):
        """theta_outer/momentum update + theta_local copy-back."""
        from opendiloco_amd.model import bump_weights_version

        bump_weights_version()
        self.lr = self.param_groups[0]["lr"]
        first = self.momentum_buf is None
        if first:
            self.momentum_buf = torch.empty_like(self.flat_outer)
        if self.flat_outer.device.type == "cuda":
            from opendiloco_amd.ops import _ext

            _ext().outer_nesterov(self.flat_outer, flat_local, self.momentum_buf, g_avg,
                                  self.lr, self.momentum, first)
        else:
            buf = self.momentum_buf
            if first:
                buf.copy_(g_avg)
            else:
                buf.mul_(self.momentum).add_(g_avg)
            d = g_avg.add(buf, alpha=self.momentum)
            self.flat_outer.add_(d, alpha=-self.lr)
            flat_local.copy_(self.flat_outer)

    def zero_grad(self, set_to_none: bool = True):
        pass  # pseudo-grads live in a caller-owned buffer

    # -- torch-style state dict (single flat param, id 0) --
    def state_dict(self) -> dict:
        state = {}
        if self.momentum_buf is not None:
            state[0] = {"momentum_buffer": self.momentum_buf}
        groups = [{k: v for k, v in self.param_groups[0].items() if k != "params"}]
        groups[0]["params"] = [0]
        return {"state": state, "param_groups": groups}

    def load_state_dict(self, sd: dict) -> None:
        groups = sd.get("param_groups", [])
        if groups:
            for k in ("lr", "momentum"):
                if k in groups[0]:
                    self.param_groups[0][k] = groups[0][k]
            self.lr = self.param_groups[0]["lr"]
            self.momentum = self.param_groups[0]["momentum"]
        state = sd.get("state", {})
        buf = None
        for st in state.values():
            if isinstance(st, dict) and "momentum_buffer" in st and st["momentum_buffer"] is not None:
                b = st["momentum_buffer"].reshape(-1).to(self.flat_outer.device)
                buf = b if buf is None else torch.cat([buf, b])
        if buf is not None:
            assert buf.numel() == self.flat_outer.numel(), \
                (buf.numel(), self.flat_outer.numel())
            self.momentum_buf = buf.clone()


def clip_grad_norm_flat_(flat_grad: torch.Tensor, max_norm: float) -> torch.Tensor:
    """Global L2 clip over the flat grad buffer (replaces clip_grad_norm_(1.0)
    at train_fsdp.py:395 / train_diloco_torch.py:323).  GPU: deterministic
    two-pass HIP kernels; CPU: torch."""
    if flat_grad.device.type == "cuda":
        from opendiloco_amd.ops import _ext, dw_stream_sync

        dw_stream_sync()  # side-stream dW accumulation must land first
        out2 = _ext().clip_grad_(flat_grad, max_norm)
        return out2[0]
    total = flat_grad.norm(2)
    coef = max_norm / (total + 1e-6)
    if coef < 1:
        flat_grad.mul_(coef)
    return total
