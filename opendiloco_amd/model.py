"""MI355X-native Llama for causal LM.

Same architecture, parameter names and initialisation as the
``transformers.LlamaForCausalLM`` the reference trains
(open_diloco/train_fsdp.py:171-174, train_diloco_torch.py:183): embedding ->
N x (RMSNorm, QKV proj, RoPE, causal SDPA, O proj, RMSNorm, SwiGLU MLP) ->
final RMSNorm -> lm_head -> shifted mean CE.  state_dict keys are identical
to HF's so the 2m/150m/1b fixtures and checkpoints interchange.

Compute policy (mirrors the reference's FSDP MixedPrecision(param_dtype=bf16)
+ fp32 master params, train_fsdp.py:239-245):
  - parameters are fp32 masters;
  - on GPU, forward casts each weight to ``compute_dtype`` (bf16 default)
    inside autograd, runs hand-written HIP kernels for RMSNorm/RoPE/
    attention/SwiGLU/CE and rocBLAS bf16 GEMMs (fp32 accumulate) for the
    dense projections; gradients flow back to fp32 masters;
  - on CPU (the reference's own CPU path / gloo tests), compute_dtype is
    fp32 and the ops run plain torch fp32 math.
"""

from __future__ import annotations

import math
import os
from dataclasses import dataclass

import torch
import torch.nn as nn
import torch.nn.functional as F

from opendiloco_amd import ops
from opendiloco_amd.llama_config import LlamaModelConfig


@dataclass
class CausalLMOutput:
    loss: torch.Tensor | None
    logits: torch.Tensor | None


def _cast(w: torch.Tensor, dtype: torch.dtype) -> torch.Tensor:
    return w if w.dtype == dtype else w.to(dtype)


# version counter for cached low-precision weight copies: bumped by whatever
# mutates master weights in place (FusedAdamW.step, the outer step, loads)
_WEIGHTS_VERSION = [0]


def bump_weights_version() -> None:
    _WEIGHTS_VERSION[0] += 1


class _CastLinearFn(torch.autograd.Function):
    """Linear over a cached low-precision weight copy, with the weight grad
    routed to the fp32 master (autocast semantics; the cache saves the
    fp32->bf16 cast per microbatch that the per-forward `.to()` would pay)."""

    @staticmethod
    def forward(ctx, x, w_master, w16):
        ctx.save_for_backward(x, w16, w_master)
        return F.linear(x, w16)

    @staticmethod
    def backward(ctx, dy):
        x, w16, w_master = ctx.saved_tensors
        dx = dy @ w16
        dy2d = dy.reshape(-1, dy.shape[-1])
        x2d = x.reshape(-1, x.shape[-1])
        wg = w_master.grad
        if wg is not None and wg.is_cuda and dy.is_cuda:
            # split-K dW: batched bf16->fp32 GEMM over token chunks +
            # deterministic fp32 reduce straight into the flat master grad
            if ops.dw_splitk_accum(dy2d, x2d, [(wg, 0)]):
                return dx, None, None
            ops._ext().accum_(wg.reshape(-1), (dy2d.t() @ x2d).reshape(-1))
            return dx, None, None
        dw = dy2d.t() @ x2d
        return dx, dw.to(torch.float32), None


class _CastEmbeddingFn(torch.autograd.Function):
    """Embedding lookup over a cached low-precision weight copy; the weight
    grad goes to the fp32 master via the same aten dense backward the
    uncached `.to()` path used (deterministic, matches previous numerics)."""

    @staticmethod
    def forward(ctx, ids, w_master, w16):
        ctx.save_for_backward(ids, w_master)
        ctx.num_weights = w_master.shape[0]
        return F.embedding(ids, w16)

    @staticmethod
    def backward(ctx, dy):
        ids, w_master = ctx.saved_tensors
        dw = torch.ops.aten.embedding_dense_backward(
            dy, ids, ctx.num_weights, -1, False)
        wg = w_master.grad
        if wg is not None and wg.is_cuda and dw.is_cuda:
            ops._ext().accum_(wg.reshape(-1), dw.reshape(-1))
            return None, None, None
        return None, dw.to(torch.float32), None


class CastLinear(nn.Module):
    """Bias-free linear with an fp32 master weight cast to compute dtype
    (cached per optimizer step); the GEMM itself is a rocBLAS bf16 GEMM with
    fp32 accumulate — a plain library GEMM, per the MFMA design rules."""

    def __init__(self, in_features: int, out_features: int):
        super().__init__()
        self.weight = nn.Parameter(torch.empty(out_features, in_features))
        self.in_features, self.out_features = in_features, out_features
        self._w16 = None
        self._ver = -1

    def _cached_w16(self, dtype: torch.dtype) -> torch.Tensor:
        if self._ver != _WEIGHTS_VERSION[0] or self._w16 is None or self._w16.dtype != dtype:
            with torch.no_grad():
                self._w16 = self.weight.to(dtype)
            self._ver = _WEIGHTS_VERSION[0]
        return self._w16

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        if x.dtype == self.weight.dtype:
            return F.linear(x, self.weight)
        if not x.is_cuda:
            return F.linear(x, _cast(self.weight, x.dtype))
        return _CastLinearFn.apply(x, self.weight, self._cached_w16(x.dtype))


class _MultiCastLinearFn(torch.autograd.Function):
    """One GEMM over a cached concatenation of several bf16 weight copies
    (QKV / gate-up batching): fewer, larger rocBLAS GEMMs; weight grads are
    sliced back to the separate fp32 masters (state_dict layout unchanged)."""

    @staticmethod
    def forward(ctx, x, w16cat, sizes, *masters):
        ctx.save_for_backward(x, w16cat, *masters)
        ctx.sizes = sizes
        return F.linear(x, w16cat)

    @staticmethod
    def backward(ctx, dy):
        x, w16cat, *masters = ctx.saved_tensors
        dx = dy @ w16cat
        dy2d = dy.reshape(-1, dy.shape[-1])
        x2d = x.reshape(-1, x.shape[-1])
        if dy.is_cuda and all(m.grad is not None and m.grad.is_cuda for m in masters):
            targets, off = [], 0
            for n, m in zip(ctx.sizes, masters):
                targets.append((m.grad, off))
                off += n
            if ops.dw_splitk_accum(dy2d, x2d, targets):
                return (dx, None, None, *([None] * len(masters)))
            ext = ops._ext()
            dwcat = dy2d.t() @ x2d
            off = 0
            for n, m in zip(ctx.sizes, masters):
                ext.accum_(m.grad.reshape(-1), dwcat[off:off + n].reshape(-1))
                off += n
            return (dx, None, None, *([None] * len(masters)))
        dwcat = dy2d.t() @ x2d
        dws = []
        off = 0
        for n in ctx.sizes:
            dws.append(dwcat[off:off + n].to(torch.float32))
            off += n
        return (dx, None, None, *dws)


class FusedProj:
    """Cache of a concatenated low-precision copy of several CastLinear
    weights, invalidated by the weights version counter."""

    def __init__(self, linears: list[CastLinear]):
        self.linears = linears
        self.sizes = [l.out_features for l in linears]
        self._w16 = None
        self._ver = -1

    def __call__(self, x: torch.Tensor) -> torch.Tensor:
        if not x.is_cuda or x.dtype == self.linears[0].weight.dtype:
            return torch.cat([l(x) for l in self.linears], dim=-1)
        if self._ver != _WEIGHTS_VERSION[0] or self._w16 is None or self._w16.dtype != x.dtype:
            with torch.no_grad():
                self._w16 = torch.cat([l.weight.to(x.dtype) for l in self.linears], dim=0)
            self._ver = _WEIGHTS_VERSION[0]
        return _MultiCastLinearFn.apply(x, self._w16, self.sizes,
                                        *[l.weight for l in self.linears])


class RMSNorm(nn.Module):
    def __init__(self, hidden: int, eps: float):
        super().__init__()
        self.weight = nn.Parameter(torch.ones(hidden))
        self.variance_epsilon = eps

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        return ops.rmsnorm(x, _cast(self.weight, x.dtype), self.variance_epsilon)


class Rotary:
    """Precomputed fp32 cos/sin tables [S, D/2] (HF inv_freq convention)."""

    def __init__(self, cfg: LlamaModelConfig):
        D = cfg.head_dim
        inv_freq = 1.0 / (cfg.rope_theta ** (torch.arange(0, D, 2, dtype=torch.float32) / D))
        t = torch.arange(cfg.max_position_embeddings, dtype=torch.float32)
        freqs = torch.outer(t, inv_freq)  # [S, D/2]
        self.cos = freqs.cos()
        self.sin = freqs.sin()

    def to_(self, device):
        self.cos = self.cos.to(device)
        self.sin = self.sin.to(device)
        return self


class Attention(nn.Module):
    def __init__(self, cfg: LlamaModelConfig):
        super().__init__()
        h, D = cfg.hidden_size, cfg.head_dim
        self.num_heads = cfg.num_attention_heads
        self.num_kv_heads = cfg.num_key_value_heads
        self.head_dim = D
        self.q_proj = CastLinear(h, self.num_heads * D)
        self.k_proj = CastLinear(h, self.num_kv_heads * D)
        self.v_proj = CastLinear(h, self.num_kv_heads * D)
        self.o_proj = CastLinear(self.num_heads * D, h)
        self.scale = D ** -0.5
        self._qkv = FusedProj([self.q_proj, self.k_proj, self.v_proj])

    def forward(self, x: torch.Tensor, rotary: Rotary) -> torch.Tensor:
        B, S, _ = x.shape
        D = self.head_dim
        qkv = self._qkv(x)
        if x.is_cuda:
            # fused: rope-gather out of the packed buffer, o written [B,S,Hq*D]
            o = ops.qkv_rope_attention(qkv, rotary.cos, rotary.sin,
                                       self.num_heads, self.num_kv_heads, D, self.scale)
            return self.o_proj(o)
        nq, nkv = self.num_heads * D, self.num_kv_heads * D
        q, k, v = qkv.split([nq, nkv, nkv], dim=-1)
        q = q.view(B, S, self.num_heads, D).transpose(1, 2).contiguous()
        k = k.view(B, S, self.num_kv_heads, D).transpose(1, 2).contiguous()
        v = v.view(B, S, self.num_kv_heads, D).transpose(1, 2).contiguous()
        q = ops.rope(q, rotary.cos, rotary.sin, S)
        k = ops.rope(k, rotary.cos, rotary.sin, S)
        o = ops.attention(q, k, v, self.scale)  # [B, Hq, S, D]
        o = o.transpose(1, 2).reshape(B, S, self.num_heads * D)
        return self.o_proj(o)


class MLP(nn.Module):
    def __init__(self, cfg: LlamaModelConfig):
        super().__init__()
        h, i = cfg.hidden_size, cfg.intermediate_size
        self.gate_proj = CastLinear(h, i)
        self.up_proj = CastLinear(h, i)
        self.down_proj = CastLinear(i, h)
        self._gu = FusedProj([self.gate_proj, self.up_proj])
        self._inter = i

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        if x.is_cuda:
            return self.down_proj(ops.swiglu_fused(self._gu(x), self._inter))
        return self.down_proj(ops.swiglu(self.gate_proj(x), self.up_proj(x)))


class DecoderLayer(nn.Module):
    def __init__(self, cfg: LlamaModelConfig):
        super().__init__()
        self.self_attn = Attention(cfg)
        self.mlp = MLP(cfg)
        self.input_layernorm = RMSNorm(cfg.hidden_size, cfg.rms_norm_eps)
        self.post_attention_layernorm = RMSNorm(cfg.hidden_size, cfg.rms_norm_eps)

    def forward(self, x: torch.Tensor, rotary: Rotary) -> torch.Tensor:
        x = x + self.self_attn(self.input_layernorm(x), rotary)
        x = x + self.mlp(self.post_attention_layernorm(x))
        return x

    def forward_fused(self, h: torch.Tensor, delta, rotary: Rotary):
        """Residual-fused path: takes (h, delta) with the pending residual
        delta from the previous layer folded into this layer's first norm
        (one fused kernel instead of add+norm); returns (h', delta')."""
        eps = self.input_layernorm.variance_epsilon
        if delta is None:
            n1 = self.input_layernorm(h)
        else:
            n1, h = ops.rmsnorm_add(h, delta, _cast(self.input_layernorm.weight, h.dtype), eps)
        a = self.self_attn(n1, rotary)
        n2, h = ops.rmsnorm_add(h, a, _cast(self.post_attention_layernorm.weight, h.dtype), eps)
        return h, self.mlp(n2)


class LlamaBackbone(nn.Module):
    def __init__(self, cfg: LlamaModelConfig):
        super().__init__()
        self.embed_tokens = nn.Embedding(cfg.vocab_size, cfg.hidden_size)
        self.layers = nn.ModuleList(DecoderLayer(cfg) for _ in range(cfg.num_hidden_layers))
        self.norm = RMSNorm(cfg.hidden_size, cfg.rms_norm_eps)


class LlamaForCausalLM(nn.Module):
    """Drop-in for the reference's model object: call signature
    model(input_ids=..., attention_mask=..., labels=...) -> .loss/.logits."""

    def __init__(self, config: LlamaModelConfig):
        super().__init__()
        self.config = config
        self.model = LlamaBackbone(config)
        self.lm_head = CastLinear(config.hidden_size, config.vocab_size)
        if config.tie_word_embeddings:
            self.lm_head.weight = self.model.embed_tokens.weight
        self.rotary = Rotary(config)
        self.compute_dtype: torch.dtype | None = None  # None -> fp32 on CPU, bf16 on GPU

    # ---- init / load ----
    def init_weights(self, seed: int | None = None):
        """HF _init_weights: Linear/Embedding ~ N(0, initializer_range), norms = 1."""
        if seed is not None:
            torch.manual_seed(seed)
        std = self.config.initializer_range
        for m in self.modules():
            if isinstance(m, CastLinear):
                m.weight.data.normal_(0.0, std)
            elif isinstance(m, nn.Embedding):
                m.weight.data.normal_(0.0, std)
            elif isinstance(m, RMSNorm):
                m.weight.data.fill_(1.0)
        return self

    @classmethod
    def from_pretrained(cls, path: str) -> "LlamaForCausalLM":
        cfg = LlamaModelConfig.from_json(path)
        model = cls(cfg)
        # load HF safetensors if present (same state_dict keys)
        import glob

        files = sorted(glob.glob(os.path.join(path, "*.safetensors")))
        if files:
            from safetensors.torch import load_file

            sd = {}
            for f in files:
                sd.update(load_file(f))
            missing, unexpected = model.load_state_dict(sd, strict=False)
            missing = [k for k in missing if "rotary" not in k]
            if cfg.tie_word_embeddings:
                missing = [k for k in missing if k != "lm_head.weight"]
            assert not missing and not unexpected, (missing, unexpected)
            model.float()
        else:
            model.init_weights()
        return model

    def _cached_embed(self, cdtype: torch.dtype) -> torch.Tensor:
        w = self.model.embed_tokens.weight
        if getattr(self, "_embed_ver", -1) != _WEIGHTS_VERSION[0]                 or getattr(self, "_embed_w16", None) is None or self._embed_w16.dtype != cdtype:
            with torch.no_grad():
                self._embed_w16 = w.to(cdtype).contiguous()
            self._embed_ver = _WEIGHTS_VERSION[0]
        return self._embed_w16

    def _dtype_for(self, device: torch.device) -> torch.dtype:
        if self.compute_dtype is not None:
            return self.compute_dtype
        return torch.float32 if device.type == "cpu" else torch.bfloat16

    def forward(self, input_ids: torch.Tensor, attention_mask: torch.Tensor | None = None,
                labels: torch.Tensor | None = None, **_ignored) -> CausalLMOutput:
        device = input_ids.device
        cdtype = self._dtype_for(device)
        # checked once per model (the .all() forces a device sync; the hot
        # loop must stay async)
        if attention_mask is not None and not getattr(self, "_mask_checked", False):
            self._mask_checked = True
            if not bool(attention_mask.bool().all()):
                raise NotImplementedError(
                    "padding attention masks are not supported: the hot path is "
                    "the reference's fake/C4 full-sequence batches (mask all "
                    "ones, open_diloco/utils.py:166)")
        if self.rotary.cos.device != device:
            self.rotary.to_(device)
        S = input_ids.shape[1]
        assert S <= self.config.max_position_embeddings

        if cdtype == self.model.embed_tokens.weight.dtype or not input_ids.is_cuda:
            h = F.embedding(input_ids, _cast(self.model.embed_tokens.weight, cdtype))
        else:
            h = _CastEmbeddingFn.apply(input_ids, self.model.embed_tokens.weight,
                                       self._cached_embed(cdtype))
        delta = None
        for layer in self.model.layers:
            h, delta = layer.forward_fused(h, delta, self.rotary)
        if delta is None:
            h = self.model.norm(h)
        else:
            h, _ = ops.rmsnorm_add(h, delta, _cast(self.model.norm.weight, cdtype),
                                   self.model.norm.variance_epsilon)
        logits = self.lm_head(h)  # [B, S, V], compute dtype

        loss = None
        if labels is not None:
            # HF shift (predict token t+1 from logits at t; mean over
            # B*(S-1)) happens inside the fused CE kernel
            loss = ops.causal_lm_loss(logits, labels)
        return CausalLMOutput(loss=loss, logits=logits)

    def train_config_summary(self) -> dict:
        return {"params": sum(p.numel() for p in self.parameters()), **self.config.to_dict()}
