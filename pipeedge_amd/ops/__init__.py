"""Compute-op dispatch: hand-written CDNA4 HIP kernels on GPU, torch on CPU.

Every model shard calls through this module. On a CUDA (ROCm) device the ops
MUST run on the in-tree HIP extension (``pipeedge_amd/ops/hip``); if the
extension is missing or an op is unimplemented, the call raises instead of
silently falling back to eager PyTorch. Set ``PIPEEDGE_AMD_FORCE_TORCH=1`` to
explicitly opt into the eager path on GPU (used for A/B comparisons only).

The reference framework has no kernels at all - its
hot path is eager HF modules; this module is where the MI355X build replaces
them.
"""
import math
import os
from typing import Optional

import torch
from torch import Tensor

from . import torch_ref

_hip = None          # the loaded extension module, or None
_hip_load_err = None  # exception from the load attempt, if any
_hip_tried = False


def _load_hip():
    global _hip, _hip_load_err, _hip_tried
    if _hip_tried:
        return _hip
    _hip_tried = True
    try:
        from . import hip_loader
        _hip = hip_loader.load()
    except Exception as exc:  # noqa: BLE001 - report it at use time
        _hip_load_err = exc
        _hip = None
    return _hip


def force_torch() -> bool:
    return os.environ.get("PIPEEDGE_AMD_FORCE_TORCH", "0") == "1"


def _use_hip(x: Tensor) -> bool:
    """HIP-kernel path iff tensor is on GPU and the eager escape is not set."""
    if not x.is_cuda or force_torch():
        return False
    if _load_hip() is None:
        raise RuntimeError(
            "pipeedge_amd HIP extension is required for GPU execution but "
            f"failed to load: {_hip_load_err!r}. Build it with "
            "`python setup.py build_ext --inplace` (or __graft_entry__.build()); "
            "set PIPEEDGE_AMD_FORCE_TORCH=1 only to explicitly run eager torch."
        )
    return True


def hip_available() -> bool:
    return _load_hip() is not None


def layernorm(x: Tensor, weight: Tensor, bias: Tensor, eps: float = 1e-6) -> Tensor:
    if _use_hip(x):
        return _hip.layernorm(x, weight, bias, eps)
    return torch_ref.layernorm(x, weight, bias, eps)


def add_layernorm(x: Tensor, residual: Tensor, weight: Tensor, bias: Tensor,
                  eps: float = 1e-6) -> Tensor:
    if _use_hip(x):
        return _hip.add_layernorm(x, residual, weight, bias, eps)
    return torch_ref.add_layernorm(x, residual, weight, bias, eps)


_ACT_CODE = {None: 0, "gelu": 1, "tanh": 2}


def linear(x: Tensor, weight_nk: Tensor, bias: Optional[Tensor] = None,
           act: Optional[str] = None, residual: Optional[Tensor] = None,
           want_stats: bool = False):
    """GEMM + fused epilogue. With ``want_stats`` (GPU path) also returns
    per-row partial (sum, sumsq) of the OUTPUT, f32 [rows, parts, 2] -
    written without atomics by each column block and summed by the next
    folded-LN GEMM (see ln_linear)."""
    if _use_hip(x):
        out, stats = _hip.linear(x, weight_nk, bias, _ACT_CODE[act],
                                 residual, 0, None, None, None, 1e-12,
                                 want_stats)
        return (out, stats) if want_stats else out
    out = torch_ref.linear(x, weight_nk, bias, act, residual)
    return (out, None) if want_stats else out


def ln_stats(x: Tensor) -> Tensor:
    """Per-row (sum, sumsq) over the last dim, f32 [rows, 1, 2]."""
    if _use_hip(x):
        return _hip.ln_stats(x.reshape(-1, x.shape[-1]))
    xf = x.reshape(-1, x.shape[-1]).float()
    return torch.stack((xf.sum(-1), (xf * xf).sum(-1)), dim=-1).unsqueeze(1)


def make_ln_fold(weight_nk: Tensor, ln_weight: Tensor, ln_bias: Tensor):
    """Precompute the folded-LN GEMM operands (GPU):
    W' = W * ln_w (per-K column), s[n] = sum_k W'[n,k], c[n] = W @ ln_b.

    Because LayerNorm is affine per row,
      LN(x) @ W^T = r*(x @ W'^T) - r*mu*s + c
    with per-row mean mu and inv-std r - so the GEMM runs on RAW x and the
    LN collapses into the epilogue (no standalone LN kernel, no extra
    activation pass)."""
    w_p = (weight_nk.float() * ln_weight.float()).to(weight_nk.dtype)         .contiguous()
    s = w_p.float().sum(dim=1).contiguous()
    c = (weight_nk.float() @ ln_bias.float()).contiguous()
    return (w_p, s, c)


def ln_linear(x: Tensor, ln_weight: Tensor, ln_bias: Tensor, eps: float,
              weight_nk: Tensor, bias: Optional[Tensor] = None,
              act: Optional[str] = None, residual: Optional[Tensor] = None,
              stats: Optional[Tensor] = None, fused=None) -> Tensor:
    """LayerNorm -> Linear; folded into ONE GEMM on GPU (see make_ln_fold).

    ``stats``: per-row (sum, sumsq) of x if already known (emitted by the
    producing GEMM); computed by a ln_stats kernel otherwise."""
    if _use_hip(x):
        assert fused is not None, "ln_linear GPU path needs make_ln_fold(...)"
        w_p, s, c = fused
        if stats is None:
            stats = _hip.ln_stats(x.reshape(-1, x.shape[-1]))
        out, _ = _hip.linear(x, w_p, bias, _ACT_CODE[act], residual, 0,
                             stats, s, c, eps, False)
        return out
    y = torch_ref.layernorm(x, ln_weight, ln_bias, eps)
    return torch_ref.linear(y, weight_nk, bias, act, residual)


def fused_ln_active(x: Tensor) -> bool:
    """Whether the folded-LN GEMM path is used for this tensor.

    Default OFF: end-to-end A/B on MI355X measured the folded path at net
    ~parity-to-slower vs the standalone LN kernels (the LN savings are offset
    by partial-stats traffic in the consumer epilogues); enable with
    PIPEEDGE_FOLDED_LN=1 to experiment."""
    return (x.is_cuda and not force_torch()
            and os.environ.get("PIPEEDGE_FOLDED_LN", "0") == "1")


def attention_core(q: Tensor, k: Tensor, v: Tensor, scale: Optional[float] = None,
                   attn_mask: Optional[Tensor] = None) -> Tensor:
    if scale is None:
        scale = 1.0 / math.sqrt(q.shape[-1])
    if _use_hip(q):
        return _hip.attention(q, k, v, scale, attn_mask)
    return torch_ref.attention_core(q, k, v, scale, attn_mask)


def qkv_attention(qkv: Tensor, num_heads: int, scale: Optional[float] = None,
                  attn_mask: Optional[Tensor] = None) -> Tensor:
    if scale is None:
        scale = 1.0 / math.sqrt(qkv.shape[-1] // 3 // num_heads)
    if _use_hip(qkv):
        return _hip.qkv_attention(qkv, num_heads, scale, attn_mask)
    return torch_ref.qkv_attention(qkv, num_heads, scale, attn_mask)


def patchify(pixels: Tensor, patch_size: int) -> Tensor:
    if _use_hip(pixels):
        return _hip.patchify(pixels, patch_size)
    return torch_ref.patchify(pixels, patch_size)


def embed_finish(x: Tensor, prefix: Tensor, pos: Tensor) -> Tensor:
    """Prepend prefix token rows (cls / cls+dist) and add position
    embeddings: out[b, s] = (prefix[s] if s < T else x[b, s-T]) + pos[s]."""
    if _use_hip(x):
        return _hip.embed_finish(x, prefix, pos)
    return torch_ref.embed_finish(x, prefix, pos)
