"""Plain-PyTorch reference implementations of the framework's compute ops.

These are the numerics ground truth for the HIP/CDNA4 kernels in
``pipeedge_amd/ops/hip`` (GPU tests compare the kernels against this module
run in fp32) and the execution path on CPU, where the reference framework ran
its shards eagerly through HF modules (reference vit.py:55-70, bert.py:41-52).

Weight layout convention (MI355X-native): all linear weights are stored
``[N, K]`` (out-features x in-features, the torch.nn.Linear layout). This is
the MFMA-optimal operand layout on gfx950: both GEMM operands are then
k-contiguous per row, so A- and B-fragments alike stage through LDS with
16-B `global_load_lds` pieces and read back as single `ds_read_b128`s
(cdna_hip_programming.md section 5).
"""
import math
from typing import Optional

import torch
from torch import Tensor


def layernorm(x: Tensor, weight: Tensor, bias: Tensor, eps: float = 1e-6) -> Tensor:
    """LayerNorm over the last dim."""
    return torch.nn.functional.layer_norm(x, (x.shape[-1],), weight, bias, eps)


def add_layernorm(x: Tensor, residual: Tensor, weight: Tensor, bias: Tensor,
                  eps: float = 1e-6) -> Tensor:
    """Fused (x + residual) -> LayerNorm (BERT post-LN pattern)."""
    return layernorm(x + residual, weight, bias, eps)


def linear(x: Tensor, weight_nk: Tensor, bias: Optional[Tensor] = None,
           act: Optional[str] = None, residual: Optional[Tensor] = None) -> Tensor:
    """GEMM with optionally fused bias + activation + residual add.

    ``weight_nk`` is ``[N, K]``; ``y = act(x @ w.T + b) + residual``.
    """
    y = x.matmul(weight_nk.t())
    if bias is not None:
        y = y + bias
    if act == "gelu":
        y = torch.nn.functional.gelu(y)
    elif act == "tanh":
        y = torch.tanh(y)
    elif act is not None:
        raise ValueError(f"unknown activation: {act}")
    if residual is not None:
        y = y + residual
    return y


def attention_core(q: Tensor, k: Tensor, v: Tensor, scale: Optional[float] = None,
                   attn_mask: Optional[Tensor] = None) -> Tensor:
    """softmax(q @ k^T * scale + mask) @ v over ``[B, H, S, D]`` inputs."""
    if scale is None:
        scale = 1.0 / math.sqrt(q.shape[-1])
    scores = torch.matmul(q, k.transpose(-1, -2)) * scale
    if attn_mask is not None:
        scores = scores + attn_mask
    probs = torch.softmax(scores, dim=-1)
    return torch.matmul(probs, v)


def qkv_attention(qkv: Tensor, num_heads: int, scale: Optional[float] = None,
                  attn_mask: Optional[Tensor] = None) -> Tensor:
    """Multi-head attention over a packed ``[B, S, 3*D]`` QKV projection.

    Columns of the last dim are ``[q | k | v]``. Returns the merged-head
    context ``[B, S, D]``. This is the layout the fused HIP attention kernel
    consumes directly (no permute kernels on GPU).
    """
    b, s, three_d = qkv.shape
    d = three_d // 3
    dh = d // num_heads
    q, k, v = qkv.split(d, dim=-1)

    def _heads(t):
        return t.reshape(b, s, num_heads, dh).permute(0, 2, 1, 3)

    if attn_mask is not None and attn_mask.dim() == 2:
        # [B, S] per-kpos additive mask (the HIP kernel's contract):
        # broadcast over heads and query rows
        attn_mask = attn_mask.reshape(b, 1, 1, s)
    ctx = attention_core(_heads(q), _heads(k), _heads(v), scale, attn_mask)
    return ctx.permute(0, 2, 1, 3).reshape(b, s, d)


def embed_finish(x: Tensor, prefix: Tensor, pos: Tensor) -> Tensor:
    """out[b, s] = (prefix[s] if s < T else x[b, s-T]) + pos[s]."""
    b = x.shape[0]
    pre = prefix.unsqueeze(0).expand(b, -1, -1).to(x.dtype)
    return torch.cat((pre, x), dim=1) + pos.to(x.dtype)


def patchify(pixels: Tensor, patch_size: int) -> Tensor:
    """[B, C, H, W] -> [B, num_patches, C*P*P] for non-overlapping P x P patches.

    The inner layout per patch is (C, Ph, Pw), matching a Conv2d(P, stride=P)
    kernel flattened as [C*P*P, D].
    """
    b, c, h, w = pixels.shape
    p = patch_size
    x = pixels.reshape(b, c, h // p, p, w // p, p)
    x = x.permute(0, 2, 4, 1, 3, 5)  # B, H/p, W/p, C, p, p
    return x.reshape(b, (h // p) * (w // p), c * p * p)
