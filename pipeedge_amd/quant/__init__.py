"""Post-training quantization of inter-stage activations (QuantPipe).

Capability parity with reference ``src/pipeedge/quantization/basic_op.py`` and
``clamp_op.py``, re-implemented as vectorized torch ops that run on CPU *and*
GPU tensors (the reference is numpy-only and forced activations through host
memory). Semantics match the reference:

- linear shift/scale to [0,1], round to ``(1<<bit)-1`` levels
  (basic_op.py:17-35 'original' mode);
- bit-pack ``32/bit`` values per 32-bit word, value ``i`` at bit offset
  ``(i % ratio) * bit`` (basic_op.py:38-55), transported as little-endian
  uint8 (basic_op.py:98-101);
- wire format per source tensor: ``[comm(uint8), shape(int32), scale(f32),
  shift(f32), bit(int8)]`` (basic_op.py:143), applied per-microbatch-element
  and stacked (basic_op.py:166-176);
- Banner et al. NIPS'19 optimal clamp, Laplace and GeLU variants
  (clamp_op.py:11-33).

Extension over the reference: ``pad_words`` lets the encoder emit a
fixed-size payload sized for a configured max bitwidth, so the wire schema
stays static while the adaptive controller changes ``bit`` at runtime.
"""
import math
from typing import List, Optional

import torch
from torch import Tensor


def compression_factor(quant_bit) -> Tensor:
    """Data-size improvement for bitwidths > 0 (reference basic_op.py:109-111)."""
    if not isinstance(quant_bit, Tensor):
        quant_bit = torch.tensor(quant_bit)
    return torch.div(32, quant_bit)


# ---------------------------------------------------------------------------
# clamping (reference clamp_op.py)
# ---------------------------------------------------------------------------

def _lambertw_real(x: float) -> float:
    """Principal-branch Lambert W for real x >= 0 (Halley iteration)."""
    if x == 0.0:
        return 0.0
    w = math.log1p(x) if x < math.e else math.log(x) - math.log(math.log(x))
    for _ in range(50):
        ew = math.exp(w)
        f = w * ew - x
        denom = ew * (w + 1.0) - (w + 2.0) * f / (2.0 * w + 2.0)
        w_next = w - f / denom
        if abs(w_next - w) <= 1e-12 * (1.0 + abs(w_next)):
            return w_next
        w = w_next
    return w


def clamp_banner2019_laplace(tensor: Tensor, bit: int) -> Tensor:
    """Optimal Laplace clamp: alpha = W(3*4^bit) * sqrt(var/2)."""
    variance = torch.var(tensor.float(), unbiased=False)
    dist_parameter = torch.sqrt(0.5 * variance)
    alpha = _lambertw_real(3.0 * 4.0 ** bit) * dist_parameter
    return tensor.clamp(min=-alpha.item(), max=alpha.item())


def clamp_banner2019_gelu(tensor: Tensor, bit: int) -> Tensor:
    """GeLU variant: half-bell distribution, W(3*4^(bit+1)) factor."""
    t = tensor.float()
    variance = 2.0 * torch.pow(t, 2).sum() / t.numel()
    dist_parameter = torch.sqrt(0.5 * variance)
    alpha = _lambertw_real(3.0 * 4.0 ** (bit + 1)) * dist_parameter
    return tensor.clamp(min=-alpha.item(), max=alpha.item())


# ---------------------------------------------------------------------------
# encode / decode
# ---------------------------------------------------------------------------

def _pack_bits(int_map: Tensor, bit: int) -> Tensor:
    """Pack int64 values in [0, 2^bit) into little-endian uint8 words."""
    ratio = 32 // bit
    n = int_map.numel()
    pad = (-n) % ratio
    if pad:
        int_map = torch.cat([int_map, int_map.new_zeros(pad)])
    grouped = int_map.view(-1, ratio)
    shifts = torch.arange(ratio, device=int_map.device, dtype=torch.int64) * bit
    words = (grouped << shifts).sum(dim=1)  # disjoint fields: sum == or
    b = torch.empty(words.numel(), 4, dtype=torch.uint8, device=words.device)
    for i in range(4):
        b[:, i] = ((words >> (8 * i)) & 0xFF).to(torch.uint8)
    return b.view(-1)


def _unpack_bits(comm: Tensor, numel: int, bit: int) -> Tensor:
    """Inverse of `_pack_bits`: little-endian uint8 -> int64 values."""
    ratio = 32 // bit
    b = comm.to(torch.int64).view(-1, 4)
    words = b[:, 0] | (b[:, 1] << 8) | (b[:, 2] << 16) | (b[:, 3] << 24)
    shifts = torch.arange(ratio, device=comm.device, dtype=torch.int64) * bit
    vals = (words.unsqueeze(1) >> shifts) & ((1 << bit) - 1)
    return vals.view(-1)[:numel]


def tensor_encode(input_data: Tensor, quant_bit: int,
                  pad_words: Optional[int] = None) -> List[Tensor]:
    """Encode one tensor (reference basic_op.py:114-143)."""
    bit_t = torch.tensor(quant_bit, dtype=torch.int8, device=input_data.device)
    if quant_bit == 0:
        return [input_data,
                torch.tensor(input_data.shape, device=input_data.device),
                torch.tensor(1.0, device=input_data.device),
                torch.tensor(0.0, device=input_data.device), bit_t]
    x = input_data.float()
    shape = torch.tensor(x.shape, dtype=torch.int32, device=x.device)
    shift = x.min()
    x = x - shift
    scale = x.max()
    safe_scale = scale if scale.item() != 0 else scale + 1.0
    x = x / safe_scale
    levels = (1 << quant_bit) - 1
    # float64 so levels*x is exact up to 2^32-1 (f32 wraps at bit >= 25)
    int_map = torch.round(levels * x.double()).to(torch.int64).view(-1)
    comm = _pack_bits(int_map, quant_bit)
    if pad_words is not None:
        want = pad_words * 4
        if comm.numel() < want:
            comm = torch.cat([comm, comm.new_zeros(want - comm.numel())])
    return [comm, shape, scale.float(), shift.float(), bit_t]


def tensor_decode(encodings: List[Tensor]) -> Tensor:
    """Decode one tensor (reference basic_op.py:146-163)."""
    comm, shape, scale, shift, bit_t = encodings
    quant_bit = int(bit_t.item() if bit_t.dim() == 0 else bit_t.view(-1)[0].item())
    if quant_bit == 0:
        return comm
    shape_list = [int(s) for s in shape.view(-1)]
    numel = 1
    for s in shape_list:
        numel *= s
    words = math.ceil(numel / (32 // quant_bit))
    comm = comm.view(-1)[:words * 4]
    int_map = _unpack_bits(comm, numel, quant_bit)
    levels = (1 << quant_bit) - 1
    x = int_map.float() / levels
    return (x * scale.float() + shift.float()).view(shape_list)


def _hip_ops():
    """The gfx950 kernel extension, or None on CPU / when forced off."""
    from .. import ops as _ops
    if _ops.force_torch() or not _ops.hip_available():
        return None
    from ..ops import hip_loader
    return hip_loader.load()


def tensor_encode_outerdim(batched_tensor: Tensor, quant_bit: int,
                           pad_words: Optional[int] = None) -> List[Tensor]:
    """Encode each element over the microbatch dim and stack
    (reference basic_op.py:166-170). On GPU the whole microbatch is encoded
    by the fused HIP min/max + bit-pack kernels (quantpack.hip)."""
    if batched_tensor.is_cuda and quant_bit > 0:
        hip = _hip_ops()
        if hip is None:
            raise RuntimeError("quant encode on GPU requires the HIP "
                               "extension (or PIPEEDGE_AMD_FORCE_TORCH=1)")
        u = batched_tensor.shape[0]
        img_shape = list(batched_tensor.shape[1:])
        comm, scale, shift = hip.quant_encode(
            batched_tensor.to(torch.bfloat16).contiguous(), quant_bit,
            pad_words or 0)
        shape = torch.tensor(img_shape, dtype=torch.int32,
                             device=batched_tensor.device).unsqueeze(0).expand(u, -1).contiguous()
        bits = torch.full((u,), quant_bit, dtype=torch.int8,
                          device=batched_tensor.device)
        return [comm, shape, scale, shift, bits]
    lists = [tensor_encode(t, quant_bit, pad_words) for t in batched_tensor]
    return [torch.stack(group, 0) for group in zip(*lists)]


def tensor_decode_outerdim(batched_encodings: List[Tensor]) -> Tensor:
    """Decode a stacked 5-tensor group (reference basic_op.py:173-176)."""
    comm, shape, scale, shift, bits = batched_encodings
    if comm.is_cuda:
        quant_bit = int(bits.view(-1)[0].item())
        if quant_bit > 0:
            hip = _hip_ops()
            if hip is None:
                raise RuntimeError("quant decode on GPU requires the HIP "
                                   "extension (or PIPEEDGE_AMD_FORCE_TORCH=1)")
            img_shape = [int(s) for s in shape.view(shape.shape[0], -1)[0]]
            return hip.quant_decode(comm.contiguous(), scale, shift,
                                    quant_bit, img_shape)
    tensors = [tensor_decode(list(enc)) for enc in zip(*batched_encodings)]
    return torch.stack(tensors, 0)
