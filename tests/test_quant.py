"""Quantization tests: encode/decode round-trip within the quantization error
bound across bitwidths and shapes (mirrors the reference's sweep,
test/quant/test_quant.py), plus clamp ops and the padded static-wire mode."""
import math

import pytest
import torch

from pipeedge_amd import quant
from pipeedge_amd.models import spec as model_spec
from pipeedge_amd.models.config import get_config


@pytest.mark.parametrize("bit", [1, 2, 3, 4, 5, 6, 8, 16, 32])
@pytest.mark.parametrize("shape", [(8, 10, 10), (1, 100, 100), (8, 197, 786)])
def test_roundtrip_within_quant_error(bit, shape):
    torch.manual_seed(bit)
    x = torch.rand(*shape)
    enc = quant.tensor_encode_outerdim(x, bit)
    assert len(enc) == 5
    assert enc[0].dtype == torch.uint8
    dec = quant.tensor_decode_outerdim(enc)
    assert dec.shape == x.shape
    levels = (1 << bit) - 1
    bound = 0.5 / levels + 1e-6 if bit < 32 else 1e-6
    assert (dec - x).abs().max().item() <= bound


def test_bit0_passthrough():
    x = torch.rand(4, 7, 9)
    enc = quant.tensor_encode_outerdim(x, 0)
    assert torch.equal(quant.tensor_decode_outerdim(enc), x)


def test_padded_wire_matches_spec():
    """Payload padded for max_bit=8 decodes correctly when encoded at bit=4,
    and its shape matches what the static schema predicts."""
    cfg = get_config("google/vit-base-patch16-224")
    specs = model_spec.boundary_spec(cfg, 4, 4, torch.float32)
    wire = model_spec.quant_wire_spec(specs, 8)
    assert len(wire) == 5
    pad_words = wire[0].shape[1] // 4
    x = torch.rand(4, 197, 768)
    enc = quant.tensor_encode_outerdim(x, 4, pad_words=pad_words)
    assert tuple(enc[0].shape) == tuple(wire[0].shape)
    dec = quant.tensor_decode_outerdim(enc)
    assert (dec - x).abs().max().item() <= 0.5 / 15 + 1e-6


def test_compression_factor():
    assert quant.compression_factor(torch.tensor(8)).item() == 4


@pytest.mark.parametrize("clamp_fn", [quant.clamp_banner2019_laplace,
                                      quant.clamp_banner2019_gelu])
def test_clamp_reduces_range(clamp_fn):
    torch.manual_seed(0)
    x = torch.randn(1000) * 3
    y = clamp_fn(x, 4)
    assert y.abs().max() <= x.abs().max()
    # clamping at high bitwidth should barely clip
    y16 = clamp_fn(x, 16)
    assert (y16 - x).abs().max() < (y - x).abs().max() + 1e-6


def test_lambertw_against_scipy():
    from scipy.special import lambertw
    from pipeedge_amd.quant import _lambertw_real
    for v in [0.5, 3.0, 3 * 4.0 ** 8, 3 * 4.0 ** 33, 1e12]:
        ref = lambertw(v).real
        assert abs(_lambertw_real(v) - ref) <= 1e-9 * (1 + abs(ref))


def test_quant_wire_words_formula():
    """floor(32/bit) values per word - bit=3 packs 10 per word, not 32/3."""
    cfg = get_config("google/vit-base-patch16-224")
    specs = model_spec.boundary_spec(cfg, 4, 1, torch.float32)
    wire = model_spec.quant_wire_spec(specs, 3)
    per_elem = 197 * 768
    assert wire[0].shape == (1, math.ceil(per_elem / 10) * 4)


def test_quant_roundtrip_additive_mask_tensor():
    """BERT boundary payloads now include the additive padding mask
    (0 / -1e9 values): the wire must carry its extremes exactly enough
    that masked positions stay hugely negative and live positions ~0."""
    import torch
    from pipeedge_amd.quant import tensor_encode_outerdim, tensor_decode_outerdim
    mask = torch.zeros(4, 128)
    mask[:, 100:] = -1e9
    for bit in (2, 4, 8):
        wire = tensor_encode_outerdim(mask.clone(), bit)
        out = tensor_decode_outerdim(wire)
        assert out.shape == mask.shape
        assert (out[:, 100:] < -1e8).all(), bit
        assert (out[:, :100].abs() < 1e7).all(), bit
