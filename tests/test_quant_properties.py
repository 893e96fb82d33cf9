"""Property-based tests (hypothesis) for the quantization wire format -
stronger than the reference's fixed sweep (reference test_quant.py:24-30)."""
import hypothesis
from hypothesis import given, settings, strategies as st
import pytest
import torch

from pipeedge_amd import quant
from pipeedge_amd.models.spec import TensorSpec, quant_wire_spec


@settings(max_examples=60, deadline=None)
@given(bit=st.sampled_from([2, 3, 4, 5, 6, 8, 12, 16, 32]),
       rows=st.integers(1, 9), cols=st.integers(1, 97),
       seed=st.integers(0, 2**16))
def test_roundtrip_error_bound(bit, rows, cols, seed):
    """decode(encode(x)) is within one quantization step of clamp(x)."""
    torch.manual_seed(seed)
    x = torch.rand(1, rows, cols)  # [0,1): clamp inactive (min >= 0)
    enc = quant.tensor_encode(x[0], bit)
    out = quant.tensor_decode(enc)
    span = (x.max() - x.min()).item()
    step = span / max(2 ** bit - 1, 1)
    assert (out - x[0]).abs().max().item() <= step * 0.51 + 1e-6


@settings(max_examples=40, deadline=None)
@given(bit=st.sampled_from([2, 3, 4, 5, 6, 8, 16, 32]),
       u=st.integers(1, 4), s=st.integers(1, 30), d=st.integers(1, 40))
def test_wire_spec_matches_encoded_sizes(bit, u, s, d):
    """quant_wire_spec's static byte counts equal the encoder's output."""
    x = torch.rand(u, s, d)
    enc = quant.tensor_encode_outerdim(x, bit)
    specs = quant_wire_spec([TensorSpec((u, s, d), torch.float32)], bit)
    assert len(specs) == 5
    # comm payload: spec reserves ceil-words * 4 uint8 per image
    comm = enc[0]
    assert comm.shape[0] == u
    assert specs[0].shape == tuple(comm.shape) or \
        specs[0].shape[1] >= comm.shape[1]
    # exact when no padding is configured
    assert specs[0].shape == (u, comm.shape[1])
