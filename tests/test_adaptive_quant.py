"""Adaptive quantization policy tests (QuantPipe controllers)."""
import pytest
import torch

from pipeedge_amd.utils.controller import AdaptiveIntegralXupController, KalmanFilter
from pipeedge_amd.utils.quant import (AdaptiveBitwidthPerformanceController,
                                      constrain_max_bitwidth)


def test_kalman_filter_converges():
    kf = KalmanFilter()
    for _ in range(100):
        est = kf(5.0)
    assert est == pytest.approx(5.0, rel=0.01)


def test_integral_controller_tracks_reference():
    # plant: y = 10 / u (workload 10, speedup u)
    ctl = AdaptiveIntegralXupController(reference=5.0, u_0=1.0, u_max=16.0)
    u = 1.0
    for _ in range(50):
        y = 10.0 * u   # measured rate rises with speedup
        u = ctl(y)
    # y = 10*u should settle near reference 5 -> u ~ 0.5 clamped to 1
    assert u == 1.0
    ctl2 = AdaptiveIntegralXupController(reference=40.0, u_0=1.0, u_max=16.0)
    u = 1.0
    for _ in range(50):
        u = ctl2(10.0 * u)
    assert 10.0 * u == pytest.approx(40.0, rel=0.05)


def test_constrain_max_bitwidth():
    # 32 Mbit at 32 Mbit/s with 1 s budget: full width fits
    out = constrain_max_bitwidth(torch.tensor(1.0), torch.tensor(32.0),
                                 torch.tensor(32.0), torch.tensor(32))
    assert int(out) == 32
    # 0.25 s budget -> need 4x compression -> 8 bits
    out = constrain_max_bitwidth(torch.tensor(0.25), torch.tensor(32.0),
                                 torch.tensor(32.0), torch.tensor(32))
    assert int(out) == 8
    # impossible budget -> 0
    out = constrain_max_bitwidth(torch.tensor(1e-9), torch.tensor(32.0),
                                 torch.tensor(32.0), torch.tensor(32))
    assert int(out) == 0


def test_bitwidth_controller_window_split():
    ctl = AdaptiveBitwidthPerformanceController(
        perf_constraint=20.0, bitwidths=[32, 16, 8, 4, 2], bitwidth_start=32)
    # measured rate below target: controller requests more speedup
    bw1, bw2, iters1 = ctl(10.0, window_len=10)
    assert bw1 in (32, 16, 8, 4, 2) and bw2 in (32, 16, 8, 4, 2)
    assert bw2 <= bw1
    assert 0 <= iters1 <= 10


def test_runtime_bitwidths_table():
    import runtime
    # unique discrete compression factors only
    factors = [32 // b for b in runtime.BITWIDTHS]
    assert factors == sorted(set(factors))
