"""Adaptive-quantization policy utilities (parity: reference utils/quant.py).

``constrain_max_bitwidth`` solves for the largest bitwidth meeting a data
movement time constraint given discrete 32-bit-word packing;
``AdaptiveBitwidthPerformanceController`` (QuantPipe's CONTROLLER policy)
drives bitwidth through the integral x-up controller, splitting each window
period between two adjacent bitwidths to hit the rate set point exactly.
"""
import math
from typing import List, Tuple

import torch

from ..quant import compression_factor
from .controller import AdaptiveIntegralXupController


def constrain_max_bitwidth(t_max: torch.Tensor, d_size: torch.Tensor,
                           d_speed: torch.Tensor,
                           bw_max: torch.Tensor) -> torch.Tensor:
    """Largest bitwidth whose (discretely packed) data moves within ``t_max``
    at ``d_speed``; 0 if even 1-bit data cannot make it."""
    bitwidths = torch.arange(int(bw_max), -1, -1, dtype=torch.int)
    # discrete packing: floor(32/bit) values per word, so the achieved scale
    # is 1/int(32/bit), not bit/32
    scales = compression_factor(bitwidths[:-1]).to(dtype=torch.int).reciprocal()
    scales = torch.hstack((scales, torch.tensor(0.0)))
    scale = torch.div(d_speed * t_max, d_size)
    return bitwidths[scale >= scales][0]


class AdaptiveBitwidthPerformanceController(AdaptiveIntegralXupController):
    """Adaptive bitwidth control for a data-rate constraint.

    Speedup is modeled as inversely proportional to bitwidth (normalized to
    the maximum bitwidth; perfect packing assumed). Each window is split
    between the two bitwidths bracketing the target speedup.
    """

    def __init__(self, perf_constraint: float, bitwidths: List[int],
                 bitwidth_start: int):
        self._bitwidths = sorted(bitwidths, reverse=True)
        self._speedups = [self._bitwidths[0] / b for b in self._bitwidths]
        u_0 = self._bitwidths[0] / bitwidth_start
        super().__init__(perf_constraint, u_0, u_max=self._speedups[-1])

    def __call__(self, perf_measured: float, window_len: int) \
            -> Tuple[int, int, int]:
        """Returns (bitwidth1, bitwidth2, iterations to spend in bitwidth1
        during the next window)."""
        xup_targ = super().__call__(perf_measured)
        idx_slow = max(0, len([s for s in self._speedups if s <= xup_targ]) - 1)
        idx_fast = min(idx_slow + 1, len(self._speedups) - 1)
        xup_slow = self._speedups[idx_slow]
        xup_fast = self._speedups[idx_fast]
        # time split x solving 1/target = x/slow + (1-x)/fast
        if math.isclose(xup_slow, xup_fast):
            frac = 0.0
        else:
            frac = (xup_slow * (xup_fast - xup_targ)) / \
                (xup_targ * (xup_fast - xup_slow))
        return (self._bitwidths[idx_slow], self._bitwidths[idx_fast],
                round(window_len * frac))
