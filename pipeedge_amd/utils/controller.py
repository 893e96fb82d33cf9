"""Feedback-control utilities (parity: reference utils/controller.py).

A scalar Kalman filter and an adaptive integral speedup ("x-up") controller
in the style of EMSOFT'13 (Hoffmann et al.): the Kalman filter tracks the
base workload, whose reciprocal serves as the integral gain, with actuator
saturation to prevent windup.
"""
from typing import Optional


class KalmanFilter:
    """Scalar Kalman filter (Welch & Bishop formulation).

    ``Q`` process noise covariance, ``R`` measurement noise covariance.
    """

    def __init__(self, x_hat_0: float = 0.0, p_0: float = 1.0):
        self._x_hat = x_hat_0
        self._p = p_0
        self.Q = 0.00001
        self.R = 0.01

    @property
    def x_hat(self) -> float:
        return self._x_hat

    def __call__(self, z: float, h: float = 1.0) -> float:
        """One discrete step: update the estimate from measurement ``z``
        with measurement model ``h``."""
        x_prior = self._x_hat
        p_prior = self._p + self.Q
        gain = (p_prior * h) / (h * p_prior * h + self.R)
        self._x_hat = x_prior + gain * (z - h * x_prior)
        self._p = (1.0 - gain * h) * p_prior
        return self._x_hat


class AdaptiveIntegralXupController:
    """Adaptive integral controller over a speedup actuator.

    ``reference`` is the set point; the control signal ``u`` is a speedup
    factor in ``[1, u_max]``. The Kalman-filtered base workload replaces the
    fixed integral gain; ``pole`` in ``[0, 1)`` trades responsiveness for
    noise robustness.
    """

    def __init__(self, reference: float, u_0: float,
                 u_max: float = float("inf"), pole: float = 0.0,
                 kf_kwargs: Optional[dict] = None):
        self.reference = reference
        self._u = u_0
        self._u_max = u_max
        self.pole = pole
        self._kalman = KalmanFilter(**(kf_kwargs or {}))

    @property
    def pole(self) -> float:
        return self._pole

    @pole.setter
    def pole(self, pole: float):
        if not 0 <= pole < 1:
            raise ValueError("pole must be in range [0, 1)")
        self._pole = pole

    def __call__(self, y: float) -> float:
        """Compute the next control signal from measurement ``y``."""
        base_workload = self._kalman(y, h=self._u)
        error = self.reference - y
        u = self._u + (1 - self._pole) * (error / base_workload)
        self._u = max(min(u, self._u_max), 1.0)
        return self._u
