"""Global device configuration (parity: reference devices.py).

The reference staged all communication through CPU memory and used forward
hooks to move tensors CPU<->device around every microbatch
(reference devices.py:8-24). In this build activations stay GPU-resident and
travel over RCCL/xGMI, so the hooks are gone; this module only holds the
process-global DEVICE choice used by shard/stage factories.
"""
from typing import Optional

import torch

DEVICE: Optional[torch.device] = None


def initialize(device: Optional[str] = None) -> torch.device:
    """Resolve and set the global DEVICE (default: cuda if available)."""
    global DEVICE
    if device is not None:
        DEVICE = torch.device(device)
    elif torch.cuda.is_available():
        DEVICE = torch.device("cuda")
    else:
        DEVICE = torch.device("cpu")
    if DEVICE.type == "cuda":
        torch.cuda.init()
        if DEVICE.index is not None:
            torch.cuda.set_device(DEVICE)
    return DEVICE
