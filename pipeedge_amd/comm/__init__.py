"""Distributed communication (capability parity with reference src/pipeedge/comm/).

The reference transports CPU-staged tensors over gloo/TCP with per-message
dtype/shape headers (reference comm/p2p/__init__.py:96-121). This build keeps
the same lifecycle contracts (`DistContext`, command broadcast, pipeline stage
factory wiring) but re-designs the data plane MI355X-first:

- data plane: torch.distributed point-to-point isend/irecv on the "nccl"
  backend (= RCCL over xGMI on ROCm) between GPU-resident buffers ("gloo" on
  CPU-only hosts, same code path);
- wire format: NO per-message headers - tensor dtypes/shapes are a static
  schema (`TensorSpec`) computed from (model, partition, ubatch, quant) when
  the schedule is distributed;
- control plane: a dedicated gloo subgroup carrying CMD_* broadcasts
  (schedule, batch announcements, stop); batch sizes are announced ahead of
  the data so every rank posts exactly the right number of recvs (no
  abandoned requests at shutdown - RCCL communicators shut down cleanly).
"""
import dataclasses
from typing import Callable, Tuple

import torch

#: callback type for received commands: (cmd, tensors)
DistCmdHandler = Callable[[int, Tuple[torch.Tensor, ...]], None]

# command values (superset of reference runtime.py:36-37)
CMD_STOP = 0
CMD_SCHED = 1
CMD_BATCH = 2


@dataclasses.dataclass(frozen=True)
class TensorSpec:
    """Static wire schema entry: one tensor's shape (incl. microbatch dim) and dtype."""
    shape: Tuple[int, ...]
    dtype: torch.dtype

    def alloc(self, device) -> torch.Tensor:
        return torch.empty(self.shape, dtype=self.dtype, device=device)

    def nbytes(self) -> int:
        n = 1
        for s in self.shape:
            n *= s
        return n * self.dtype.itemsize


class DistContext:
    """Parent class for distributed context managers (reference comm/__init__.py:7-32)."""

    def __init__(self, world_size: int, rank: int):
        self._world_size = world_size
        self._rank = rank
        self._initialized = False

    @property
    def rank(self) -> int:
        return self._rank

    @property
    def world_size(self) -> int:
        return self._world_size

    def init(self) -> None:
        self._initialized = True

    def shutdown(self) -> None:
        self._initialized = False

    def __enter__(self):
        self.init()
        return self

    def __exit__(self, exc_type, exc_val, exc_tb):
        if exc_type is not None:
            # surface the error BEFORE shutdown - the shutdown barrier can
            # block on peers, hiding the root cause
            import logging
            import traceback
            logging.getLogger(__name__).error(
                "exception in distributed context:\n%s",
                "".join(traceback.format_exception(exc_type, exc_val, exc_tb)))
        self.shutdown()
