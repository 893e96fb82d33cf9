"""hipGraph capture of shard forwards.

Per-microbatch shapes are static in this framework (the same property that
enables the static wire schemas), so each stage's forward is captured once
into a hipGraph (`torch.cuda.CUDAGraph` == hipGraph on ROCm) and replayed
per microbatch - eliminating per-kernel launch overhead for the ~90-kernel
ViT forward. The reference has no analogue (eager HF modules; SURVEY.md
section 2.6).

Inputs are copied into capture-static buffers before replay; outputs are
cloned out after (the send path keeps clones alive while RCCL transfers
run). Capture is keyed by input shapes, so re-scheduling with a different
partition or microbatch size re-captures.
"""
import logging
import os
from typing import Dict, Tuple

import torch

logger = logging.getLogger(__name__)

ENV_DISABLE = "PIPEEDGE_NO_HIPGRAPH"


def graphs_enabled() -> bool:
    return torch.cuda.is_available() and \
        os.environ.get(ENV_DISABLE, "0") != "1"


class GraphedModule:
    """Wraps a module: eager warmup calls, then capture + replay."""

    def __init__(self, module, warmup_iters: int = 3):
        self._module = module
        self._warmup_iters = warmup_iters
        self._graphs: Dict[Tuple, dict] = {}

    def __getattr__(self, name):
        return getattr(self._module, name)

    @staticmethod
    def _key(data):
        # keyed by stream too: concurrent replays on different streams must
        # not share capture-static buffers
        sid = torch.cuda.current_stream().cuda_stream
        if isinstance(data, torch.Tensor):
            return (sid, tuple(data.shape), data.dtype)
        return (sid,) + tuple((tuple(t.shape), t.dtype) for t in data)

    def _capture(self, data):
        static_in = data.clone() if isinstance(data, torch.Tensor) \
            else tuple(t.clone() for t in data)
        # warmup on a side stream (allocator + lazy-init work must not be
        # captured)
        side = torch.cuda.Stream()
        side.wait_stream(torch.cuda.current_stream())
        with torch.cuda.stream(side):
            for _ in range(self._warmup_iters):
                out = self._module(static_in)
        torch.cuda.current_stream().wait_stream(side)
        graph = torch.cuda.CUDAGraph()
        with torch.cuda.graph(graph):
            static_out = self._module(static_in)
        entry = {"graph": graph, "in": static_in, "out": static_out}
        logger.info("captured hipGraph for input %s", self._key(data))
        return entry

    def __call__(self, data):
        key = self._key(data)
        entry = self._graphs.get(key)
        if entry is None:
            entry = self._capture(data)
            self._graphs[key] = entry
        # copy inputs into the capture-static buffers
        if isinstance(data, torch.Tensor):
            entry["in"].copy_(data, non_blocking=True)
        else:
            for dst, src in zip(entry["in"], data):
                dst.copy_(src, non_blocking=True)
        entry["graph"].replay()
        out = entry["out"]
        if isinstance(out, torch.Tensor):
            return out.clone()
        return tuple(t.clone() for t in out)


def maybe_graph(module):
    """Wrap a shard for graph replay when running on GPU (no-op on CPU)."""
    if graphs_enabled():
        return GraphedModule(module)
    return module
