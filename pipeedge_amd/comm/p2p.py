"""Point-to-point pipeline transport.

Replaces the reference's gloo/TCP tagged send/recv with per-message headers +
CPU staging + 3-thread stage engine (reference comm/p2p/__init__.py) with an
MI355X-native design:

- GPU-resident activations move rank-to-rank with ``dist.isend``/``irecv`` on
  the "nccl" backend (RCCL over xGMI); on CPU hosts the identical code runs on
  gloo (used by the CPU test suite);
- tensor shapes/dtypes are static per (model, partition, ubatch, quant) and
  are exchanged ONCE via the control plane, so the data plane carries raw
  payloads only;
- each rank runs a single-threaded event loop with double-buffered recv slots
  (RCCL transfers overlap compute via their own HIP streams; a recorded event
  guards slot reuse) instead of recv/work/send Python threads;
- batches are announced on the control plane (CMD_BATCH, microbatch count)
  before any data moves, so every rank posts exactly the matching number of
  recvs and shutdown never abandons an in-flight RCCL op.

The control plane is a dedicated gloo subgroup: tiny, latency-insensitive
CPU tensors for CMD_SCHED / CMD_BATCH / CMD_STOP (reference runtime.py:36-44).
"""
import collections
import logging
import threading
from typing import Callable, List, Optional, Sequence

import torch
import torch.distributed as dist

from . import CMD_STOP, DistCmdHandler, DistContext, TensorSpec
from ..models import ModuleShardData

logger = logging.getLogger(__name__)

TAG_CMD = 10
TAG_CMD_META = 11
TAG_CMD_DATA = 12
TAG_DATA = 1

_DTYPE_CODES = [torch.float32, torch.float64, torch.float16, torch.bfloat16,
                torch.uint8, torch.int8, torch.int16, torch.int32, torch.int64,
                torch.bool]
_DTYPE_TO_CODE = {d: i for i, d in enumerate(_DTYPE_CODES)}
_CMD_META_LEN = 10  # dtype_code, ndim, up to 8 dims


class CommandThread(threading.Thread):
    """Background listener for control-plane commands (any source).

    Mirrors the reference's CommandThread (comm/p2p/__init__.py:298-331) but
    on a dedicated gloo group; the sender's rank rides in the header so
    follow-up payloads can be received from the right peer.
    """

    def __init__(self, group, cmd_cb: DistCmdHandler, stop_event: threading.Event):
        super().__init__(daemon=True, name="pipeedge-cmd")
        self._group = group
        self._cmd_cb = cmd_cb
        self._stop_event = stop_event

    def run(self):
        while not self._stop_event.is_set():
            hdr = torch.zeros(3, dtype=torch.int64)
            req = dist.irecv(hdr, src=None, group=self._group, tag=TAG_CMD)
            # gloo's is_completed() does not fire for any-source recvs (same
            # issue the reference works around, p2p/util.py:8-24): block in a
            # short-lived daemon and poll it against stop_event instead.
            waiter = threading.Thread(target=req.wait, daemon=True)
            waiter.start()
            while waiter.is_alive():
                if self._stop_event.is_set():
                    return  # abandoned gloo irecv; daemon thread, harmless
                waiter.join(timeout=0.005)
            cmd, n_tensors, src = int(hdr[0]), int(hdr[1]), int(hdr[2])
            tensors = []
            for _ in range(n_tensors):
                meta = torch.zeros(_CMD_META_LEN, dtype=torch.int64)
                dist.recv(meta, src=src, group=self._group, tag=TAG_CMD_META)
                dtype = _DTYPE_CODES[int(meta[0])]
                shape = [int(d) for d in meta[2:2 + int(meta[1])]]
                payload = torch.zeros(shape, dtype=dtype)
                dist.recv(payload, src=src, group=self._group, tag=TAG_CMD_DATA)
                tensors.append(payload)
            try:
                self._cmd_cb(cmd, tuple(tensors))
            except Exception:  # noqa: BLE001
                logger.exception("command handler failed for cmd=%d", cmd)
            if cmd == CMD_STOP:
                break


class DistP2pContext(DistContext):
    """Distributed context for the p2p pipeline.

    ``backend`` defaults to "nccl" (RCCL) when a GPU is visible, else "gloo".
    """

    def __init__(self, world_size: int, rank: int, cmd_cb: DistCmdHandler,
                 backend: Optional[str] = None):
        super().__init__(world_size, rank)
        if backend is None:
            # PIPEEDGE_P2P_BACKEND=gloo forces a CPU-staged wire with GPU
            # compute - lets N pipeline stages share one GPU for validation
            # (RCCL forbids two ranks per device)
            import os
            backend = os.environ.get(
                "PIPEEDGE_P2P_BACKEND",
                "nccl" if torch.cuda.is_available() else "gloo")
        self._backend = backend
        self._cmd_cb = cmd_cb
        self._ctrl_group = None
        self._cmd_thread = None
        self.stop_event = threading.Event()

    @property
    def ctrl_group(self):
        return self._ctrl_group

    def init(self):
        super().init()
        dist.init_process_group(self._backend, rank=self._rank,
                                world_size=self._world_size)
        # control plane always on gloo (CPU tensors, any-source recv, tags)
        self._ctrl_group = dist.new_group(backend="gloo")
        # dedicated communicator for the results back-edge (see
        # DistP2pPipelineStage)
        self.results_group = dist.new_group(backend=self._backend)
        # Eagerly initialize the subgroup communicator: NCCL comm init is
        # COLLECTIVE over the group, but only the data rank and the last
        # stage ever issue ops on this group - lazy first-use init would
        # wait forever for the other ranks. A one-time barrier makes every
        # rank participate in the init.
        dist.barrier(group=self.results_group)
        self._cmd_thread = CommandThread(self._ctrl_group, self._cmd_cb,
                                         self.stop_event)
        self._cmd_thread.start()
        return self

    def shutdown(self):
        if not self._initialized:
            return
        self.stop_event.set()
        self._cmd_thread.join(timeout=5.0)
        dist.barrier(group=self._ctrl_group)
        dist.destroy_process_group()
        super().shutdown()

    def cmd_broadcast(self, cmd: int, tensors: Sequence[torch.Tensor] = ()):
        """Send a command (+ small CPU tensors) to every *other* rank."""
        tensors = [t.cpu() for t in tensors]
        hdr = torch.tensor([cmd, len(tensors), self._rank], dtype=torch.int64)
        metas = []
        for t in tensors:
            assert t.dim() <= 8
            meta = torch.zeros(_CMD_META_LEN, dtype=torch.int64)
            meta[0] = _DTYPE_TO_CODE[t.dtype]
            meta[1] = t.dim()
            for i, d in enumerate(t.shape):
                meta[2 + i] = d
            metas.append(meta)
        for dst in range(self._world_size):
            if dst == self._rank:
                continue
            dist.send(hdr, dst=dst, group=self._ctrl_group, tag=TAG_CMD)
            for meta, t in zip(metas, tensors):
                dist.send(meta, dst=dst, group=self._ctrl_group, tag=TAG_CMD_META)
                dist.send(t.contiguous(), dst=dst, group=self._ctrl_group,
                          tag=TAG_CMD_DATA)


def _default_recv_transform(tensors: List[torch.Tensor]) -> ModuleShardData:
    return tensors[0] if len(tensors) == 1 else tuple(tensors)


def _default_send_transform(data: ModuleShardData) -> List[torch.Tensor]:
    return [data] if isinstance(data, torch.Tensor) else list(data)


class _RecvPipeline:
    """Double-buffered in-order irecv stream from one peer."""

    def __init__(self, src: int, specs: Sequence[TensorSpec], device,
                 depth: int = 2, group=None):
        self.src = src
        self.group = group
        self.device = torch.device(device)
        self.specs = list(specs)
        self.depth = depth
        self.slots = [[s.alloc(self.device) for s in self.specs]
                      for _ in range(depth)]
        self.events = [None] * depth
        self.works = {}
        self.next_post = 0
        self.use_events = self.device.type == "cuda"

    def post(self, upto: int, n: int):
        """Post irecvs for microbatches < upto (bounded by slot availability)."""
        while self.next_post < min(upto, n):
            i = self.next_post
            s = i % self.depth
            if self.events[s] is not None:
                self.events[s].synchronize()
                self.events[s] = None
            self.works[i] = [dist.irecv(t, src=self.src, group=self.group,
                                        tag=TAG_DATA)
                             for t in self.slots[s]]
            self.next_post += 1

    def wait(self, i: int) -> List[torch.Tensor]:
        for w in self.works.pop(i):
            w.wait()
        return self.slots[i % self.depth]

    def mark_consumed(self, i: int):
        """Record an event after the consumer launched its reads of slot i."""
        s = i % self.depth
        if self.use_events:
            ev = torch.cuda.Event()
            ev.record()
            self.events[s] = ev


class _SendWindow:
    """Outstanding isend tracker: keeps tensors alive, bounds in-flight count."""

    def __init__(self, dst: int, max_inflight: int = 8, group=None):
        self.dst = dst
        self.group = group
        self.max_inflight = max_inflight
        self.pending = collections.deque()

    def send(self, tensors: List[torch.Tensor]):
        while len(self.pending) >= self.max_inflight:
            works, _ = self.pending.popleft()
            for w in works:
                w.wait()
        tensors = [t.contiguous() for t in tensors]
        works = [dist.isend(t, dst=self.dst, group=self.group, tag=TAG_DATA)
                 for t in tensors]
        self.pending.append((works, tensors))

    def drain(self):
        while self.pending:
            works, _ = self.pending.popleft()
            for w in works:
                w.wait()


class DistP2pPipelineStage:
    """One rank's role in the pipeline (stage, data rank, or both).

    Parity with reference DistP2pPipelineStage (comm/p2p/__init__.py:334-450),
    re-shaped for the static-schema single-threaded engine:

    - ``rank_src`` / ``rank_dst``: upstream / downstream peers (None at chain
      ends), with the reference's data-rank wiring semantics
      (reference model_cfg.py:128-166);
    - ``recv_specs`` / ``send_specs``: static wire schemas for this rank's
      inbound / outbound edges;
    - ``work_cb``: the module shard (None for a pure data rank);
    - ``results_cb``: output consumer (data rank only).
    """

    def __init__(self, rank_src: Optional[int], rank_dst: Optional[int],
                 recv_specs: Sequence[TensorSpec], send_specs: Sequence[TensorSpec],
                 work_cb: Optional[Callable], results_cb: Optional[Callable],
                 device=None, recv_group=None, send_group=None):
        # Separate process groups break the pipeline's only CYCLE (the
        # results back-edge last-stage -> data-rank): with a single NCCL
        # communicator, the data rank's pre-posted result irecvs could
        # serialize ahead of its input isends on the comm stream and
        # deadlock the ring. The forward chain is acyclic and safe on one
        # group; the back-edge gets its own communicator.
        self.rank_src = rank_src
        self.rank_dst = rank_dst
        self.recv_group = recv_group
        self.send_group = send_group
        self.recv_specs = list(recv_specs) if recv_specs else []
        self.send_specs = list(send_specs) if send_specs else []
        self.work_cb = work_cb
        self.results_cb = results_cb
        self.device = torch.device(device) if device is not None else \
            (torch.device("cuda", torch.cuda.current_device())
             if torch.cuda.is_available() else torch.device("cpu"))
        # Wire device: where tensors live for dist send/recv. RCCL moves
        # GPU-resident buffers directly; gloo requires CPU staging, so a
        # gloo data plane with GPU compute (e.g. multiple stages sharing
        # one GPU for validation) stages through pinned-less CPU copies.
        self.wire_device = self.device
        if self.device.type == "cuda" and dist.is_initialized() and \
                dist.get_backend() == "gloo":
            self.wire_device = torch.device("cpu")
        # transforms (quantization hooks attach here)
        self.recv_transform = _default_recv_transform
        self.send_transform = _default_send_transform
        # monitoring hooks
        self._recv_pre_hook = None
        self._recv_post_hook = None
        self._send_pre_hook = None
        self._send_post_hook = None

    # -- hook registration (parity: reference p2p/__init__.py:411-433) -------
    def register_recv_pre_hook(self, hook, args=()):
        self._recv_pre_hook = (hook, args)

    def register_recv_post_hook(self, hook, args=()):
        self._recv_post_hook = (hook, args)

    def register_send_pre_hook(self, hook, args=()):
        self._send_pre_hook = (hook, args)

    def register_send_post_hook(self, hook, args=()):
        self._send_post_hook = (hook, args)

    @staticmethod
    def _call_hook(hook_entry, *extra):
        if hook_entry is not None:
            hook, args = hook_entry
            hook(*args, *extra)

    # -- microbatch stream overlap -------------------------------------------
    def _ustreams(self):
        """Two HIP streams alternate over consecutive microbatches so their
        kernels co-reside on the chip (the per-microbatch kernels underfill
        256 CUs; this is the stream-level analogue of the reference's 3-deep
        in-flight stage queues, comm/p2p/__init__.py:374-376). Disable with
        PIPEEDGE_UBATCH_STREAMS=0."""
        import os
        n = int(os.environ.get("PIPEEDGE_UBATCH_STREAMS", "2"))
        if self.device.type != "cuda" or n < 2:
            return None
        if not hasattr(self, "_ustream_pair"):
            self._ustream_pair = [torch.cuda.Stream() for _ in range(n)]
        return self._ustream_pair

    @staticmethod
    def _stream_ctx(streams, i):
        import contextlib
        if streams is None:
            return contextlib.nullcontext()
        return torch.cuda.stream(streams[i % len(streams)])

    _offset_done = False

    def _maybe_offset_stream(self, streams, i):
        """One-time de-phasing of the later microbatch streams: without it
        both streams start IDENTICAL per-microbatch kernel chains at the
        same instant and stay phase-locked - both sit in the low-occupancy
        LayerNorm/attention phases simultaneously. A one-time spin delay of
        ~half a chain pairs one stream's GEMM phases against the other's
        small-kernel phases. PIPEEDGE_STREAM_OFFSET_US (default 0 = off)."""
        import os
        if streams is None or self._offset_done or i == 0:
            return
        if i >= len(streams):
            self._offset_done = True
            return
        us = int(os.environ.get("PIPEEDGE_STREAM_OFFSET_US", "0"))
        if us > 0:
            # ~2.0e3 device cycles per us on MI355X; exactness irrelevant
            torch.cuda._sleep(int(us * 2000) * i)
        if i == len(streams) - 1:
            self._offset_done = True

    # -- stage-rank batch processing -----------------------------------------
    def _from_wire(self, tensors: List[torch.Tensor]) -> List[torch.Tensor]:
        if self.wire_device == self.device:
            return tensors
        return [t.to(self.device) for t in tensors]

    def _to_wire(self, tensors: List[torch.Tensor]) -> List[torch.Tensor]:
        if self.wire_device == self.device:
            return tensors
        return [t.to(self.wire_device) for t in tensors]

    def stage_run_batch(self, n_ubatch: int):
        """Process one announced batch: exactly ``n_ubatch`` recv->work->send."""
        assert self.work_cb is not None
        streams = self._ustreams()
        depth = max(2, len(streams) if streams else 2)
        recv = _RecvPipeline(self.rank_src, self.recv_specs, self.wire_device,
                             depth=depth, group=self.recv_group) \
            if self.rank_src is not None else None
        send = _SendWindow(self.rank_dst, group=self.send_group) \
            if self.rank_dst is not None else None
        if recv is not None:
            recv.post(depth, n_ubatch)
        for i in range(n_ubatch):
            with self._stream_ctx(streams, i):
                self._maybe_offset_stream(streams, i)
                self._call_hook(self._recv_pre_hook)
                wire_in = self._from_wire(recv.wait(i))
                self._call_hook(self._recv_post_hook, wire_in)
                inputs = self.recv_transform(wire_in)
                outputs = self.work_cb(inputs)
                recv.mark_consumed(i)
                wire_out = self._to_wire(self.send_transform(outputs))
                self._call_hook(self._send_pre_hook)
                send.send(wire_out)
                self._call_hook(self._send_post_hook, wire_out)
            recv.post(i + depth + 1, n_ubatch)
        send.drain()
        if streams is not None:
            for s in streams:
                s.synchronize()

    # -- data-rank batch processing ------------------------------------------
    def data_run_batch(self, ubatches: Sequence[ModuleShardData]) -> None:
        """Feed a batch of microbatches into the pipeline and collect all
        results via ``results_cb`` (in order)."""
        assert self.results_cb is not None
        n = len(ubatches)
        if self.work_cb is not None and self.rank_dst is None:
            # single-rank degenerate pipeline (reference runtime.py:324-329)
            streams = self._ustreams()
            for i, ub in enumerate(ubatches):
                with self._stream_ctx(streams, i):
                    self._maybe_offset_stream(streams, i)
                    self.results_cb(self.work_cb(self._to_device(ub)))
            if streams is not None:
                for s in streams:
                    s.synchronize()
            return
        send = _SendWindow(self.rank_dst, group=self.send_group)
        recv = _RecvPipeline(self.rank_src, self.recv_specs, self.wire_device,
                             group=self.recv_group)
        streams = self._ustreams()
        recv.post(2, n)
        collected = 0
        # ADMISSION CONTROL - the results back-edge closes a CYCLE (data
        # rank -> stages -> data rank). If more microbatches are admitted
        # than the chain can buffer (~stages x (send window 8 + recv slots
        # 2)), every rank ends up blocked on a full send window while the
        # data rank itself is blocked sending instead of draining results:
        # a distributed deadlock (observed at ~83 in-flight on an 8-stage
        # continuous stream). Cap admissions well under capacity - steady
        # state only needs ~2x pipeline depth in flight.
        depth_cap = max(8, 4 * int(getattr(self, "pipeline_depth", 8)))
        for i, ub in enumerate(ubatches):
            while i - collected >= depth_cap:
                collected = self._consume_result(recv, collected, n)
            with self._stream_ctx(streams, i):
                if self.work_cb is not None:
                    outputs = self.work_cb(self._to_device(ub))
                    wire = self._to_wire(self.send_transform(outputs))
                else:
                    wire = self._to_wire(
                        _default_send_transform(self._to_device(ub)))
                self._call_hook(self._send_pre_hook)
                send.send(wire)
                self._call_hook(self._send_post_hook, wire)
            # opportunistically drain finished results to bound buffering
            while (collected < i and collected in recv.works and
                   all(w.is_completed() for w in recv.works[collected])):
                collected = self._consume_result(recv, collected, n)
        while collected < n:
            collected = self._consume_result(recv, collected, n)
        send.drain()

    def _consume_result(self, recv: "_RecvPipeline", j: int, n: int) -> int:
        self._call_hook(self._recv_pre_hook)
        wire = recv.wait(j)
        self._call_hook(self._recv_post_hook, wire)
        result = self.recv_transform(wire)
        # results are handed off: clone so slot reuse can't alias
        if isinstance(result, torch.Tensor):
            result = result.clone()
        else:
            result = tuple(t.clone() for t in result)
        self.results_cb(result)
        recv.mark_consumed(j)
        recv.post(j + 3, n)
        return j + 1

    def _to_device(self, data: ModuleShardData) -> ModuleShardData:
        if isinstance(data, torch.Tensor):
            return data.to(self.device, non_blocking=True)
        return tuple(t.to(self.device, non_blocking=True) for t in data)
