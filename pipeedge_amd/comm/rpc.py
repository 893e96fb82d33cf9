"""RPC-based pipeline transport (parity: reference comm/rpc/__init__.py).

The alternative ``-c rpc`` transport: ``torch.distributed.rpc`` (TensorPipe)
with remote-owned stages linked by RRefs and semaphore backpressure
(reference rpc/__init__.py:42-121). Activations are CPU-staged over TCP like
the reference; the RCCL p2p engine (comm/p2p.py) is the performance path on
MI355X - this transport exists for capability parity and for control-plane
apps like the reverse auction (revauct.py).

Note (parity): RPC does NOT guarantee microbatch ordering
(reference rpc/__init__.py:44).
"""
import logging
import threading
from typing import Any, Callable, List, Optional, Tuple

import torch
from torch.distributed import rpc

from . import DistContext

logger = logging.getLogger(__name__)


def tensorpipe_rpc_backend_options_factory(*args, **kwargs):
    return rpc.TensorPipeRpcBackendOptions(*args, **kwargs)


class DistRpcContext(DistContext):
    """Context manager for torch.distributed.rpc lifecycles
    (reference rpc/__init__.py:15-39)."""

    def __init__(self, world_size: int = None, rank: int = None,
                 name: Optional[str] = None, **rpc_kwargs):
        # accept both (world_size, rank) and the reference's tuple/dict style
        if isinstance(world_size, (tuple, list)):
            # reference style: DistRpcContext((name,), {'world_size': W, 'rank': R})
            name = world_size[0]
            kwargs = rank or {}
            world_size = kwargs["world_size"]
            rank = kwargs["rank"]
        super().__init__(world_size, rank)
        self._name = name or f"worker{rank}"
        self._rpc_kwargs = rpc_kwargs

    @property
    def name(self) -> str:
        return self._name

    def init(self):
        super().init()
        rpc.init_rpc(self._name, rank=self._rank,
                     world_size=self._world_size, **self._rpc_kwargs)
        return self

    def shutdown(self):
        if not self._initialized:
            return
        rpc.shutdown()
        super().shutdown()

    def cmd_broadcast(self, remote_cmd_handler: Callable, cmd: int,
                      tensors: Optional[Tuple[torch.Tensor, ...]] = None):
        """Broadcast a command to all other ranks (rpc/__init__.py:28-39)."""
        futs = []
        for r in range(self._world_size):
            if r == self._rank:
                continue
            futs.append(rpc.rpc_async(r, remote_cmd_handler,
                                      args=(cmd, tensors or ())))
        torch.futures.wait_all(futs)


class DistRpcPipelineStage:
    """One remote-owned pipeline stage (reference rpc/__init__.py:42-121).

    Backpressure: ``_sem_fwd`` caps in-flight microbatches per stage (3),
    ``_sem_mod`` serializes the module (1); upstream blocks in
    ``wait_for_ready`` before pushing (reference rpc/__init__.py:53-99).
    """

    def __init__(self, module_cls, module_args=(), module_kwargs=None,
                 stage_id: Optional[int] = None):
        self._module = module_cls(*module_args, **(module_kwargs or {}))
        self._stage_id = stage_id
        self._next_rref: Optional[rpc.RRef] = None
        self._results_to: Optional[int] = None
        self._results_cb: Optional[Callable] = None
        self._sem_fwd = threading.Semaphore(3)
        self._sem_mod = threading.Semaphore(1)

    def module_to(self, *args, **kwargs):
        self._module = self._module.to(*args, **kwargs)

    def set_next(self, stage_rref: rpc.RRef):
        self._next_rref = stage_rref

    def set_results(self, results_to: int, results_cb: Callable):
        self._results_to = results_to
        self._results_cb = results_cb

    def wait_for_ready(self):
        """Block the (upstream) caller until this stage can accept work."""
        self._sem_fwd.acquire()  # released when the microbatch leaves

    def mod_register_buffer(self, *args, **kwargs):
        return self._module.register_buffer(*args, **kwargs)

    def mod_register_forward_pre_hook(self, *args, **kwargs):
        return self._module.register_forward_pre_hook(*args, **kwargs)

    def mod_register_forward_hook(self, *args, **kwargs):
        return self._module.register_forward_hook(*args, **kwargs)

    def __call__(self, inputs: Any):
        """Process one microbatch and forward it down the pipeline."""
        try:
            with self._sem_mod:
                device = next(self._module.parameters(), torch.empty(0)).device
                dtype = getattr(self._module, "compute_dtype", None)
                # a quantized wire (5-tensor groups headed by a uint8 comm
                # tensor) must keep its f32 scale/shift exactly - the decode
                # pre-hook casts AFTER dequantization
                is_wire = (isinstance(inputs, tuple) and len(inputs) >= 5
                           and len(inputs) % 5 == 0
                           and inputs[0].dtype == torch.uint8)

                def to_dev(t):
                    t = t.to(device)
                    if not is_wire and dtype is not None \
                            and t.is_floating_point():
                        t = t.to(dtype)
                    return t
                if isinstance(inputs, torch.Tensor):
                    inputs_dev = to_dev(inputs)
                else:
                    inputs_dev = tuple(to_dev(t) for t in inputs)
                outputs = self._module(inputs_dev)
                # CPU-stage for TensorPipe transport
                if isinstance(outputs, torch.Tensor):
                    outputs = outputs.cpu()
                else:
                    outputs = tuple(t.cpu() for t in outputs)
            if self._next_rref is not None:
                self._next_rref.rpc_sync().wait_for_ready()
                self._next_rref.rpc_async().__call__(outputs)
            else:
                assert self._results_to is not None
                rpc.rpc_sync(self._results_to, self._results_cb,
                             args=(outputs,))
        finally:
            self._sem_fwd.release()


class DistRpcPipeline:
    """A chain of remote pipeline stages (reference rpc/__init__.py:122-163)."""

    def __init__(self, stage_rrefs: List[rpc.RRef], results_to: int,
                 results_cb: Callable):
        self._rref_list = stage_rrefs
        self._link_pipeline(results_to, results_cb)

    def rpc_register_buffer(self, name: str, tensors: List[Optional[torch.Tensor]],
                            **kwargs):
        futs = [rref.rpc_async().mod_register_buffer(name, tensor, **kwargs)
                for rref, tensor in zip(self._rref_list, tensors)]
        torch.futures.wait_all(futs)

    def rpc_register_forward_pre_hook(self, hook: Callable, first: bool = True):
        rrefs = self._rref_list if first else self._rref_list[1:]
        futs = [rref.rpc_async().mod_register_forward_pre_hook(hook)
                for rref in rrefs]
        torch.futures.wait_all(futs)

    def rpc_register_forward_hook(self, hook: Callable, last: bool = True):
        rrefs = self._rref_list if last else self._rref_list[:-1]
        futs = [rref.rpc_async().mod_register_forward_hook(hook)
                for rref in rrefs]
        torch.futures.wait_all(futs)

    def _link_pipeline(self, results_to: int, results_cb: Callable):
        n_stages = len(self._rref_list)
        futs = [self._rref_list[i].rpc_async().set_next(self._rref_list[i + 1])
                for i in range(n_stages - 1)]
        futs.append(self._rref_list[-1].rpc_async().set_results(results_to,
                                                                results_cb))
        torch.futures.wait_all(futs)

    def enqueue_tensor(self, tensor: torch.Tensor):
        """Push one microbatch into the first stage (asynchronous)."""
        self._rref_list[0].rpc_sync().wait_for_ready()
        self._rref_list[0].rpc_async().__call__(tensor)


# ---------------------------------------------------------------------------
# runtime driver for `-c rpc` (reference runtime.py:514-578)
# ---------------------------------------------------------------------------

def run_pipeline_rpc(world_size: int, rank: int, model_name: str,
                     model_file, batch_size: int, ubatch_size: int,
                     partition, quant_list, rank_order, data_rank: int,
                     hosts, dataset_cfg: dict, sched_models_file,
                     sched_dev_types_file, sched_dev_file,
                     rpc_num_worker_threads: int = 16,
                     num_batches: int = 1) -> None:
    import monitoring as monitoring_glue  # root-level wrapper
    import model_cfg
    import runtime as runtime_mod

    monitoring_glue.init(runtime_mod.MONITORING_KEY_MODEL,
                         runtime_mod.get_window_size(), work_type="tensors",
                         acc_type="layers")
    monitoring_glue.add_key(runtime_mod.MONITORING_KEY_OUTPUT,
                            work_type="classifications", acc_type="correct")
    monitoring_glue.add_key(runtime_mod.MONITORING_KEY_QUANT_DECODE,
                            work_type="tensors", acc_type="bits")
    monitoring_glue.add_key(runtime_mod.MONITORING_KEY_QUANT_ENCODE,
                            work_type="tensors", acc_type="bits")
    monitoring_glue.add_key(runtime_mod.MONITORING_KEY_RECV, work_type="Mbits")
    monitoring_glue.add_key(runtime_mod.MONITORING_KEY_SEND, work_type="Mbits")
    opts = tensorpipe_rpc_backend_options_factory(
        num_worker_threads=rpc_num_worker_threads)
    with DistRpcContext(world_size, rank,
                        rpc_backend_options=opts) as ctx:
        if rank == data_rank:
            device = None  # stages place themselves
            stage_layers, stage_quant, stage_ranks = \
                runtime_mod.get_pipeline_sched(
                    world_size, hosts, partition, quant_list, rank_order,
                    model_name, ubatch_size, "torch.float32",
                    sched_models_file, sched_dev_types_file, sched_dev_file)
            pipeline = model_cfg.dist_rpc_pipeline_factory(
                model_name, model_file, stage_ranks, stage_layers, rank,
                runtime_mod.handle_results)
            # quant + monitoring wiring (reference runtime.py:558-565):
            # per-stage quant_bit / rate_constraint buffers, monitor hooks on
            # every stage, quant encode on all-but-last, decode on
            # all-but-first
            import os
            send_constraint = float(
                os.getenv(runtime_mod.ENV_SEND_CONSTRAINT, "0"))
            pipeline.rpc_register_buffer(
                "quant_bit", [torch.tensor(q) for q in stage_quant],
                persistent=False)
            pipeline.rpc_register_buffer(
                "rate_constraint",
                [torch.tensor(send_constraint) for _ in stage_quant],
                persistent=False)
            pipeline.rpc_register_forward_hook(
                runtime_mod.forward_hook_monitor)
            pipeline.rpc_register_forward_hook(
                runtime_mod.forward_hook_quant_encode, last=False)
            # send/recv monitors see wire sizes: send AFTER encode, recv
            # BEFORE decode (torch runs hooks in registration order)
            pipeline.rpc_register_forward_hook(
                runtime_mod.forward_hook_rpc_send_monitor, last=False)
            pipeline.rpc_register_forward_pre_hook(
                runtime_mod.forward_pre_hook_rpc_recv_monitor, first=False)
            pipeline.rpc_register_forward_pre_hook(
                runtime_mod.forward_pre_hook_quant_decode, first=False)
            pipeline.rpc_register_forward_pre_hook(
                runtime_mod.forward_pre_hook_monitor)
            dataset = runtime_mod.load_dataset(dataset_cfg, model_name,
                                               batch_size, ubatch_size)
            start = runtime_mod.results_counter.value
            import time
            tik = time.time()
            for _ in range(num_batches):
                for u0 in range(0, len(dataset), ubatch_size):
                    items = [dataset[i][0] for i in
                             range(u0, min(u0 + ubatch_size, len(dataset)))]
                    pipeline.enqueue_tensor(torch.stack(items))
            runtime_mod.results_counter.wait_gte(
                start + len(dataset) * num_batches)
            latency = time.time() - tik
            throughput = batch_size * num_batches / latency
            logger.info("Latency is %f, throughput is %f items/sec",
                        latency, throughput)
    monitoring_glue.finish()
