"""Pipeline-parallel inference runtime CLI (parity: reference runtime.py).

Same CLI surface as the reference (positional rank/worldsize; -d/-s/--addr/
--port/-c/-w/-m/-M/-b/-u; dataset group; user-scheduling -pt/-q/-r/-D;
auto-scheduling -H/-sm/-sdt/-sd) with MI355X-native internals:

- one process per GPU; activations stay GPU-resident and move over RCCL/xGMI
  ("nccl" backend) - gloo only for the CPU control plane and CPU-only hosts;
- the schedule broadcast carries everything needed for each rank to derive
  its static wire schemas (no per-message headers);
- quantization (QuantPipe) runs as stage send/recv transforms with a wire
  payload sized for the configured max bitwidth, so adaptive bitwidth changes
  never change the schema.
"""
import argparse
import logging
import os
import queue
import sys
import threading
import time
from typing import List, Optional, Tuple

import numpy as np
import torch

import devices
import model_cfg
import monitoring
from pipeedge_amd import models, quant
from pipeedge_amd.comm import CMD_BATCH, CMD_SCHED, CMD_STOP
from pipeedge_amd.comm.p2p import DistP2pContext
from pipeedge_amd.models import spec as model_spec
from pipeedge_amd.quant import (clamp_banner2019_gelu, clamp_banner2019_laplace,
                                tensor_decode_outerdim, tensor_encode_outerdim)
from pipeedge_amd.sched.scheduler import sched_pipeline
from pipeedge_amd.utils.data import load_dataset_synthetic
from pipeedge_amd.utils.threads import ThreadSafeCounter

logger = logging.getLogger(__name__)

# monitoring keys (parity: reference runtime.py:54-59)
MONITORING_KEY_MODEL = "shard"
MONITORING_KEY_OUTPUT = "output"
MONITORING_KEY_QUANT_DECODE = "quant_decode"
MONITORING_KEY_QUANT_ENCODE = "quant_encode"
MONITORING_KEY_RECV = "recv"
MONITORING_KEY_SEND = "send"

ENV_WINDOW_SIZE = "WINDOW_SIZE"
WINDOW_SIZE_DEFAULT = 10

ENV_SEND_CONSTRAINT = "SEND_CONSTRAINT"
ENV_ADAPTIVE_QUANT = "ADAPTIVE_QUANT"
ADAPTIVE_QUANT_HEURISTIC = "HEURISTIC"
ADAPTIVE_QUANT_HEURISTIC2 = "HEURISTIC2"
ADAPTIVE_QUANT_CONTROLLER = "CONTROLLER"


def get_window_size() -> int:
    return int(os.getenv(ENV_WINDOW_SIZE, str(WINDOW_SIZE_DEFAULT)))


# ---------------------------------------------------------------------------
# command handling / schedule distribution
# ---------------------------------------------------------------------------

sched_q = queue.Queue()
batch_q = queue.Queue()
stop_event = threading.Event()


def handle_cmd(cmd: int, tensors: Tuple[torch.Tensor, ...]) -> None:
    """Process control-plane commands (parity: reference runtime.py:406-415)."""
    if cmd == CMD_STOP:
        logger.info("handle_cmd: stop")
        stop_event.set()
    elif cmd == CMD_SCHED:
        logger.info("handle_cmd: sched")
        sched_q.put((tensors[0].tolist(),      # stage layers
                     tensors[1].tolist(),      # stage quant
                     tensors[2].tolist(),      # stage ranks
                     int(tensors[3].item()),   # data rank
                     int(tensors[4].item())))  # seq len (0 = model default)
    elif cmd == CMD_BATCH:
        batch_q.put(int(tensors[0].item()))
    else:
        logger.warning("handle_cmd: unknown command: %d", cmd)


def parse_yaml_sched(sched: List[dict], hosts: Optional[List[str]]) \
        -> Tuple[List[Tuple[int, int]], List[int]]:
    """Turn scheduler YAML output into (stage_layers, stage_ranks)
    (parity: reference runtime.py:260-289)."""
    stage_layers = []
    stage_ranks = []
    if len(sched) == 0:
        raise RuntimeError("No viable schedule found")
    for stage in sched:
        assert len(stage) == 1
        host = list(stage.keys())[0]
        layers = stage[host]
        stage_layers.append((int(layers[0]), int(layers[1])))
        if hosts:
            try:
                stage_ranks.append(hosts.index(host))
            except ValueError:
                logger.error("Scheduling: host not found in hosts list: %s", host)
                raise
        else:
            try:
                stage_ranks.append(int(host))
            except ValueError:
                logger.error("Scheduling: 'hosts' not specified and host is "
                             "not an int: %s", host)
                raise
    return stage_layers, stage_ranks


def get_pipeline_sched(world_size: int, hosts: Optional[List[str]],
                       partition: Optional[List[Tuple[int, int]]],
                       quant: Optional[List[int]],
                       rank_order: Optional[List[int]], model_name: str,
                       ubatch_size: int, dtype_str: str,
                       s_models_file: Optional[str],
                       s_dev_types_file: Optional[str],
                       s_dev_file: Optional[str]) \
        -> Tuple[List[Tuple[int, int]], List[int], List[int]]:
    """Resolve the pipeline schedule (parity: reference runtime.py:291-356)."""
    if partition:
        stage_layers = partition
        stage_quant = quant if quant else [0] * len(stage_layers)
        stage_ranks = rank_order if rank_order else list(range(len(stage_layers)))
    elif world_size <= 1:
        # degenerate single-stage pipeline (reference runtime.py:324-329)
        stage_layers = [(1, model_cfg.get_model_layers(model_name))]
        stage_quant = quant if quant else [0]
        stage_ranks = rank_order if rank_order else [0]
    elif s_models_file and s_dev_types_file and s_dev_file:
        sched = sched_pipeline(model_name, 2, 2, ubatch_size, dtype=dtype_str,
                               models_file=s_models_file,
                               dev_types_file=s_dev_types_file,
                               dev_file=s_dev_file)
        stage_layers, stage_ranks = parse_yaml_sched(sched, hosts)
        stage_quant = quant if quant else [0] * len(stage_layers)
    else:
        raise RuntimeError("Scheduling: must specify either -pt or all of "
                           "-sm/-sdt/-sd (or run with world size <= 1)")
    assert len(stage_layers) == len(stage_quant) == len(stage_ranks)
    logger.info("Scheduling: stage-to-layer mapping: %s", stage_layers)
    logger.info("Scheduling: stage output quantization: %s", stage_quant)
    logger.info("Scheduling: stage-to-rank mapping: %s", stage_ranks)
    return stage_layers, stage_quant, stage_ranks


# ---------------------------------------------------------------------------
# quantization stage transforms (replace the reference's forward hooks,
# runtime.py:73-119; wire format unchanged)
# ---------------------------------------------------------------------------

# ---------------------------------------------------------------------------
# adaptive quantization policies (QuantPipe; parity: reference
# runtime.py:121-216). Invoked per microbatch inside the quant send
# transform - the reference registers them as module forward hooks, which is
# the same call point (after the shard computes, before encoding).
# ---------------------------------------------------------------------------

from pipeedge_amd.quant import compression_factor  # noqa: E402
from pipeedge_amd.utils import quant as quantutil  # noqa: E402

# largest bitwidths in [2, 32] with unique discrete compression factors
BITWIDTHS = [i for i in range(32, 1, -1)
             if int(compression_factor(i)) > int(compression_factor(i + 1))]
_MODULE_QUANT_CONTROLLERS = {}
_MODULE_QUANT_CONTROLLERS_LOCK = threading.Lock()


def set_quant_bandwidth_heuristic(module, outputs) -> None:
    """Pick a bitwidth so window send work fits the rate constraint."""
    with monitoring.get_locked_context(MONITORING_KEY_SEND) as mctx:
        tag = mctx.get_tag(key=MONITORING_KEY_SEND)
        window_size = mctx.get_window_size(key=MONITORING_KEY_SEND)
        bandwidth = mctx.get_window_perf(key=MONITORING_KEY_SEND)
        send_work = mctx.get_window_work(key=MONITORING_KEY_SEND)
    if tag == 0 or tag % window_size != 0:
        return
    target_rate = module.rate_constraint.item()
    if target_rate > 0:
        ubatch_size = models.get_microbatch_size(outputs, verify=True)
        target_time = ubatch_size * window_size / target_rate
    else:
        target_time = float("inf")
    target_datasize = target_time * bandwidth
    quant_bit = module.quant_bit.item()
    if target_datasize <= 0:
        return
    if quant_bit > 0:
        compress_ratio = int(send_work * (32 / quant_bit) / target_datasize) + 1
    else:
        compress_ratio = int(send_work / target_datasize) + 1
    for limit, bit in ((1, 0), (2, 16), (4, 8), (5, 6), (8, 4)):
        if compress_ratio <= limit:
            module.quant_bit = torch.tensor(bit)
            break
    else:
        module.quant_bit = torch.tensor(2)
    logger.info("Adaptive quantization (heuristic): bitwidth=%d",
                int(module.quant_bit))


def set_quant_bandwidth_heuristic2(module, outputs) -> None:
    """Solve the discrete-packing constraint directly."""
    with monitoring.get_locked_context(MONITORING_KEY_SEND) as mctx:
        tag = mctx.get_tag(key=MONITORING_KEY_SEND)
        window_size = mctx.get_window_size(key=MONITORING_KEY_SEND)
        bandwidth = mctx.get_window_perf(key=MONITORING_KEY_SEND)
    if tag == 0 or tag % window_size != 0:
        return
    tensors = (outputs,) if isinstance(outputs, torch.Tensor) else outputs
    ubatch_size = models.get_microbatch_size(outputs, verify=True)
    rate = module.rate_constraint.item()
    ubatch_time = torch.tensor(ubatch_size / rate if rate > 0
                               else float("inf"))
    ubatch_mbits = sum(t.numel() * t.element_size() for t in tensors)         * 8 / 1e6
    src_bit = torch.tensor(tensors[0].element_size() * 8)
    quant_bit = quantutil.constrain_max_bitwidth(
        ubatch_time, torch.tensor(ubatch_mbits), torch.tensor(bandwidth),
        src_bit)
    module.quant_bit = max(torch.tensor(2), quant_bit) % src_bit
    logger.info("Adaptive quantization (heuristic2): bitwidth=%d",
                int(module.quant_bit))


def set_quant_controller(module, outputs) -> None:
    """EMSOFT'13-style integral controller splitting windows between two
    bitwidths."""
    try:
        bw1 = module.bitwidth1.item()
        bw2 = module.bitwidth2.item()
        bw1_iters = module.bitwidth1_iters.item()
    except AttributeError:
        # first window period, before the controller sets these buffers:
        # hold the CONFIGURED bitwidth (the reference zeroes quant_bit here,
        # reference runtime.py:183-191, disabling quantization for the first
        # window even when -q was given; we keep the configured bit instead)
        bw1 = bw2 = module.quant_bit.item()
        bw1_iters = 0
    with monitoring.get_locked_context(MONITORING_KEY_SEND) as mctx:
        tag = mctx.get_tag(key=MONITORING_KEY_SEND)
        window_size = mctx.get_window_size(key=MONITORING_KEY_SEND)
        heartrate = mctx.get_window_heartrate(key=MONITORING_KEY_SEND)
    if tag > 0 and tag % window_size == 0:
        with _MODULE_QUANT_CONTROLLERS_LOCK:
            if module not in _MODULE_QUANT_CONTROLLERS:
                bw_start = module.quant_bit.item() or max(BITWIDTHS)
                _MODULE_QUANT_CONTROLLERS[module] =                     quantutil.AdaptiveBitwidthPerformanceController(
                        0, BITWIDTHS, bw_start)
        bw_ctlr = _MODULE_QUANT_CONTROLLERS[module]
        bw_ctlr.reference = module.rate_constraint.item()
        ubatch_size = models.get_microbatch_size(outputs, verify=True)
        bw1, bw2, bw1_iters = bw_ctlr(heartrate * ubatch_size, window_size)
        module.register_buffer("bitwidth1", torch.tensor(bw1),
                               persistent=False)
        module.register_buffer("bitwidth2", torch.tensor(bw2),
                               persistent=False)
        logger.info("Adaptive quantization (controller): bitwidth1=%d "
                    "(iters=%d), bitwidth2=%d", bw1, bw1_iters, bw2)
    bitwidth = bw1 if bw1_iters > 0 else bw2
    module.quant_bit = torch.tensor(bitwidth % max(BITWIDTHS))
    module.register_buffer("bitwidth1_iters",
                           torch.tensor(max(0, bw1_iters - 1)),
                           persistent=False)


_ADAPTIVE_POLICIES = {
    ADAPTIVE_QUANT_HEURISTIC: set_quant_bandwidth_heuristic,
    ADAPTIVE_QUANT_HEURISTIC2: set_quant_bandwidth_heuristic2,
    ADAPTIVE_QUANT_CONTROLLER: set_quant_controller,
}


def make_quant_send_transform(module, pad_words_per_tensor: List[int],
                              max_bit: int):
    """Encode stage outputs with the module's current (adaptive) quant_bit.

    The static wire schema is sized for the stage's CONFIGURED bitwidth
    (``max_bit``), so adaptive policies may only tighten compression:
    bitwidths are clamped into [2, max_bit] (the reference could also turn
    quantization fully off mid-run, which would change message sizes -
    runtime.py:121-216; on xGMI the fixed ceiling-sized payload is not a
    bottleneck)."""
    policy = _ADAPTIVE_POLICIES.get(os.getenv(ENV_ADAPTIVE_QUANT, ""))

    def transform(outputs: models.ModuleShardData) -> List[torch.Tensor]:
        if policy is not None:
            try:
                policy(module, outputs)
            except Exception:  # noqa: BLE001
                logger.exception("adaptive quant policy failed")
        monitoring.iteration_start(MONITORING_KEY_QUANT_ENCODE)
        if isinstance(outputs, torch.Tensor):
            outputs = (outputs,)
        quant_bit = int(module.quant_bit.item())
        if quant_bit <= 0 or quant_bit > max_bit:
            quant_bit = max_bit  # schema ceiling; see docstring
        wire: List[torch.Tensor] = []
        for tensor, pad_words in zip(outputs, pad_words_per_tensor):
            if quant_bit > 0:
                clamp = clamp_banner2019_laplace if tensor.min() < 0.2 \
                    else clamp_banner2019_gelu
                tensor = clamp(tensor, quant_bit)
            wire += tensor_encode_outerdim(tensor, quant_bit, pad_words=pad_words)
        n_items = models.get_microbatch_size(outputs[0]) if quant_bit > 0 else 0
        monitoring.iteration(MONITORING_KEY_QUANT_ENCODE, work=n_items,
                             accuracy=quant_bit)
        return wire
    return transform


def make_quant_recv_transform(dtype: torch.dtype):
    """Decode 5-tensor groups back into stage inputs."""
    def transform(wire: List[torch.Tensor]) -> models.ModuleShardData:
        monitoring.iteration_start(MONITORING_KEY_QUANT_DECODE)
        assert len(wire) % 5 == 0 and len(wire) >= 5
        quant_bit = int(wire[4].view(-1)[0].item())
        tensors = []
        for i in range(len(wire) // 5):
            group = wire[i * 5:i * 5 + 5]
            tensors.append(tensor_decode_outerdim(group).to(dtype))
        n_items = models.get_microbatch_size(tensors[0]) if quant_bit > 0 else 0
        monitoring.iteration(MONITORING_KEY_QUANT_DECODE, work=n_items,
                             accuracy=quant_bit)
        outputs = tensors[0] if len(tensors) == 1 else tuple(tensors)
        return outputs
    return transform


# ---------------------------------------------------------------------------
# RPC-path quantization/monitoring forward hooks (parity: reference
# runtime.py:60-119,558-565). The p2p engine attaches the same logic as
# stage send/recv TRANSFORMS (above); the RPC transport owns nn.Modules
# remotely, so there it rides torch's hook mechanism, fanned out via
# DistRpcPipeline.rpc_register_*. Unlike the p2p wire, TensorPipe frames are
# dynamic, so adaptive bitwidths are NOT clamped to a schema ceiling here -
# bit 0 (quantization off) passes through unquantized like the reference.
# ---------------------------------------------------------------------------

def forward_pre_hook_monitor(_module, _inputs) -> None:
    """Mark shard-forward iteration start (reference runtime.py:60-62)."""
    monitoring.iteration_start(MONITORING_KEY_MODEL)


def forward_hook_monitor(module, _inputs, outputs) -> None:
    """Record shard-forward completion (reference runtime.py:64-71)."""
    n_items = models.get_microbatch_size(outputs, verify=True)
    n_layers = module.shard_config.layer_end - module.shard_config.layer_start + 1
    monitoring.iteration(MONITORING_KEY_MODEL, work=n_items, accuracy=n_layers)


def forward_hook_quant_encode(module, _inputs, outputs):
    """Quantize stage outputs into the 5-tensor wire format (reference
    runtime.py:73-91), applying the adaptive policy first when configured."""
    policy = _ADAPTIVE_POLICIES.get(os.getenv(ENV_ADAPTIVE_QUANT, ""))
    if policy is not None:
        try:
            policy(module, outputs)
        except Exception:  # noqa: BLE001
            logger.exception("adaptive quant policy failed")
    monitoring.iteration_start(MONITORING_KEY_QUANT_ENCODE)
    if isinstance(outputs, torch.Tensor):
        outputs = (outputs,)
    quant_bit = int(module.quant_bit.item())
    wire: List[torch.Tensor] = []
    for tensor in outputs:
        if quant_bit > 0:
            clamp = clamp_banner2019_laplace if tensor.min() < 0.2 \
                else clamp_banner2019_gelu
            tensor = clamp(tensor, quant_bit)
        wire += tensor_encode_outerdim(tensor, quant_bit)
    n_items = models.get_microbatch_size(outputs[0]) if quant_bit > 0 else 0
    monitoring.iteration(MONITORING_KEY_QUANT_ENCODE, work=n_items,
                         accuracy=quant_bit)
    return tuple(wire)


def forward_hook_rpc_send_monitor(_module, _inputs, outputs) -> None:
    """Record outbound wire traffic on the RPC path. The p2p engine measures
    this in its comm-thread hooks (p2p_*_hook_monitor below); TensorPipe
    pushes happen inside the remote stage call, so the wire size is sampled
    from the (post-encode) outputs instead. The adaptive policies key off
    this MONITORING_KEY_SEND window state."""
    tensors = (outputs,) if isinstance(outputs, torch.Tensor) else outputs
    n_bytes = sum(t.numel() * t.element_size() for t in tensors)
    monitoring.iteration_start(MONITORING_KEY_SEND)
    monitoring.iteration(MONITORING_KEY_SEND, work=n_bytes * 8 / 1e6)


def forward_pre_hook_rpc_recv_monitor(_module, input_arg) -> None:
    """Record inbound wire traffic on the RPC path (mirror of the send
    monitor; registered before the decode pre-hook so it sees wire sizes)."""
    wire = input_arg[0]
    tensors = (wire,) if isinstance(wire, torch.Tensor) else wire
    n_bytes = sum(t.numel() * t.element_size() for t in tensors)
    monitoring.iteration_start(MONITORING_KEY_RECV)
    monitoring.iteration(MONITORING_KEY_RECV, work=n_bytes * 8 / 1e6)


def forward_pre_hook_quant_decode(module, input_arg):
    """Decode 5-tensor wire groups back into stage inputs (reference
    runtime.py:93-119). ``input_arg`` is the 1-tuple of forward() args."""
    monitoring.iteration_start(MONITORING_KEY_QUANT_DECODE)
    assert isinstance(input_arg, tuple) and len(input_arg) == 1
    wire = input_arg[0]
    assert len(wire) % 5 == 0 and len(wire) >= 5
    quant_bit = int(wire[4].view(-1)[0].item())
    dtype = getattr(module, "compute_dtype", None)
    tensors = []
    for i in range(len(wire) // 5):
        decoded = tensor_decode_outerdim(list(wire[i * 5:i * 5 + 5]))
        if dtype is not None:
            decoded = decoded.to(dtype)
        tensors.append(decoded)
    outputs = tuple(tensors) if len(tensors) == 1 else (tuple(tensors),)
    n_items = models.get_microbatch_size(tensors[0]) if quant_bit > 0 else 0
    monitoring.iteration(MONITORING_KEY_QUANT_DECODE, work=n_items,
                         accuracy=quant_bit)
    return outputs


def cast_recv_transform(dtype: torch.dtype):
    """Un-quantized inbound edge: cast to the compute dtype."""
    def transform(wire: List[torch.Tensor]) -> models.ModuleShardData:
        tensors = [t.to(dtype) if t.is_floating_point() else t for t in wire]
        return tensors[0] if len(tensors) == 1 else tuple(tensors)
    return transform


# p2p comm monitoring hooks (parity: reference runtime.py:219-234)
def p2p_pre_hook_monitor(key: str) -> None:
    monitoring.iteration_start(key)


def p2p_post_hook_monitor(key: str, tensors: List[torch.Tensor]) -> None:
    n_bytes = sum(t.numel() * t.element_size() for t in tensors)
    mbits = n_bytes * 8 / 1e6
    monitoring.iteration(key, work=mbits)


# ---------------------------------------------------------------------------
# results handling (parity: reference runtime.py:236-257)
# ---------------------------------------------------------------------------

results_counter = ThreadSafeCounter()
label_queue = queue.Queue()

# Synthetic datasets carry RANDOM labels, so the per-microbatch accuracy
# compare is noise - and its `.cpu()`/`.item()` forces a device sync per
# microbatch inside the results path, serializing the microbatch streams
# (measured: single-GPU runtime.py throughput roughly halves). Real
# datasets (CoLA/ImageNet/npz) keep the reference's per-microbatch
# accuracy bookkeeping; load_dataset sets this flag.
_compute_accuracy = True


def handle_results(tensors: models.ModuleShardData) -> None:
    """Process pipeline outputs: accuracy bookkeeping + counting."""
    if isinstance(tensors, tuple):
        tensors = tensors[0]
    n_items = models.get_microbatch_size(tensors)
    monitoring.iteration_start(MONITORING_KEY_OUTPUT)
    acc = 0
    if _compute_accuracy and not label_queue.empty() and tensors.dim() == 2:
        labels = label_queue.get()
        pred = tensors.float().argmax(dim=-1).cpu()
        acc = int((pred == labels[:pred.shape[0]]).sum().item())
    monitoring.iteration(MONITORING_KEY_OUTPUT, work=n_items, accuracy=acc)
    results_counter.add(n_items)


# ---------------------------------------------------------------------------
# pipeline drivers
# ---------------------------------------------------------------------------

def _compute_dtype(device: torch.device) -> torch.dtype:
    return torch.bfloat16 if device.type == "cuda" else torch.float32


def load_dataset(dataset_cfg: dict, model_name: str, batch_size: int,
                 ubatch_size: int):
    """Dataset dispatch (reference runtime.py:358-401). Default: synthetic
    random inputs (offline environment)."""
    config = model_cfg.get_model_config(model_name)
    name = dataset_cfg.get("name")
    indices = dataset_cfg.get("indices")
    if name == "CoLA":
        from pipeedge_amd.utils.data import load_dataset_glue
        dataset = load_dataset_glue(model_name, "cola",
                                    dataset_cfg.get("split", "train"),
                                    batch_size, indices=indices)
    elif name == "npz":
        from pipeedge_amd.utils.data import load_dataset_npz
        dataset = load_dataset_npz(dataset_cfg["root"], batch_size)
    elif name == "ImageNet":
        from pipeedge_amd.utils.data import load_dataset_imagenet
        dataset = load_dataset_imagenet(dataset_cfg["root"],
                                        dataset_cfg.get("split", "train"),
                                        batch_size, indices=indices)
    else:
        dataset = load_dataset_synthetic(config, batch_size)
        if indices is not None:
            from pipeedge_amd.utils.data import RolloverTensorDataset
            sel = [dataset[int(i) % len(dataset)] for i in indices]
            dataset = RolloverTensorDataset(
                batch_size, *(torch.stack(ts) for ts in zip(*sel)))
    if dataset_cfg.get("shuffle"):
        from pipeedge_amd.utils.data import ShuffledDataset
        dataset = ShuffledDataset(dataset)
    global _compute_accuracy
    _compute_accuracy = name in ("CoLA", "ImageNet", "npz")
    return dataset


def run_pipeline_p2p(world_size: int, rank: int, model_name: str,
                     model_file: Optional[str], batch_size: int,
                     ubatch_size: int, partition, quant_list, rank_order,
                     data_rank: int, hosts, dataset_cfg: dict,
                     sched_models_file, sched_dev_types_file, sched_dev_file,
                     num_batches: int = 1) -> None:
    """Run the P2P pipeline (parity: reference runtime.py:418-511)."""
    monitoring.init(MONITORING_KEY_MODEL, get_window_size(),
                    work_type="tensors", acc_type="layers")
    monitoring.add_key(MONITORING_KEY_OUTPUT, work_type="classifications",
                       acc_type="correct")
    monitoring.add_key(MONITORING_KEY_QUANT_DECODE, work_type="tensors",
                       acc_type="bits")
    monitoring.add_key(MONITORING_KEY_QUANT_ENCODE, work_type="tensors",
                       acc_type="bits")
    monitoring.add_key(MONITORING_KEY_RECV, work_type="Mbits")
    monitoring.add_key(MONITORING_KEY_SEND, work_type="Mbits")

    device = devices.DEVICE if devices.DEVICE is not None else torch.device("cpu")
    dtype = _compute_dtype(device)
    dtype_str = f"torch.{str(dtype).split('.')[-1]}"

    with DistP2pContext(world_size, rank, handle_cmd) as dist_ctx:
        # distribute the schedule (reference runtime.py:431-448)
        if rank == 0:
            try:
                stage_layers, stage_quant, stage_ranks = get_pipeline_sched(
                    world_size, hosts, partition, quant_list, rank_order,
                    model_name, ubatch_size, dtype_str, sched_models_file,
                    sched_dev_types_file, sched_dev_file)
                # seq_len: data-dependent for text models; broadcast with
                # the schedule
                seq_len = 0
                if model_cfg.get_model_config(model_name).model_type == "bert":
                    dataset_probe = load_dataset(dataset_cfg, model_name, 1, 1)
                    seq_len = dataset_probe[0][0].shape[-1]
            except Exception:
                # a scheduling failure must not strand the other ranks in
                # their schedule wait forever (observed: a bad YAML hung a
                # whole 8-rank job until an external timeout killed it)
                logger.exception("Scheduling failed; broadcasting stop")
                dist_ctx.cmd_broadcast(CMD_STOP)
                stop_event.set()
                raise
            logger.info("Scheduling: data rank: %s", data_rank)
            dist_ctx.cmd_broadcast(CMD_SCHED, (
                torch.tensor(stage_layers),
                torch.tensor(stage_quant),
                torch.tensor(stage_ranks),
                torch.tensor(data_rank),
                torch.tensor(seq_len)))
        else:
            logger.info("Waiting for schedule")
            while True:
                try:
                    sched = sched_q.get(timeout=0.25)
                    break
                except queue.Empty:
                    if stop_event.is_set():
                        raise RuntimeError(
                            "stopped before a schedule arrived (rank 0 "
                            "failed or the run was aborted)") from None
            stage_layers, stage_quant, stage_ranks, data_rank, seq_len = sched
            stage_layers = [tuple(sl) for sl in stage_layers]

        try:
            stage = stage_ranks.index(rank)
        except ValueError:
            stage = None

        module = None
        if stage is not None:
            module = model_cfg.module_shard_factory(
                model_name, model_file, stage_layers[stage][0],
                stage_layers[stage][1], stage, dtype=dtype)
            module.register_buffer("quant_bit",
                                   torch.tensor(stage_quant[stage]),
                                   persistent=False)
            send_constraint = float(os.getenv(ENV_SEND_CONSTRAINT, "0"))
            module.register_buffer("rate_constraint",
                                   torch.tensor(send_constraint),
                                   persistent=False)


        # hipGraph-replayed forward on GPU; shard monitoring wraps the
        # callable (module forward hooks would not fire during graph replay)
        from pipeedge_amd.utils.graph import maybe_graph
        work_cb = None
        if module is not None:
            inner_cb = maybe_graph(module)
            n_layers = stage_layers[stage][1] - stage_layers[stage][0] + 1

            def work_cb(data, _cb=inner_cb, _n_layers=n_layers):
                monitoring.iteration_start(MONITORING_KEY_MODEL)
                outputs = _cb(data)
                n_items = models.get_microbatch_size(outputs, verify=True)
                monitoring.iteration(MONITORING_KEY_MODEL, work=n_items,
                                     accuracy=_n_layers)
                return outputs
        stage_ctx = model_cfg.dist_p2p_pipeline_stage_factory(
            stage_ranks, data_rank, rank, stage, work_cb, handle_results,
            model_name=model_name, ubatch_size=ubatch_size,
            stage_layers=stage_layers, stage_quant=stage_quant, dtype=dtype,
            results_group=dist_ctx.results_group, seq_len=seq_len or None)

        # attach quant / cast transforms to the stage's edges
        config = model_cfg.get_model_config(model_name)
        total_layers = model_cfg.get_model_layers(model_name)
        if stage is not None:
            # outbound: quantize unless we're the last stage
            if stage_layers[stage][1] < total_layers and stage_quant[stage] > 0:
                out_specs = model_spec.stage_boundary_spec(
                    config, stage_layers[stage][1], total_layers, ubatch_size,
                    dtype, seq_len or None)
                pad_words = [_pad_words(s, stage_quant[stage])
                             for s in out_specs]
                stage_ctx.send_transform = make_quant_send_transform(
                    module, pad_words, stage_quant[stage])
            elif stage_layers[stage][1] < total_layers and \
                    os.getenv(ENV_ADAPTIVE_QUANT):
                # static wire schema: a stage configured quant=0 has no quant
                # send transform, so adaptive policies cannot enable
                # quantization for it mid-run (the reference could; its wire
                # re-negotiates per message). Surface it instead of no-op'ing.
                logger.warning(
                    "%s is set but stage %d has quant=0: adaptive "
                    "quantization is DISABLED for this stage (configure a "
                    "nonzero -q ceiling to enable; policy bitwidth 0 is "
                    "coerced to that ceiling)", ENV_ADAPTIVE_QUANT, stage)
            # inbound: dequantize unless we're stage 0
            if stage > 0 and stage_quant[stage - 1] > 0:
                stage_ctx.recv_transform = make_quant_recv_transform(dtype)
            else:
                stage_ctx.recv_transform = cast_recv_transform(dtype)
        stage_ctx.register_recv_pre_hook(p2p_pre_hook_monitor,
                                         (MONITORING_KEY_RECV,))
        stage_ctx.register_recv_post_hook(p2p_post_hook_monitor,
                                          (MONITORING_KEY_RECV,))
        stage_ctx.register_send_pre_hook(p2p_pre_hook_monitor,
                                         (MONITORING_KEY_SEND,))
        stage_ctx.register_send_post_hook(p2p_post_hook_monitor,
                                          (MONITORING_KEY_SEND,))

        if rank == data_rank:
            dataset = load_dataset(dataset_cfg, model_name, batch_size,
                                   ubatch_size)
            ubatches = []
            labels = []
            for start in range(0, len(dataset), ubatch_size):
                items = [dataset[i] for i in
                         range(start, min(start + ubatch_size, len(dataset)))]
                data = torch.stack([it[0] for it in items])
                lbl = torch.stack([torch.as_tensor(it[1]) for it in items])
                if data.is_floating_point():
                    data = data.to(dtype)
                ubatches.append(data)
                labels.append(lbl)
            tik_data = time.time()
            start_count = results_counter.value
            # stream ALL batches continuously (the reference enqueues every
            # microbatch before waiting, runtime.py:498-501; a per-batch
            # engine drain cost ~12% at batch 64 / ubatch 8)
            for _ in range(num_batches):
                for lbl in labels:
                    label_queue.put(lbl)
            dist_ctx.cmd_broadcast(
                CMD_BATCH, (torch.tensor(len(ubatches) * num_batches),))
            stage_ctx.data_run_batch(ubatches * num_batches)
            results_counter.wait_gte(start_count + len(dataset) * num_batches)
            if device.type == "cuda":
                torch.cuda.synchronize()
            tok_data = time.time()
            latency = tok_data - tik_data
            throughput = batch_size * num_batches / latency
            logger.info("Latency is %f, throughput is %f items/sec",
                        latency, throughput)
            dist_ctx.cmd_broadcast(CMD_STOP)
            stop_event.set()
        elif stage is not None:
            while not stop_event.is_set():
                try:
                    n_ubatch = batch_q.get(timeout=0.1)
                except queue.Empty:
                    continue
                stage_ctx.stage_run_batch(n_ubatch)
        else:
            stop_event.wait()
    monitoring.finish()


def _pad_words(tensor_spec, quant_bit: int) -> Optional[int]:
    if quant_bit <= 0:
        return None
    import math
    per_elem = 1
    for d in tensor_spec.shape[1:]:
        per_elem *= d
    return math.ceil(per_elem / (32 // quant_bit))


def run_pipeline_rpc(*args, **kwargs) -> None:
    """RPC transport (reference runtime.py:514-578). See comm/rpc.py."""
    from pipeedge_amd.comm import rpc as rpc_mod
    rpc_mod.run_pipeline_rpc(*args, **kwargs)


def init_env(device: Optional[str], net_addr: str, net_port: int,
             net_ifname: str, rank: int = 0) -> None:
    """Initialize the runtime environment (reference runtime.py:581-602).

    A bare ``cuda`` device (or default) on a multi-GPU host maps this rank
    to ``cuda:<LOCAL_RANK or rank % device_count>`` - one GPU per rank, the
    RCCL requirement (single-node multi-rank; the reference ran one process
    per HOST so never needed this).
    """
    if device in (None, "cuda") and torch.cuda.is_available() \
            and torch.cuda.device_count() > 1:
        local = int(os.environ.get("LOCAL_RANK",
                                   rank % torch.cuda.device_count()))
        device = f"cuda:{local}"
    devices.initialize(device)
    os.environ.setdefault("MASTER_ADDR", net_addr)
    os.environ.setdefault("MASTER_PORT", str(net_port))
    if net_ifname and net_ifname != "lo0":
        os.environ.setdefault("GLOO_SOCKET_IFNAME", net_ifname)
        os.environ.setdefault("TP_SOCKET_IFNAME", net_ifname)


def main() -> None:
    parser = argparse.ArgumentParser(
        description="MI355X Pipeline Parallelism Runtime",
        formatter_class=argparse.ArgumentDefaultsHelpFormatter)
    parser.add_argument("rank", type=int, help="the rank for the current node")
    parser.add_argument("worldsize", type=int, help="the world size")
    parser.add_argument("-d", "--device", type=str, default=None,
                        help="compute device: 'cpu', 'cuda', 'cuda:1', ...")
    parser.add_argument("-s", "--socket-ifname", type=str, default="lo0",
                        help="socket interface name")
    parser.add_argument("--addr", type=str, default="127.0.0.1",
                        help="ip address for the master node")
    parser.add_argument("--port", type=int, default=29500,
                        help="communication port for the master node")
    parser.add_argument("-c", "--comm", type=str, default="p2p",
                        choices=["p2p", "rpc"],
                        help="the communication implementation")
    parser.add_argument("-w", "--worker-threads", default=16, type=int,
                        help="worker threads for the 'rpc' backend")
    parser.add_argument("-m", "--model-name", type=str,
                        default="google/vit-base-patch16-224",
                        choices=model_cfg.get_model_names(),
                        help="the neural network model for loading")
    parser.add_argument("-M", "--model-file", type=str,
                        help="the model weights file")
    parser.add_argument("-b", "--batch-size", default=64, type=int)
    parser.add_argument("-u", "--ubatch-size", default=8, type=int)
    parser.add_argument("--num-batches", default=1, type=int,
                        help="number of batches to stream (throughput runs)")
    dset = parser.add_argument_group("Dataset arguments")
    dset.add_argument("--dataset-name", type=str,
                      choices=["CoLA", "ImageNet", "npz", "synthetic"],
                      help="dataset to use (default: synthetic random)")
    dset.add_argument("--dataset-root", type=str,
                      help="dataset root directory")
    dset.add_argument("--dataset-split", default="train", type=str)
    dset.add_argument("--dataset-indices-file", default=None, type=str)
    dset.add_argument("--dataset-shuffle", type=bool, nargs="?", const=True,
                      default=False)
    usched = parser.add_argument_group("User-defined scheduling")
    usched.add_argument("-pt", "--partition", type=str,
                        help="comma-delimited layer pairs, e.g. '1,24,25,48'")
    usched.add_argument("-q", "--quant", type=str,
                        help="comma-delimited quantization bits per stage")
    usched.add_argument("-r", "--rank-order", type=str, default=None,
                        help="comma-delimited ranks in stage order")
    usched.add_argument("-D", "--data-rank", type=int, default=0,
                        help="rank where inputs are loaded / outputs handled")
    asched = parser.add_argument_group("Automated scheduling")
    asched.add_argument("-H", "--hosts", type=str,
                        help="comma-delimited hosts in rank order")
    asched.add_argument("-sm", "--sched-models-file", default=None, type=str)
    asched.add_argument("-sdt", "--sched-dev-types-file", default=None, type=str)
    asched.add_argument("-sd", "--sched-dev-file", default=None, type=str)
    args = parser.parse_args()

    if args.dataset_indices_file is None:
        indices = None
    elif args.dataset_indices_file.endswith(".pt"):
        indices = torch.load(args.dataset_indices_file)
    else:
        indices = np.load(args.dataset_indices_file)
    dataset_cfg = {
        "name": args.dataset_name,
        "root": args.dataset_root,
        "split": args.dataset_split,
        "indices": indices,
        "shuffle": args.dataset_shuffle,
    }

    partition = None
    if args.partition is not None:
        parts = [int(i) for i in args.partition.split(",")]
        assert len(parts) % 2 == 0
        partition = [(parts[i], parts[i + 1]) for i in range(0, len(parts), 2)]
    quant_list = None if args.quant is None else \
        [int(i) for i in args.quant.split(",")]
    rank_order = None if args.rank_order is None else \
        [int(i) for i in args.rank_order.split(",")]
    hosts = None if args.hosts is None else args.hosts.split(",")

    tik = time.time()
    init_env(args.device, args.addr, args.port, args.socket_ifname,
             rank=args.rank)
    logger.info("Device: %s", devices.DEVICE)
    if args.comm == "p2p":
        run_pipeline_p2p(args.worldsize, args.rank, args.model_name,
                         args.model_file, args.batch_size, args.ubatch_size,
                         partition, quant_list, rank_order, args.data_rank,
                         hosts, dataset_cfg, args.sched_models_file,
                         args.sched_dev_types_file, args.sched_dev_file,
                         num_batches=args.num_batches)
    else:
        run_pipeline_rpc(args.worldsize, args.rank, args.model_name,
                         args.model_file, args.batch_size, args.ubatch_size,
                         partition, quant_list, rank_order, args.data_rank,
                         hosts, dataset_cfg, args.sched_models_file,
                         args.sched_dev_types_file, args.sched_dev_file,
                         args.worker_threads)
    tok = time.time()
    logger.info("Total program execution time = %f", tok - tik)


if __name__ == "__main__":
    logging.basicConfig(filename="runtime.log", level=logging.DEBUG)
    console_hndlr = logging.StreamHandler(sys.stdout)
    console_hndlr.setFormatter(logging.Formatter(fmt="%(message)s"))
    console_hndlr.setLevel(logging.INFO)
    logging.getLogger().addHandler(console_hndlr)
    main()
