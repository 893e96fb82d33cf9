"""Model configurations and factory glue (parity: reference model_cfg.py).

Maps the supported model zoo to layer counts / default weight files / shard
classes (reference model_cfg.py:24-43) and wires shards to the p2p pipeline
engine (reference model_cfg.py:128-166), extended with the static wire-schema
computation this build uses instead of per-message headers.
"""
import logging
from typing import Any, Callable, List, Optional, Tuple

import torch

import devices
from pipeedge_amd.comm import p2p
from pipeedge_amd.models import ModuleShard, ModuleShardConfig
from pipeedge_amd.models import bert, deit, spec, vit
from pipeedge_amd.models.config import get_config

_logger = logging.getLogger(__name__)

_MODEL_CONFIGS = {}


def _model_cfg_add(name, layers, weights_file, shard_module):
    _MODEL_CONFIGS[name] = {
        "name": name,
        "layers": layers,
        "weights_file": weights_file,
        "shard_module": shard_module,
    }


# Transformer blocks split 4 ways: ViT-Base 12 blocks -> 48 schedulable layers
_model_cfg_add("google/vit-base-patch16-224", 48, "ViT-B_16-224.npz",
               vit.ViTShardForImageClassification)
_model_cfg_add("google/vit-large-patch16-224", 96, "ViT-L_16-224.npz",
               vit.ViTShardForImageClassification)
_model_cfg_add("google/vit-huge-patch14-224-in21k", 128, "ViT-H_14.npz",
               vit.ViTShardForImageClassification)
_model_cfg_add("bert-base-uncased", 48, "BERT-B.npz", bert.BertModelShard)
_model_cfg_add("bert-large-uncased", 96, "BERT-L.npz", bert.BertModelShard)
_model_cfg_add("textattack/bert-base-uncased-CoLA", 48, "BERT-B-CoLA.npz",
               bert.BertShardForSequenceClassification)
_model_cfg_add("facebook/deit-base-distilled-patch16-224", 48,
               "DeiT_B_distilled.npz", deit.DeiTShardForImageClassification)
_model_cfg_add("facebook/deit-small-distilled-patch16-224", 48,
               "DeiT_S_distilled.npz", deit.DeiTShardForImageClassification)
_model_cfg_add("facebook/deit-tiny-distilled-patch16-224", 48,
               "DeiT_T_distilled.npz", deit.DeiTShardForImageClassification)


def get_model_names() -> List[str]:
    return list(_MODEL_CONFIGS.keys())


def get_model_dict(model_name: str) -> dict:
    return _MODEL_CONFIGS[model_name]


def get_model_layers(model_name: str) -> int:
    return _MODEL_CONFIGS[model_name]["layers"]


def get_model_config(model_name: str) -> Any:
    """Architecture config from the offline registry (no AutoConfig/network)."""
    return get_config(model_name)


def get_model_default_weights_file(model_name: str) -> str:
    return _MODEL_CONFIGS[model_name]["weights_file"]


def save_model_weights_file(model_name: str, model_file: Optional[str] = None) -> None:
    if model_file is None:
        model_file = get_model_default_weights_file(model_name)
    module = _MODEL_CONFIGS[model_name]["shard_module"]
    module.save_weights(model_name, model_file)


def module_shard_factory(model_name: str, model_file: Optional[str],
                         layer_start: int, layer_end: int, stage: int,
                         dtype: Optional[torch.dtype] = None) -> ModuleShard:
    """Build a shard on the globally-configured ``devices.DEVICE``.

    Unlike the reference (model_cfg.py:80-95), a missing weights file
    random-initializes deterministically (offline synthetic benchmarking).
    """
    import os
    if model_file is None:
        model_file = get_model_default_weights_file(model_name)
    if not os.path.isfile(model_file):
        _logger.warning("weights file %s not found: using random init", model_file)
        model_file = None
    config = get_model_config(model_name)
    is_first = layer_start == 1
    is_last = layer_end == get_model_layers(model_name)
    shard_config = ModuleShardConfig(layer_start=layer_start, layer_end=layer_end,
                                     is_first=is_first, is_last=is_last)
    module = _MODEL_CONFIGS[model_name]["shard_module"]
    shard = module(config, shard_config, model_file)
    _logger.info("======= %s Stage %d layers [%d, %d] =======",
                 module.__name__, stage, layer_start, layer_end)
    device = devices.DEVICE if devices.DEVICE is not None else torch.device("cpu")
    if dtype is None:
        dtype = torch.bfloat16 if device.type == "cuda" else torch.float32
    shard = shard.to(device=device, dtype=dtype)
    shard.compute_dtype = dtype
    return shard


def _dist_rpc_pipeline_stage_factory(*args, **kwargs):
    """Build a DistRpcPipelineStage on the globally-configured device
    (reference model_cfg.py:98-103); runs on the stage's owner rank."""
    from pipeedge_amd.comm import rpc as rpc_mod
    stage = rpc_mod.DistRpcPipelineStage(*args, **kwargs)
    return stage


def dist_rpc_pipeline_factory(model_name: str, model_file: Optional[str],
                              stage_ranks: List[int],
                              stage_layers: List[Tuple[int, int]],
                              results_to: int,
                              results_cb: Callable[[Any], None]):
    """Build an RPC pipeline with remotely-constructed stages
    (reference model_cfg.py:103-126)."""
    from torch.distributed import rpc as trpc
    from pipeedge_amd.comm import rpc as rpc_mod
    stage_rrefs = []
    assert len(stage_ranks) == len(stage_layers) > 0
    for i, (dst_rank, layers) in enumerate(zip(stage_ranks, stage_layers)):
        rref = trpc.remote(
            dst_rank, _dist_rpc_pipeline_stage_factory,
            args=(module_shard_factory,
                  (model_name, model_file, layers[0], layers[1], i)),
            kwargs={"stage_id": i})
        stage_rrefs.append(rref)
    return rpc_mod.DistRpcPipeline(stage_rrefs, results_to, results_cb)


def stage_wire_schemas(model_name: str, stage_layers: List[Tuple[int, int]],
                       ubatch_size: int, stage_quant: List[int],
                       dtype: torch.dtype, seq_len: Optional[int] = None):
    """Compute every pipeline edge's wire schema.

    Edges: data->stage0 (model input), stage i -> stage i+1 (boundary,
    quantized with stage_quant[i] - reference quant hooks encode on every
    stage but the last, runtime.py:468-482), last stage -> data (raw output).

    Returns a list of ``len(stage_layers) + 1`` schemas: edge e feeds stage e.
    """
    config = get_model_config(model_name)
    total = get_model_layers(model_name)
    edges = [spec.input_spec(config, ubatch_size, dtype, seq_len)]
    for i, (_, layer_end) in enumerate(stage_layers):
        base = spec.stage_boundary_spec(config, layer_end, total, ubatch_size,
                                        dtype, seq_len)
        if layer_end < total:
            qbit = stage_quant[i] if stage_quant else 0
            base = spec.quant_wire_spec(base, qbit)
        edges.append(base)
    return edges


def dist_p2p_pipeline_stage_factory(stage_ranks: List[int], data_rank: int,
                                    rank: int, stage: Optional[int],
                                    module: Optional[ModuleShard],
                                    handle_results_cb: Callable[[Any], None],
                                    model_name: str = None,
                                    ubatch_size: int = 8,
                                    stage_layers: List[Tuple[int, int]] = None,
                                    stage_quant: List[int] = None,
                                    dtype: torch.dtype = torch.float32,
                                    results_group=None,
                                    seq_len: Optional[int] = None) \
        -> p2p.DistP2pPipelineStage:
    """Wire this rank's role in the p2p pipeline (reference model_cfg.py:128-166)
    and attach the static recv/send schemas for its edges."""
    edges = stage_wire_schemas(model_name, stage_layers, ubatch_size,
                               stage_quant or [0] * len(stage_layers), dtype,
                               seq_len)
    num_stages = len(stage_ranks)
    recv_group = send_group = None
    if rank == data_rank:
        if stage is None:
            rank_src = stage_ranks[-1]
            rank_dst = stage_ranks[0]
            work_cb = None
        else:
            if stage != 0:
                raise ValueError(f"Data rank must be stage=0 or stage=None, got {stage}")
            rank_src = stage_ranks[-1] if num_stages > 1 else None
            rank_dst = stage_ranks[1] if num_stages > 1 else None
            work_cb = module
        assert handle_results_cb is not None
        results_cb = handle_results_cb
        recv_specs = edges[-1]          # results edge
        send_specs = edges[1] if work_cb is not None else edges[0]
        recv_group = results_group      # back-edge: its own communicator
    elif stage is None:
        return p2p.DistP2pPipelineStage(None, None, [], [], None, None)
    else:
        rank_src = data_rank if stage == 0 else stage_ranks[stage - 1]
        rank_dst = data_rank if stage == num_stages - 1 else stage_ranks[stage + 1]
        work_cb = module
        results_cb = None
        recv_specs = edges[stage]
        send_specs = edges[stage + 1]
        if stage == num_stages - 1 and rank_dst == data_rank:
            send_group = results_group  # back-edge sender side
    ctx = p2p.DistP2pPipelineStage(rank_src, rank_dst, recv_specs, send_specs,
                                   work_cb, results_cb,
                                   device=devices.DEVICE,
                                   recv_group=recv_group,
                                   send_group=send_group)
    # admission-control depth for the data rank (see DistP2pPipelineStage
    # data_run_batch: the results back-edge makes unbounded admission a
    # distributed-deadlock risk)
    ctx.pipeline_depth = num_stages
    return ctx
