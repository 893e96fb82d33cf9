"""Static wire-schema computation.

The reference sends dtype/shape headers with every tensor message
(reference comm/p2p/__init__.py:96-121). In this build, shapes crossing every
stage boundary are fully determined by (model config, partition, microbatch
size, quant bits), so each rank derives its recv/send schemas locally when the
schedule is distributed and the data plane carries raw payloads only.

Boundary shapes follow the 4-sublayer split:
after sublayer 0 of a block: (context[u,S,D], residual[u,S,D]);
after sublayer 1 or 3:        hidden[u,S,D];
after sublayer 2:             (mlp_hidden[u,S,I], residual[u,S,D]).
"""
import math
from typing import List, Optional

import torch

from ..comm import TensorSpec
from .config import TransformerConfig


def input_spec(config: TransformerConfig, ubatch_size: int,
               dtype: torch.dtype,
               seq_len: Optional[int] = None) -> List[TensorSpec]:
    """Schema of the model input (what the data rank sends to stage 0)."""
    if config.model_type in ("vit", "deit"):
        return [TensorSpec((ubatch_size, config.num_channels, config.image_size,
                            config.image_size), dtype)]
    s = seq_len if seq_len else config.max_position_embeddings
    return [TensorSpec((ubatch_size, s), torch.int64)]


def boundary_spec(config: TransformerConfig, layer_end: int, ubatch_size: int,
                  dtype: torch.dtype,
                  seq_len: Optional[int] = None) -> List[TensorSpec]:
    """Schema of activations leaving a stage whose last layer is ``layer_end``
    (1-indexed global schedulable layer)."""
    s = seq_len if seq_len is not None else _default_seq(config)
    d = config.hidden_size
    i = config.intermediate_size
    sub = (layer_end - 1) % 4
    if sub == 0:
        specs = [TensorSpec((ubatch_size, s, d), dtype),
                 TensorSpec((ubatch_size, s, d), dtype)]
    elif sub == 2:
        specs = [TensorSpec((ubatch_size, s, i), dtype),
                 TensorSpec((ubatch_size, s, d), dtype)]
    else:
        specs = [TensorSpec((ubatch_size, s, d), dtype)]
    if config.model_type == "bert":
        # padding mask rides every BERT boundary as the LAST tensor
        # (additive [u, S]; derived from token ids by the first shard)
        specs.append(TensorSpec((ubatch_size, s), dtype))
    return specs


def output_spec(config: TransformerConfig, ubatch_size: int,
                dtype: torch.dtype) -> List[TensorSpec]:
    """Schema of the final model output (last stage -> data rank)."""
    if config.model_type in ("vit", "deit"):
        return [TensorSpec((ubatch_size, config.num_labels), dtype)]
    if config.num_labels > 0:
        return [TensorSpec((ubatch_size, config.num_labels), dtype)]
    return [TensorSpec((ubatch_size, config.hidden_size), dtype)]  # pooled


def _default_seq(config: TransformerConfig) -> int:
    if config.model_type in ("vit", "deit"):
        return config.seq_len
    return config.max_position_embeddings


def stage_boundary_spec(config: TransformerConfig, layer_end: int, total_layers: int,
                        ubatch_size: int, dtype: torch.dtype,
                        seq_len: Optional[int] = None) -> List[TensorSpec]:
    """Schema leaving the stage ending at ``layer_end`` - the final-stage
    boundary is the model output."""
    if layer_end >= total_layers:
        return output_spec(config, ubatch_size, dtype)
    return boundary_spec(config, layer_end, ubatch_size, dtype, seq_len)


def quant_wire_spec(specs: List[TensorSpec], quant_bit: int) -> List[TensorSpec]:
    """Wire schema after per-microbatch-element quantization encoding.

    Mirrors the reference's 5-tensor wire format per source tensor
    (reference quantization/basic_op.py:143,166-170):
    [comm(uint8), shape(int32), scale(f32), shift(f32), bit(int8)], each
    stacked over the microbatch dim. The comm payload is sized for
    ``quant_bit`` (the configured ceiling when bitwidth adapts at runtime).
    """
    if quant_bit <= 0:
        return specs
    out: List[TensorSpec] = []
    for spec in specs:
        u = spec.shape[0]
        per_elem = 1
        for dim in spec.shape[1:]:
            per_elem *= dim
        # floor(32/bit) values per 32-bit word (reference basic_op.py:44)
        words = math.ceil(per_elem / (32 // quant_bit))
        out.append(TensorSpec((u, words * 4), torch.uint8))
        out.append(TensorSpec((u, len(spec.shape) - 1), torch.int32))
        out.append(TensorSpec((u,), torch.float32))
        out.append(TensorSpec((u,), torch.float32))
        out.append(TensorSpec((u,), torch.int8))
    return out
