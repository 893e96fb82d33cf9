"""Model shard abstractions.

Capability parity with the reference's ``src/pipeedge/models/__init__.py``
(ModuleShardConfig at :9, ModuleShard at :25, get_microbatch_size at :39),
re-designed for one-process-per-GPU execution: a shard owns a contiguous range
of *schedulable layers* ``[layer_start, layer_end]`` (1-indexed; each
transformer block contributes 4 schedulable sub-layers) and runs them on a
single device with GPU-resident activations.
"""
import dataclasses
from typing import Tuple, Union

import torch
from torch import Tensor, nn

#: Activations crossing a shard boundary: one tensor, or a tuple when a block
#: is split mid-attention / mid-MLP (e.g. (attn_context, residual)).
ModuleShardData = Union[Tensor, Tuple[Tensor, ...]]


@dataclasses.dataclass
class ModuleShardConfig:
    """Static configuration for a shard: its layer range and boundary flags."""
    layer_start: int = 0
    layer_end: int = 0
    is_first: bool = False
    is_last: bool = False
    # extra framework knobs (MI355X build additions)
    dtype: torch.dtype = torch.float32


class ModuleShard(nn.Module):
    """A partition of a model: layers ``[layer_start, layer_end]`` inclusive.

    ``is_first`` shards own input embeddings; ``is_last`` shards own the final
    norm / pooler / classifier head.
    """

    def __init__(self, config, shard_config: ModuleShardConfig):
        super().__init__()
        self.config = config
        self.shard_config = shard_config

    def has_layer(self, layer: int) -> bool:
        """Whether the given 1-indexed schedulable layer belongs to this shard."""
        return self.shard_config.layer_start <= layer <= self.shard_config.layer_end

    @property
    def layer_range(self) -> Tuple[int, int]:
        return (self.shard_config.layer_start, self.shard_config.layer_end)


def get_microbatch_size(shard_data: ModuleShardData, verify: bool = False) -> int:
    """Microbatch size of shard input/output data (outer dim of the tensors)."""
    if isinstance(shard_data, Tensor):
        shard_data = (shard_data,)
    if verify:
        sizes = {t.shape[0] for t in shard_data}
        assert len(sizes) <= 1, f"inconsistent microbatch sizes: {sizes}"
    return int(shard_data[0].shape[0]) if len(shard_data) > 0 else 0
