"""DeiT (distilled) model shards.

Same 4-sublayer split and forward semantics as ViT (reference deit.py:27-69 is
a copy of vit.py with different checkpoint keys); only the embeddings (extra
distillation token) and the ``.npz`` key layout (timm state_dict, fused-QKV
``[3D, D]`` weights split at load - reference deit.py:121-157) differ.

Note: the reference loads only ``cls_token`` and leaves HF's distillation
token at its default init (deit.py:121-124); we additionally load
``dist_token`` when the checkpoint provides it.
"""
import logging
from collections.abc import Mapping
from typing import Union

import numpy as np
import torch
from torch import Tensor, nn

from .. import ops
from . import ModuleShard, ModuleShardConfig
from .config import TransformerConfig
from .vit import (ViTLayerShard, ViTModelShard,
                  ViTShardForImageClassification, _param, _t, init_param)

logger = logging.getLogger(__name__)


class DeiTEmbeddings(nn.Module):
    """Patch projection + cls token + distillation token + position embeddings."""

    def __init__(self, config: TransformerConfig):
        super().__init__()
        self.config = config
        d = config.hidden_size
        pdim = config.num_channels * config.patch_size ** 2
        self.patch_weight = _param((d, pdim))
        self.patch_bias = _param((d,))
        self.cls_token = _param((1, 1, d))
        self.dist_token = _param((1, 1, d))
        self.pos_embed = _param((1, config.seq_len, d))  # seq = patches + 2
        self._prefix = None  # cached [2, d] cls+dist rows (built on first use)

    def init_random(self, prefix: str):
        for name, p in self.named_parameters():
            kind = "zeros" if name.endswith("_bias") else "trunc"
            init_param(p, f"{prefix}/emb/{name}", kind)

    def forward(self, pixels: Tensor) -> Tensor:
        patches = ops.patchify(pixels, self.config.patch_size)
        x = ops.linear(patches, self.patch_weight, self.patch_bias)
        prefix = self._prefix
        if prefix is None or prefix.dtype != x.dtype \
                or prefix.device != x.device:
            prefix = torch.cat((self.cls_token.reshape(1, -1),
                                self.dist_token.reshape(1, -1))).to(x.dtype)
            self._prefix = prefix
        return ops.embed_finish(x, prefix, self.pos_embed[0].to(x.dtype))


class DeiTLayerShard(ViTLayerShard):
    """Identical compute to ViTLayerShard (reference deit.py:27-69)."""


class DeiTModelShard(ViTModelShard):
    """ViT encoder body with DeiT embeddings and timm npz key layout."""

    embeddings_cls = DeiTEmbeddings

    @torch.no_grad()
    def load_npz(self, weights: Mapping):
        d = self.config.hidden_size
        if self.embeddings is not None:
            emb = self.embeddings
            emb.cls_token.copy_(_t(weights["cls_token"]))
            if "dist_token" in weights:
                emb.dist_token.copy_(_t(weights["dist_token"]))
            else:
                emb.dist_token.zero_()
            emb.pos_embed.copy_(_t(weights["pos_embed"]))
            # timm conv weight is [D, C, P, P] == [N, (c,ph,pw)]: direct
            kern = weights["patch_embed.proj.weight"]
            emb.patch_weight.copy_(_t(kern).reshape(emb.patch_weight.shape))
            emb.patch_bias.copy_(_t(weights["patch_embed.proj.bias"]))
        for layer in self.layers:
            root = f"blocks.{layer.block_id}."
            if layer.has_layer(0):
                layer.ln1_weight.copy_(_t(weights[root + "norm1.weight"]))
                layer.ln1_bias.copy_(_t(weights[root + "norm1.bias"]))
                # timm fused qkv weight is [3D, D] == our [N, K]: direct
                layer.qkv_weight.copy_(_t(weights[root + "attn.qkv.weight"]))
                layer.qkv_bias.copy_(_t(weights[root + "attn.qkv.bias"]))
            if layer.has_layer(1):
                layer.out_weight.copy_(_t(weights[root + "attn.proj.weight"]))
                layer.out_bias.copy_(_t(weights[root + "attn.proj.bias"]))
            if layer.has_layer(2):
                layer.ln2_weight.copy_(_t(weights[root + "norm2.weight"]))
                layer.ln2_bias.copy_(_t(weights[root + "norm2.bias"]))
                layer.fc1_weight.copy_(_t(weights[root + "mlp.fc1.weight"]))
                layer.fc1_bias.copy_(_t(weights[root + "mlp.fc1.bias"]))
            if layer.has_layer(3):
                layer.fc2_weight.copy_(_t(weights[root + "mlp.fc2.weight"]))
                layer.fc2_bias.copy_(_t(weights[root + "mlp.fc2.bias"]))
        if self.shard_config.is_last:
            self.layernorm_weight.copy_(_t(weights["norm.weight"]))
            self.layernorm_bias.copy_(_t(weights["norm.bias"]))


class DeiTShardForImageClassification(ViTShardForImageClassification):
    """DeiT shard + classifier head (classifies on the CLS token, matching the
    reference's ``data[:, 0, :]`` - deit.py:220-226 via vit.py pattern)."""

    model_shard_cls = DeiTModelShard

    def __init__(self, config: TransformerConfig, shard_config: ModuleShardConfig,
                 model_weights: Union[str, Mapping, None]):
        if isinstance(model_weights, str):
            model_weights = dict(np.load(model_weights))
        ModuleShard.__init__(self, config, shard_config)
        self.classifier_weight = None
        self.vit = DeiTModelShard(config, shard_config, model_weights)
        if shard_config.is_last and config.num_labels > 0:
            self.classifier_weight = _param((config.num_labels, config.hidden_size))
            self.classifier_bias = _param((config.num_labels,))
            if model_weights is None:
                init_param(self.classifier_weight, f"{config.name}/head/weight")
                init_param(self.classifier_bias, f"{config.name}/head/bias", "zeros")
            else:
                with torch.no_grad():
                    self.classifier_weight.copy_(_t(model_weights["head.weight"]))
                    self.classifier_bias.copy_(_t(model_weights["head.bias"]))

    @staticmethod
    def save_weights(model_name: str, model_file: str, seed: int = 0) -> None:
        """Synthesize a random-init checkpoint in the timm npz key layout
        (no network in this environment; reference downloads via torch.hub)."""
        from .config import get_config
        cfg = get_config(model_name)
        rng = np.random.default_rng(seed)
        d, i = cfg.hidden_size, cfg.intermediate_size
        p, c = cfg.patch_size, cfg.num_channels

        def trunc(*shape):
            return np.clip(rng.standard_normal(shape) * 0.02, -0.04, 0.04).astype(np.float32)

        def zeros(*shape):
            return np.zeros(shape, dtype=np.float32)

        def ones(*shape):
            return np.ones(shape, dtype=np.float32)

        out = {
            "cls_token": trunc(1, 1, d),
            "dist_token": trunc(1, 1, d),
            "pos_embed": trunc(1, cfg.seq_len, d),
            "patch_embed.proj.weight": trunc(d, c, p, p),
            "patch_embed.proj.bias": zeros(d),
            "norm.weight": ones(d),
            "norm.bias": zeros(d),
            "head.weight": trunc(cfg.num_labels, d),
            "head.bias": zeros(cfg.num_labels),
        }
        for blk in range(cfg.num_hidden_layers):
            root = f"blocks.{blk}."
            out[root + "norm1.weight"] = ones(d)
            out[root + "norm1.bias"] = zeros(d)
            out[root + "norm2.weight"] = ones(d)
            out[root + "norm2.bias"] = zeros(d)
            out[root + "attn.qkv.weight"] = trunc(3 * d, d)
            out[root + "attn.qkv.bias"] = zeros(3 * d)
            out[root + "attn.proj.weight"] = trunc(d, d)
            out[root + "attn.proj.bias"] = zeros(d)
            out[root + "mlp.fc1.weight"] = trunc(i, d)
            out[root + "mlp.fc1.bias"] = zeros(i)
            out[root + "mlp.fc2.weight"] = trunc(d, i)
            out[root + "mlp.fc2.bias"] = zeros(d)
        np.savez(model_file, **out)
