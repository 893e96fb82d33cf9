"""BERT model shards.

Capability parity with reference ``src/pipeedge/models/transformers/bert.py``:
4 schedulable sublayers per block with BERT's post-LN ordering
(reference bert.py:41-52), first shard owns embeddings (:76-80), last owns the
pooler (:98-102); ``BertShardForSequenceClassification`` strips the ``bert.``
key prefix of fine-tuned checkpoints (:191-196). HF state_dict npz key layout
(``[N, K]`` weights == our native MFMA GEMM layout: loaded directly).

Sublayer semantics (post-LN, reference bert.py:41-52):
  0: self-attention                      -> (context, hidden_in)
  1: attn out dense + residual + LN      -> hidden
  2: intermediate dense + GeLU           -> (mlp_hidden, attn_out)
  3: output dense + residual + LN        -> hidden
"""
import logging
import math
from collections.abc import Mapping
from typing import Union

import numpy as np
import torch
from torch import Tensor, nn

from .. import ops
from . import ModuleShard, ModuleShardConfig
from .config import TransformerConfig
from .vit import (_param, _t, init_param, iter_block_ranges, load_npz_or_random)

logger = logging.getLogger(__name__)


class BertEmbeddings(nn.Module):
    """Word + position + token-type embeddings -> LayerNorm."""

    def __init__(self, config: TransformerConfig):
        super().__init__()
        self.config = config
        d = config.hidden_size
        self.word_embeddings = _param((config.vocab_size, d))
        self.position_embeddings = _param((config.max_position_embeddings, d))
        self.token_type_embeddings = _param((config.type_vocab_size, d))
        self.ln_weight = _param((d,))
        self.ln_bias = _param((d,))

    def init_random(self, prefix: str):
        for name, p in self.named_parameters():
            if name == "ln_weight":
                kind = "ones"
            elif name == "ln_bias":
                kind = "zeros"
            else:
                kind = "trunc"
            init_param(p, f"{prefix}/emb/{name}", kind)

    def forward(self, input_ids: Tensor) -> Tensor:
        seq = input_ids.shape[1]
        x = self.word_embeddings[input_ids]
        x = x + self.position_embeddings[:seq].unsqueeze(0).to(x.dtype)
        x = x + self.token_type_embeddings[0].to(x.dtype)  # all-zero token types
        return ops.layernorm(x, self.ln_weight, self.ln_bias,
                             self.config.layer_norm_eps)


class BertLayerShard(ModuleShard):
    """Sublayers [layer_start, layer_end] (subset of 0..3) of ONE block."""

    def __init__(self, config: TransformerConfig, shard_config: ModuleShardConfig):
        super().__init__(config, shard_config)
        d = config.hidden_size
        i = config.intermediate_size
        self.scale = 1.0 / math.sqrt(config.head_dim)
        if self.has_layer(0):
            self.qkv_weight = _param((3 * d, d))
            self.qkv_bias = _param((3 * d,))
        if self.has_layer(1):
            self.out_weight = _param((d, d))
            self.out_bias = _param((d,))
            self.out_ln_weight = _param((d,))
            self.out_ln_bias = _param((d,))
        if self.has_layer(2):
            self.fc1_weight = _param((i, d))
            self.fc1_bias = _param((i,))
        if self.has_layer(3):
            self.fc2_weight = _param((d, i))
            self.fc2_bias = _param((d,))
            self.fc2_ln_weight = _param((d,))
            self.fc2_ln_bias = _param((d,))

    def init_random(self, prefix: str):
        for name, p in self.named_parameters():
            if name.endswith("ln_weight"):
                kind = "ones"
            elif name.endswith("_bias"):
                kind = "zeros"
            else:
                kind = "trunc"
            init_param(p, f"{prefix}/block{self.block_id}/{name}", kind)

    @torch.no_grad()
    def forward(self, data, attn_mask=None):
        cfg = self.config
        eps = cfg.layer_norm_eps
        if self.has_layer(0):
            qkv = ops.linear(data, self.qkv_weight, self.qkv_bias)
            ctx = ops.qkv_attention(qkv, cfg.num_attention_heads, self.scale,
                                    attn_mask=attn_mask)
            data = (ctx, data)
        if self.has_layer(1):
            y = ops.linear(data[0], self.out_weight, self.out_bias)
            data = ops.add_layernorm(y, data[1], self.out_ln_weight,
                                     self.out_ln_bias, eps)
        if self.has_layer(2):
            mid = ops.linear(data, self.fc1_weight, self.fc1_bias, act="gelu")
            data = (mid, data)
        if self.has_layer(3):
            y = ops.linear(data[0], self.fc2_weight, self.fc2_bias)
            data = ops.add_layernorm(y, data[1], self.fc2_ln_weight,
                                     self.fc2_ln_bias, eps)
        return data


class BertModelShard(ModuleShard):
    """Layers [layer_start, layer_end] of the BERT encoder."""

    def __init__(self, config: TransformerConfig, shard_config: ModuleShardConfig,
                 model_weights: Union[str, Mapping, None]):
        super().__init__(config, shard_config)
        self.embeddings = None
        self.layers = nn.ModuleList()
        self._build()
        load_npz_or_random(self, model_weights)

    def _build(self):
        if self.shard_config.is_first:
            self.embeddings = BertEmbeddings(self.config)
        for block_id, sub_start, sub_end in iter_block_ranges(
                self.shard_config.layer_start, self.shard_config.layer_end):
            cfg = ModuleShardConfig(layer_start=sub_start, layer_end=sub_end)
            layer = BertLayerShard(self.config, cfg)
            layer.block_id = block_id
            self.layers.append(layer)
        if self.shard_config.is_last:
            d = self.config.hidden_size
            self.pooler_weight = _param((d, d))
            self.pooler_bias = _param((d,))

    def init_random(self, prefix: str):
        if self.embeddings is not None:
            self.embeddings.init_random(prefix)
        for layer in self.layers:
            layer.init_random(prefix)
        if self.shard_config.is_last:
            init_param(self.pooler_weight, f"{prefix}/pooler/weight")
            init_param(self.pooler_bias, f"{prefix}/pooler/bias", "zeros")

    @torch.no_grad()
    def load_npz(self, weights: Mapping):
        if self.embeddings is not None:
            emb = self.embeddings
            emb.word_embeddings.copy_(_t(weights["embeddings.word_embeddings.weight"]))
            emb.position_embeddings.copy_(_t(weights["embeddings.position_embeddings.weight"]))
            emb.token_type_embeddings.copy_(_t(weights["embeddings.token_type_embeddings.weight"]))
            emb.ln_weight.copy_(_t(weights["embeddings.LayerNorm.weight"]))
            emb.ln_bias.copy_(_t(weights["embeddings.LayerNorm.bias"]))
        for layer in self.layers:
            root = f"encoder.layer.{layer.block_id}."
            if layer.has_layer(0):
                # HF [N, K] weights: direct
                qkv_w = [_t(weights[root + f"attention.self.{n}.weight"])
                         for n in ("query", "key", "value")]
                layer.qkv_weight.copy_(torch.cat(qkv_w, dim=0))
                qkv_b = [_t(weights[root + f"attention.self.{n}.bias"])
                         for n in ("query", "key", "value")]
                layer.qkv_bias.copy_(torch.cat(qkv_b))
            if layer.has_layer(1):
                layer.out_weight.copy_(_t(weights[root + "attention.output.dense.weight"]))
                layer.out_bias.copy_(_t(weights[root + "attention.output.dense.bias"]))
                layer.out_ln_weight.copy_(_t(weights[root + "attention.output.LayerNorm.weight"]))
                layer.out_ln_bias.copy_(_t(weights[root + "attention.output.LayerNorm.bias"]))
            if layer.has_layer(2):
                layer.fc1_weight.copy_(_t(weights[root + "intermediate.dense.weight"]))
                layer.fc1_bias.copy_(_t(weights[root + "intermediate.dense.bias"]))
            if layer.has_layer(3):
                layer.fc2_weight.copy_(_t(weights[root + "output.dense.weight"]))
                layer.fc2_bias.copy_(_t(weights[root + "output.dense.bias"]))
                layer.fc2_ln_weight.copy_(_t(weights[root + "output.LayerNorm.weight"]))
                layer.fc2_ln_bias.copy_(_t(weights[root + "output.LayerNorm.bias"]))
        if self.shard_config.is_last:
            self.pooler_weight.copy_(_t(weights["pooler.dense.weight"]))
            self.pooler_bias.copy_(_t(weights["pooler.dense.bias"]))

    @torch.no_grad()
    def forward(self, data):
        """Forward with padding-mask plumbing.

        The first shard derives the additive attention mask from the token
        ids (pad_token_id positions -> -1e9; the reference's shards run
        MASK-FREE over padding, reference bert.py:41-52 - this build fixes
        that) and every stage boundary carries the mask as its LAST payload
        tensor (spec.boundary_spec appends it) so downstream attention
        sublayers mask correctly; the last shard drops it.
        """
        attn_mask = None
        if self.shard_config.is_first:
            ids = data
            pad = self.config.pad_token_id
            dt = self.embeddings.word_embeddings.dtype
            attn_mask = torch.where(
                ids == pad,
                torch.full((), -1e9, dtype=dt, device=ids.device),
                torch.zeros((), dtype=dt, device=ids.device))
            data = self.embeddings(ids)
        elif isinstance(data, tuple):
            *inner, attn_mask = data
            data = inner[0] if len(inner) == 1 else tuple(inner)
        kernel_mask = attn_mask
        if attn_mask is not None and attn_mask.is_cuda:
            # pre-pad ONCE per shard forward to the HIP kernel's f32
            # [B, ceil128(S)] contract; every block reuses it (the per-call
            # pad inside the op cost 3 extra launches per attention)
            b, s = attn_mask.shape
            sp = (s + 127) & ~127
            km = torch.zeros(b, sp, dtype=torch.float32,
                             device=attn_mask.device)
            km[:, :s] = attn_mask.float()
            kernel_mask = km
        for layer in self.layers:
            data = layer(data, kernel_mask)
        if self.shard_config.is_last:
            # pooler: tanh(dense(CLS token))
            return ops.linear(data[:, 0], self.pooler_weight,
                              self.pooler_bias, act="tanh")
        if isinstance(data, torch.Tensor):
            return (data, attn_mask)
        return (*data, attn_mask)

    @staticmethod
    def save_weights(model_name: str, model_file: str, seed: int = 0) -> None:
        save_random_bert_npz(model_name, model_file, seed, classifier=False)


class BertShardForSequenceClassification(ModuleShard):
    """BERT shard + classification head (reference bert.py:164-219)."""

    def __init__(self, config: TransformerConfig, shard_config: ModuleShardConfig,
                 model_weights: Union[str, Mapping, None]):
        super().__init__(config, shard_config)
        self.classifier_weight = None
        if isinstance(model_weights, str):
            model_weights = dict(np.load(model_weights))
        bert_weights = None
        if model_weights is not None:
            # fine-tuned checkpoints prefix the encoder keys with 'bert.'
            bert_weights = {k[len("bert."):]: v for k, v in model_weights.items()
                            if k.startswith("bert.")}
        self.bert = BertModelShard(config, shard_config, bert_weights)
        if shard_config.is_last:
            self.classifier_weight = _param((config.num_labels, config.hidden_size))
            self.classifier_bias = _param((config.num_labels,))
            if model_weights is None:
                init_param(self.classifier_weight, f"{config.name}/classifier/weight")
                init_param(self.classifier_bias, f"{config.name}/classifier/bias",
                           "zeros")
            else:
                with torch.no_grad():
                    self.classifier_weight.copy_(_t(model_weights["classifier.weight"]))
                    self.classifier_bias.copy_(_t(model_weights["classifier.bias"]))

    @torch.no_grad()
    def forward(self, data):
        data = self.bert(data)
        if self.shard_config.is_last:
            data = ops.linear(data, self.classifier_weight, self.classifier_bias)
        return data

    @staticmethod
    def save_weights(model_name: str, model_file: str, seed: int = 0) -> None:
        save_random_bert_npz(model_name, model_file, seed, classifier=True)


def save_random_bert_npz(model_name: str, model_file: str, seed: int = 0,
                         classifier: bool = False) -> None:
    """Synthesize a random-init checkpoint in the HF BERT npz key layout."""
    from .config import get_config
    cfg = get_config(model_name)
    rng = np.random.default_rng(seed)
    d, i = cfg.hidden_size, cfg.intermediate_size

    def trunc(*shape):
        return np.clip(rng.standard_normal(shape) * 0.02, -0.04, 0.04).astype(np.float32)

    def zeros(*shape):
        return np.zeros(shape, dtype=np.float32)

    def ones(*shape):
        return np.ones(shape, dtype=np.float32)

    prefix = "bert." if classifier else ""
    out = {
        prefix + "embeddings.position_ids":
            np.arange(cfg.max_position_embeddings, dtype=np.int64)[None],
        prefix + "embeddings.word_embeddings.weight": trunc(cfg.vocab_size, d),
        prefix + "embeddings.position_embeddings.weight":
            trunc(cfg.max_position_embeddings, d),
        prefix + "embeddings.token_type_embeddings.weight":
            trunc(cfg.type_vocab_size, d),
        prefix + "embeddings.LayerNorm.weight": ones(d),
        prefix + "embeddings.LayerNorm.bias": zeros(d),
        prefix + "pooler.dense.weight": trunc(d, d),
        prefix + "pooler.dense.bias": zeros(d),
    }
    for blk in range(cfg.num_hidden_layers):
        root = f"{prefix}encoder.layer.{blk}."
        for n in ("query", "key", "value"):
            out[root + f"attention.self.{n}.weight"] = trunc(d, d)
            out[root + f"attention.self.{n}.bias"] = zeros(d)
        out[root + "attention.output.dense.weight"] = trunc(d, d)
        out[root + "attention.output.dense.bias"] = zeros(d)
        out[root + "attention.output.LayerNorm.weight"] = ones(d)
        out[root + "attention.output.LayerNorm.bias"] = zeros(d)
        out[root + "intermediate.dense.weight"] = trunc(i, d)
        out[root + "intermediate.dense.bias"] = zeros(i)
        out[root + "output.dense.weight"] = trunc(d, i)
        out[root + "output.dense.bias"] = zeros(d)
        out[root + "output.LayerNorm.weight"] = ones(d)
        out[root + "output.LayerNorm.bias"] = zeros(d)
    if classifier:
        out["classifier.weight"] = trunc(cfg.num_labels, d)
        out["classifier.bias"] = zeros(cfg.num_labels)
    np.savez(model_file, **out)
