"""ViT model shards, MI355X-native.

Capability parity with reference ``src/pipeedge/models/transformers/vit.py``:
the same 4-schedulable-sublayers-per-block split (reference vit.py:41-70),
the same Google ``.npz`` checkpoint key layout (reference vit.py:120-159), and
the same shard classes (`ViTModelShard`, `ViTShardForImageClassification`).

Design differences (deliberate, MI355X-first):
- no HF modules: parameters are held directly in the MFMA-native ``[N, K]``
  GEMM layout (k-contiguous rows for both operands - see
  pipeedge_amd/ops/torch_ref.py) and all compute is routed through
  ``pipeedge_amd.ops`` (HIP kernels on gfx950, torch on CPU);
- QKV is packed into one ``[D, 3D]`` weight so self-attention is a single
  GEMM + one fused attention kernel;
- missing weights file => deterministic random init (synthetic benchmarking -
  there is no network in the deployment environment).

Sublayer semantics (identical to reference vit.py:55-70):
  0: LN_before + self-attention  -> (context, residual_input)
  1: attention out-proj + residual add -> hidden
  2: LN_after + MLP fc1 + GeLU   -> (mlp_hidden, residual_input)
  3: MLP fc2 + residual add      -> hidden
"""
import logging
import math
import zlib
from collections.abc import Mapping
from typing import Union

import numpy as np
import torch
from torch import Tensor, nn

from .. import ops
from . import ModuleShard, ModuleShardConfig
from .config import TransformerConfig

logger = logging.getLogger(__name__)


def _param(shape, dtype=torch.float32):
    return nn.Parameter(torch.empty(*shape, dtype=dtype), requires_grad=False)


def seed_for(*parts) -> int:
    """Deterministic cross-process seed from string parts (hash() is salted)."""
    return zlib.crc32("|".join(str(p) for p in parts).encode()) & 0x7FFFFFFF


def init_param(p: Tensor, key: str, kind: str = "trunc"):
    """Init one parameter from its own seeded generator, so random-init is
    identical regardless of how the model is sharded."""
    gen = torch.Generator().manual_seed(seed_for(key))
    with torch.no_grad():
        if kind == "trunc":
            nn.init.trunc_normal_(p, std=0.02, generator=gen)
        elif kind == "zeros":
            p.zero_()
        elif kind == "ones":
            p.fill_(1.0)
        else:
            raise ValueError(kind)


def _t(arr) -> Tensor:
    return torch.from_numpy(np.ascontiguousarray(arr))


class ViTEmbeddings(nn.Module):
    """Patch projection + cls token + position embeddings.

    The 16x16/s16 conv is expressed as patchify (pure data movement) + GEMM,
    which is the shape MFMA wants; weight layout ``[D, C*P*P]`` with patch
    pixels ordered (c, ph, pw).
    """

    def __init__(self, config: TransformerConfig):
        super().__init__()
        self.config = config
        d = config.hidden_size
        pdim = config.num_channels * config.patch_size ** 2
        self.patch_weight = _param((d, pdim))
        self.patch_bias = _param((d,))
        self.cls_token = _param((1, 1, d))
        self.pos_embed = _param((1, config.seq_len, d))

    def init_random(self, prefix: str):
        for name, p in self.named_parameters():
            kind = "zeros" if name.endswith("_bias") else "trunc"
            init_param(p, f"{prefix}/emb/{name}", kind)

    def forward(self, pixels: Tensor) -> Tensor:
        patches = ops.patchify(pixels, self.config.patch_size)
        x = ops.linear(patches, self.patch_weight, self.patch_bias)
        return ops.embed_finish(x, self.cls_token.reshape(1, -1).to(x.dtype),
                                self.pos_embed[0].to(x.dtype))


class ViTLayerShard(ModuleShard):
    """Sublayers [layer_start, layer_end] (subset of 0..3) of ONE block."""

    def __init__(self, config: TransformerConfig, shard_config: ModuleShardConfig):
        super().__init__(config, shard_config)
        d = config.hidden_size
        i = config.intermediate_size
        self.scale = 1.0 / math.sqrt(config.head_dim)
        if self.has_layer(0):
            self.ln1_weight = _param((d,))
            self.ln1_bias = _param((d,))
            self.qkv_weight = _param((3 * d, d))
            self.qkv_bias = _param((3 * d,))
        if self.has_layer(1):
            self.out_weight = _param((d, d))
            self.out_bias = _param((d,))
        if self.has_layer(2):
            self.ln2_weight = _param((d,))
            self.ln2_bias = _param((d,))
            self.fc1_weight = _param((i, d))
            self.fc1_bias = _param((i,))
        if self.has_layer(3):
            self.fc2_weight = _param((d, i))
            self.fc2_bias = _param((d,))

    def init_random(self, prefix: str):
        for name, p in self.named_parameters():
            if name.endswith("_bias"):
                kind = "zeros"
            elif name.startswith("ln"):
                kind = "ones"
            else:
                kind = "trunc"
            init_param(p, f"{prefix}/block{self.block_id}/{name}", kind)

    def _ln_folds(self):
        """Lazy folded-LN GEMM operands (GPU path; see ops.make_ln_fold)."""
        if getattr(self, "_fold_cache", None) is None:
            cache = {}
            with torch.no_grad():
                if self.has_layer(0):
                    cache["qkv"] = ops.make_ln_fold(
                        self.qkv_weight, self.ln1_weight, self.ln1_bias)
                if self.has_layer(2):
                    cache["fc1"] = ops.make_ln_fold(
                        self.fc1_weight, self.ln2_weight, self.ln2_bias)
            self._fold_cache = cache
        return self._fold_cache

    @torch.no_grad()
    def forward(self, data, stats=None):
        """Compute the shard's sublayers.

        ``stats``: per-row (sum, sumsq) of the incoming hidden state when the
        producing GEMM emitted them (GPU folded-LN path); returns
        (data, stats) when called via forward_with_stats.
        """
        data, _ = self.forward_with_stats(data, stats)
        return data

    @torch.no_grad()
    def forward_with_stats(self, data, stats=None):
        cfg = self.config
        eps = cfg.layer_norm_eps
        if self.has_layer(0):
            if ops.fused_ln_active(data):
                qkv = ops.ln_linear(data, self.ln1_weight, self.ln1_bias,
                                    eps, self.qkv_weight, self.qkv_bias,
                                    stats=stats,
                                    fused=self._ln_folds()["qkv"])
            else:
                normed = ops.layernorm(data, self.ln1_weight, self.ln1_bias,
                                       eps)
                qkv = ops.linear(normed, self.qkv_weight, self.qkv_bias)
            ctx = ops.qkv_attention(qkv, cfg.num_attention_heads, self.scale)
            data = (ctx, data)
            stats = None
        if self.has_layer(1):
            if ops.fused_ln_active(data[0]):
                data, stats = ops.linear(data[0], self.out_weight,
                                         self.out_bias, residual=data[1],
                                         want_stats=True)
            else:
                data = ops.linear(data[0], self.out_weight, self.out_bias,
                                  residual=data[1])
                stats = None
        if self.has_layer(2):
            if ops.fused_ln_active(data):
                mid = ops.ln_linear(data, self.ln2_weight, self.ln2_bias,
                                    eps, self.fc1_weight, self.fc1_bias,
                                    act="gelu", stats=stats,
                                    fused=self._ln_folds()["fc1"])
            else:
                normed = ops.layernorm(data, self.ln2_weight, self.ln2_bias,
                                       eps)
                mid = ops.linear(normed, self.fc1_weight, self.fc1_bias,
                                 act="gelu")
            data = (mid, data)
            stats = None
        if self.has_layer(3):
            if ops.fused_ln_active(data[0]):
                data, stats = ops.linear(data[0], self.fc2_weight,
                                         self.fc2_bias, residual=data[1],
                                         want_stats=True)
            else:
                data = ops.linear(data[0], self.fc2_weight, self.fc2_bias,
                                  residual=data[1])
                stats = None
        return data, stats


class ViTModelShard(ModuleShard):
    """Layers [layer_start, layer_end] of the full ViT encoder.

    First shard owns embeddings, last owns the final LayerNorm
    (reference vit.py:94-118).
    """

    # embeddings class is overridden by DeiT
    embeddings_cls = ViTEmbeddings

    def __init__(self, config: TransformerConfig, shard_config: ModuleShardConfig,
                 model_weights: Union[str, Mapping, None]):
        super().__init__(config, shard_config)
        self.embeddings = None
        self.layers = nn.ModuleList()
        self.layernorm = None
        self._build()
        load_npz_or_random(self, model_weights)

    def _build(self):
        if self.shard_config.is_first:
            self.embeddings = self.embeddings_cls(self.config)
        # map global layer range -> (block, sublayer range) pieces
        for block_id, sub_start, sub_end in iter_block_ranges(
                self.shard_config.layer_start, self.shard_config.layer_end):
            cfg = ModuleShardConfig(layer_start=sub_start, layer_end=sub_end)
            layer = ViTLayerShard(self.config, cfg)
            layer.block_id = block_id
            self.layers.append(layer)
        if self.shard_config.is_last:
            d = self.config.hidden_size
            self.layernorm_weight = _param((d,))
            self.layernorm_bias = _param((d,))

    def init_random(self, prefix: str):
        if self.embeddings is not None:
            self.embeddings.init_random(prefix)
        for layer in self.layers:
            layer.init_random(prefix)
        if self.shard_config.is_last:
            init_param(self.layernorm_weight, f"{prefix}/ln_f/weight", "ones")
            init_param(self.layernorm_bias, f"{prefix}/ln_f/bias", "zeros")

    @torch.no_grad()
    def load_npz(self, weights: Mapping):
        if self.embeddings is not None:
            emb = self.embeddings
            emb.cls_token.copy_(_t(weights["cls"]))
            emb.pos_embed.copy_(_t(weights["Transformer/posembed_input/pos_embedding"]))
            # npz conv kernel is [P, P, C, D]; ours is [D, (c, ph, pw)]
            kern = weights["embedding/kernel"].transpose(3, 2, 0, 1)
            emb.patch_weight.copy_(_t(kern).reshape(emb.patch_weight.shape))
            emb.patch_bias.copy_(_t(weights["embedding/bias"]))
        d = self.config.hidden_size
        for layer in self.layers:
            root = f"Transformer/encoderblock_{layer.block_id}/"
            attn = root + "MultiHeadDotProductAttention_1/"
            if layer.has_layer(0):
                layer.ln1_weight.copy_(_t(weights[root + "LayerNorm_0/scale"]))
                layer.ln1_bias.copy_(_t(weights[root + "LayerNorm_0/bias"]))
                # npz kernels are [D(in), H, dh] -> transpose to [N, K]
                qkv_w = [_t(weights[attn + f"{n}/kernel"]).reshape(d, d).t()
                         for n in ("query", "key", "value")]
                layer.qkv_weight.copy_(torch.cat(qkv_w, dim=0))
                qkv_b = [_t(weights[attn + f"{n}/bias"]).reshape(-1)
                         for n in ("query", "key", "value")]
                layer.qkv_bias.copy_(torch.cat(qkv_b))
            if layer.has_layer(1):
                layer.out_weight.copy_(_t(weights[attn + "out/kernel"]).reshape(d, d).t())
                layer.out_bias.copy_(_t(weights[attn + "out/bias"]).reshape(-1))
            if layer.has_layer(2):
                layer.ln2_weight.copy_(_t(weights[root + "LayerNorm_2/scale"]))
                layer.ln2_bias.copy_(_t(weights[root + "LayerNorm_2/bias"]))
                layer.fc1_weight.copy_(_t(weights[root + "MlpBlock_3/Dense_0/kernel"]).t())
                layer.fc1_bias.copy_(_t(weights[root + "MlpBlock_3/Dense_0/bias"]))
            if layer.has_layer(3):
                layer.fc2_weight.copy_(_t(weights[root + "MlpBlock_3/Dense_1/kernel"]).t())
                layer.fc2_bias.copy_(_t(weights[root + "MlpBlock_3/Dense_1/bias"]))
        if self.shard_config.is_last:
            self.layernorm_weight.copy_(_t(weights["Transformer/encoder_norm/scale"]))
            self.layernorm_bias.copy_(_t(weights["Transformer/encoder_norm/bias"]))

    @torch.no_grad()
    def forward(self, data):
        if self.shard_config.is_first:
            data = self.embeddings(data)
        stats = None  # per-row hidden stats threaded between folded-LN GEMMs
        for layer in self.layers:
            data, stats = layer.forward_with_stats(data, stats)
        if self.shard_config.is_last:
            data = ops.layernorm(data, self.layernorm_weight, self.layernorm_bias,
                                 self.config.layer_norm_eps)
        return data


class ViTShardForImageClassification(ModuleShard):
    """ViT shard + classifier head on the last shard (reference vit.py:189-226)."""

    model_shard_cls = ViTModelShard

    def __init__(self, config: TransformerConfig, shard_config: ModuleShardConfig,
                 model_weights: Union[str, Mapping, None]):
        super().__init__(config, shard_config)
        self.classifier_weight = None
        if isinstance(model_weights, str):
            model_weights = open_npz(model_weights)
        self.vit = self.model_shard_cls(config, shard_config, model_weights)
        if shard_config.is_last and config.num_labels > 0:
            self.classifier_weight = _param((config.num_labels, config.hidden_size))
            self.classifier_bias = _param((config.num_labels,))
            if model_weights is None:
                init_param(self.classifier_weight, f"{config.name}/head/weight")
                init_param(self.classifier_bias, f"{config.name}/head/bias", "zeros")
            else:
                with torch.no_grad():
                    # npz head/kernel is [D, num_labels] -> [N, K]
                    self.classifier_weight.copy_(_t(model_weights["head/kernel"]).t())
                    self.classifier_bias.copy_(_t(model_weights["head/bias"]))

    @torch.no_grad()
    def forward(self, data):
        data = self.vit(data)
        if self.shard_config.is_last and self.classifier_weight is not None:
            # classify on the CLS token
            data = ops.linear(data[:, 0, :], self.classifier_weight,
                              self.classifier_bias)
        return data

    @staticmethod
    def save_weights(model_name: str, model_file: str, seed: int = 0) -> None:
        """Write a weights file in the Google npz layout.

        The reference downloads pretrained checkpoints (vit.py:172-186); this
        environment has no network, so we synthesize a deterministic
        random-init checkpoint with the same key layout instead.
        """
        save_random_npz(model_name, model_file, seed)


# ---------------------------------------------------------------------------
# helpers shared with deit.py
# ---------------------------------------------------------------------------

def iter_block_ranges(layer_start: int, layer_end: int):
    """Yield (block_id, sublayer_start, sublayer_end) covering the 1-indexed
    global layer range, 4 sublayers per block (reference vit.py:99-113)."""
    layer_curr = layer_start
    while layer_curr <= layer_end:
        block_id = (layer_curr - 1) // 4
        sub_start = (layer_curr - 1) % 4
        if block_id == (layer_end - 1) // 4:
            sub_end = (layer_end - 1) % 4
        else:
            sub_end = 3
        yield block_id, sub_start, sub_end
        layer_curr += sub_end - sub_start + 1


def open_npz(path: str) -> Mapping:
    return np.load(path)


def load_npz_or_random(shard, model_weights: Union[str, Mapping, None]):
    """Load shard weights from an npz path/mapping, or random-init if None."""
    if isinstance(model_weights, str):
        with np.load(model_weights) as weights:
            shard.load_npz(weights)
    elif model_weights is not None:
        shard.load_npz(model_weights)
    else:
        shard.init_random(shard.config.name)
        logger.warning("no weights file: random-initialized %s layers [%d,%d]",
                       shard.config.name, shard.shard_config.layer_start,
                       shard.shard_config.layer_end)


def save_random_npz(model_name: str, model_file: str, seed: int = 0) -> None:
    """Synthesize a random-init checkpoint in the Google ViT npz layout."""
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
        "cls": trunc(1, 1, d),
        "Transformer/posembed_input/pos_embedding": trunc(1, cfg.seq_len, d),
        "embedding/kernel": trunc(p, p, c, d),
        "embedding/bias": zeros(d),
        "Transformer/encoder_norm/scale": ones(d),
        "Transformer/encoder_norm/bias": zeros(d),
        "head/kernel": trunc(d, cfg.num_labels),
        "head/bias": zeros(cfg.num_labels),
    }
    dh = cfg.head_dim
    h = cfg.num_attention_heads
    for blk in range(cfg.num_hidden_layers):
        root = f"Transformer/encoderblock_{blk}/"
        attn = root + "MultiHeadDotProductAttention_1/"
        out[root + "LayerNorm_0/scale"] = ones(d)
        out[root + "LayerNorm_0/bias"] = zeros(d)
        out[root + "LayerNorm_2/scale"] = ones(d)
        out[root + "LayerNorm_2/bias"] = zeros(d)
        for n in ("query", "key", "value"):
            out[attn + f"{n}/kernel"] = trunc(d, h, dh)
            out[attn + f"{n}/bias"] = zeros(h, dh)
        out[attn + "out/kernel"] = trunc(h, dh, d)
        out[attn + "out/bias"] = zeros(d)
        out[root + "MlpBlock_3/Dense_0/kernel"] = trunc(d, i)
        out[root + "MlpBlock_3/Dense_0/bias"] = zeros(i)
        out[root + "MlpBlock_3/Dense_1/kernel"] = trunc(i, d)
        out[root + "MlpBlock_3/Dense_1/bias"] = zeros(d)
    np.savez(model_file, **out)
