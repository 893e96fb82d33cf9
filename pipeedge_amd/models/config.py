"""Offline model-architecture registry.

The reference resolves architectures with HF ``AutoConfig.from_pretrained``
(reference model_cfg.py:57-66), which needs network access. This build ships
the (public) architecture hyperparameters for the supported model zoo so the
framework is fully offline.
"""
import dataclasses


@dataclasses.dataclass
class TransformerConfig:
    """Architecture hyperparameters for one supported model."""
    name: str
    model_type: str                 # 'vit' | 'deit' | 'bert'
    hidden_size: int
    num_hidden_layers: int          # transformer blocks (schedulable layers = 4x)
    num_attention_heads: int
    intermediate_size: int
    num_labels: int = 0
    layer_norm_eps: float = 1e-12
    # vision
    image_size: int = 224
    patch_size: int = 16
    num_channels: int = 3
    distilled: bool = False         # DeiT: cls + distillation tokens
    # text (BERT)
    vocab_size: int = 0
    max_position_embeddings: int = 0
    type_vocab_size: int = 0
    pad_token_id: int = 0           # BERT [PAD]; masks attention over padding

    @property
    def head_dim(self) -> int:
        return self.hidden_size // self.num_attention_heads

    @property
    def num_patches(self) -> int:
        return (self.image_size // self.patch_size) ** 2

    @property
    def seq_len(self) -> int:
        """Token count for vision models (patches + cls [+ dist])."""
        return self.num_patches + (2 if self.distilled else 1)


def _vit(name, d, l, h, i, labels, patch=16, ln_eps=1e-12):
    return TransformerConfig(name=name, model_type="vit", hidden_size=d,
                             num_hidden_layers=l, num_attention_heads=h,
                             intermediate_size=i, num_labels=labels,
                             patch_size=patch, layer_norm_eps=ln_eps)


def _deit(name, d, l, h, i, labels=1000):
    return TransformerConfig(name=name, model_type="deit", hidden_size=d,
                             num_hidden_layers=l, num_attention_heads=h,
                             intermediate_size=i, num_labels=labels,
                             distilled=True)


def _bert(name, d, l, h, i, labels=0):
    return TransformerConfig(name=name, model_type="bert", hidden_size=d,
                             num_hidden_layers=l, num_attention_heads=h,
                             intermediate_size=i, num_labels=labels,
                             vocab_size=30522, max_position_embeddings=512,
                             type_vocab_size=2)


# Same model zoo as reference model_cfg.py:24-43.
MODEL_CONFIGS = {cfg.name: cfg for cfg in [
    _vit("google/vit-base-patch16-224", 768, 12, 12, 3072, 1000),
    _vit("google/vit-large-patch16-224", 1024, 24, 16, 4096, 1000),
    # ViT-Huge has no classifier in the public checkpoint; labels from in21k
    _vit("google/vit-huge-patch14-224-in21k", 1280, 32, 16, 5120, 21843, patch=14),
    _bert("bert-base-uncased", 768, 12, 12, 3072),
    _bert("bert-large-uncased", 1024, 24, 16, 4096),
    _bert("textattack/bert-base-uncased-CoLA", 768, 12, 12, 3072, labels=2),
    _deit("facebook/deit-base-distilled-patch16-224", 768, 12, 12, 3072),
    _deit("facebook/deit-small-distilled-patch16-224", 384, 12, 6, 1536),
    _deit("facebook/deit-tiny-distilled-patch16-224", 192, 12, 3, 768),
]}


def get_config(model_name: str) -> TransformerConfig:
    return MODEL_CONFIGS[model_name]
