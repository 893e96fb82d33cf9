"""Model shard tests: full-model vs stitched-shard equivalence (the test the
reference lacks - SURVEY.md section 4), npz checkpoint round-trips, and layer
range mapping."""
import numpy as np
import pytest
import torch

from pipeedge_amd.models import ModuleShardConfig, get_microbatch_size
from pipeedge_amd.models.config import get_config
from pipeedge_amd.models import bert, deit, vit
from pipeedge_amd.models.vit import iter_block_ranges


def _shard_cfg(a, b, total):
    return ModuleShardConfig(layer_start=a, layer_end=b,
                             is_first=(a == 1), is_last=(b == total))


def test_iter_block_ranges():
    # layers 1..8 = blocks 0,1 full
    assert list(iter_block_ranges(1, 8)) == [(0, 0, 3), (1, 0, 3)]
    # mid-block split: layers 2..7
    assert list(iter_block_ranges(2, 7)) == [(0, 1, 3), (1, 0, 2)]
    # single sublayer
    assert list(iter_block_ranges(6, 6)) == [(1, 1, 1)]


def test_get_microbatch_size():
    t = torch.zeros(8, 3)
    assert get_microbatch_size(t) == 8
    assert get_microbatch_size((t, torch.zeros(8, 5)), verify=True) == 8


@pytest.mark.parametrize("partition", [
    [(1, 48)],
    [(1, 24), (25, 48)],
    [(1, 17), (18, 30), (31, 48)],
    [(1, 2), (3, 13), (14, 14), (15, 48)],  # mid-block boundaries
])
def test_vit_stitched_equals_full(partition):
    cfg = get_config("google/vit-base-patch16-224")
    full = vit.ViTShardForImageClassification(cfg, _shard_cfg(1, 48, 48), None)
    torch.manual_seed(0)
    x = torch.randn(2, 3, 224, 224)
    y_full = full(x)
    data = x
    for a, b in partition:
        shard = vit.ViTShardForImageClassification(cfg, _shard_cfg(a, b, 48), None)
        data = shard(data)
    assert torch.equal(data, y_full)


def test_vit_npz_roundtrip(tmp_weights_dir):
    name = "google/vit-base-patch16-224"
    path = str(tmp_weights_dir / "vitb.npz")
    vit.ViTShardForImageClassification.save_weights(name, path, seed=1)
    cfg = get_config(name)
    m1 = vit.ViTShardForImageClassification(cfg, _shard_cfg(1, 48, 48), path)
    m2 = vit.ViTShardForImageClassification(cfg, _shard_cfg(1, 48, 48), path)
    torch.manual_seed(0)
    x = torch.randn(1, 3, 224, 224)
    assert torch.equal(m1(x), m2(x))
    # weights actually loaded (not random-init): classifier comes from npz
    with np.load(path) as w:
        assert np.allclose(m1.classifier_weight.numpy(), w["head/kernel"].T)


def test_deit_stitched_equals_full(tmp_weights_dir):
    name = "facebook/deit-tiny-distilled-patch16-224"
    path = str(tmp_weights_dir / "deit-t.npz")
    deit.DeiTShardForImageClassification.save_weights(name, path, seed=2)
    cfg = get_config(name)
    full = deit.DeiTShardForImageClassification(cfg, _shard_cfg(1, 48, 48), path)
    torch.manual_seed(0)
    x = torch.randn(2, 3, 224, 224)
    y_full = full(x)
    data = x
    for a, b in [(1, 10), (11, 35), (36, 48)]:
        data = deit.DeiTShardForImageClassification(cfg, _shard_cfg(a, b, 48), path)(data)
    assert torch.equal(data, y_full)
    # distilled: sequence has 198 tokens
    assert cfg.seq_len == 198


def test_bert_stitched_equals_full(tmp_weights_dir):
    name = "textattack/bert-base-uncased-CoLA"
    path = str(tmp_weights_dir / "bert-cola.npz")
    bert.BertShardForSequenceClassification.save_weights(name, path, seed=3)
    cfg = get_config(name)
    full = bert.BertShardForSequenceClassification(cfg, _shard_cfg(1, 48, 48), path)
    torch.manual_seed(0)
    ids = torch.randint(0, cfg.vocab_size, (2, 64))
    y_full = full(ids)
    assert y_full.shape == (2, 2)
    data = ids
    for a, b in [(1, 23), (24, 24), (25, 48)]:
        data = bert.BertShardForSequenceClassification(cfg, _shard_cfg(a, b, 48), path)(data)
    assert torch.equal(data, y_full)


def test_bert_model_shard_pooler_output():
    cfg = get_config("bert-base-uncased")
    m = bert.BertModelShard(cfg, _shard_cfg(1, 48, 48), None)
    ids = torch.randint(0, cfg.vocab_size, (3, 32))
    y = m(ids)
    assert y.shape == (3, cfg.hidden_size)  # pooled CLS
    assert torch.all(y.abs() <= 1.0)  # tanh range


def test_random_init_is_deterministic_across_processes():
    # seeds derive from crc32, not salted hash()
    from pipeedge_amd.models.vit import seed_for
    assert seed_for("a", 1, "b") == seed_for("a", 1, "b")
    assert seed_for("x") != seed_for("y")


def test_tuple_activation_boundary_shapes():
    """A boundary mid-attention passes (context, residual) tuples; shapes must
    match the reference's documented wire shapes."""
    cfg = get_config("google/vit-base-patch16-224")
    s1 = vit.ViTShardForImageClassification(cfg, _shard_cfg(1, 1, 48), None)
    torch.manual_seed(0)
    x = torch.randn(4, 3, 224, 224)
    out = s1(x)
    assert isinstance(out, tuple) and len(out) == 2
    assert out[0].shape == (4, 197, 768)
    assert out[1].shape == (4, 197, 768)
    s2 = vit.ViTShardForImageClassification(cfg, _shard_cfg(2, 3, 48), None)
    out2 = s2(out)
    assert isinstance(out2, tuple)
    assert out2[0].shape == (4, 197, 3072)  # mlp hidden after sublayer 2


def test_vit_huge_head_dim_80_shard():
    """ViT-Huge: patch 14, head dim 80 (the DHP=96 attention path)."""
    cfg = get_config("google/vit-huge-patch14-224-in21k")
    assert cfg.head_dim == 80
    assert cfg.seq_len == 257
    shard = vit.ViTShardForImageClassification(cfg, _shard_cfg(1, 4, 128), None)
    out = shard(torch.randn(1, 3, 224, 224))
    assert out.shape == (1, 257, 1280)


def test_bert_padding_mask_matches_unpadded():
    """Padded-batch outputs must equal per-sequence unpadded outputs: the
    first shard derives the additive mask from pad ids and attention ignores
    padding (the reference's shards run mask-free over padding and get this
    wrong - reference bert.py:41-52)."""
    cfg = get_config("textattack/bert-base-uncased-CoLA")
    total = 48
    full = bert.BertShardForSequenceClassification(
        cfg, _shard_cfg(1, total, total), None)
    gen = torch.Generator().manual_seed(7)
    s_full, s_short = 64, 40
    ids = torch.randint(1000, cfg.vocab_size, (2, s_full), generator=gen)
    ids[1, s_short:] = cfg.pad_token_id  # sequence 1 padded
    y_padded = full(ids)
    # unpadded reference for sequence 1: same tokens, no pad positions
    y_short = full(ids[1:2, :s_short])
    assert torch.allclose(y_padded[1], y_short[0], atol=1e-4, rtol=1e-4), \
        (y_padded[1] - y_short[0]).abs().max()
    # and sequence 0 (no padding) unchanged vs its own unpadded run
    # (allclose: batch-1 vs batch-2 matmul reduction order differs on CPU)
    y0 = full(ids[0:1])
    assert torch.allclose(y_padded[0], y0[0], atol=1e-6, rtol=1e-6)


@pytest.mark.parametrize("partition", [
    [(1, 24), (25, 48)],
    [(1, 2), (3, 13), (14, 14), (15, 48)],  # mid-block boundaries
])
def test_bert_masked_stitched_equals_full(partition):
    """The mask must survive every stage boundary type (1-, 2-tensor
    payloads + trailing mask)."""
    cfg = get_config("textattack/bert-base-uncased-CoLA")
    full = bert.BertShardForSequenceClassification(
        cfg, _shard_cfg(1, 48, 48), None)
    gen = torch.Generator().manual_seed(8)
    ids = torch.randint(1000, cfg.vocab_size, (2, 48), generator=gen)
    ids[0, 30:] = cfg.pad_token_id
    ids[1, 17:] = cfg.pad_token_id
    y_full = full(ids)
    data = ids
    for a, b in partition:
        shard = bert.BertShardForSequenceClassification(
            cfg, _shard_cfg(a, b, 48), None)
        data = shard(data)
    assert torch.equal(data, y_full)


def test_bert_mask_broadcast_equivalence():
    """torch_ref [B,S] mask must equal an explicit [B,1,1,S] extended mask."""
    from pipeedge_amd.ops import torch_ref
    torch.manual_seed(2)
    b, s, h, dh = 2, 32, 4, 16
    qkv = torch.randn(b, s, 3 * h * dh)
    mask = torch.zeros(b, s)
    mask[0, 20:] = -1e9
    mask[1, 10:] = -1e9
    y2d = torch_ref.qkv_attention(qkv, h, attn_mask=mask)
    y4d = torch_ref.qkv_attention(qkv, h,
                                  attn_mask=mask.reshape(b, 1, 1, s))
    assert torch.allclose(y2d, y4d)
