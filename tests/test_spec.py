"""Wire-schema unit tests (models/spec.py): the static shapes that replace
the reference's per-message headers must match the documented boundary
shapes."""
import torch

import model_cfg
from pipeedge_amd.models import spec
from pipeedge_amd.models.config import get_config


def test_vit_boundary_shapes():
    cfg = get_config("google/vit-base-patch16-224")
    # after sublayer 0 (layer 1): (context, residual)
    s = spec.boundary_spec(cfg, 1, 8, torch.bfloat16)
    assert [t.shape for t in s] == [(8, 197, 768), (8, 197, 768)]
    # after sublayer 2 (layer 3): (mlp_hidden, residual)
    s = spec.boundary_spec(cfg, 3, 8, torch.bfloat16)
    assert [t.shape for t in s] == [(8, 197, 3072), (8, 197, 768)]
    # after sublayer 1/3: single hidden
    for layer in (2, 4, 24):
        s = spec.boundary_spec(cfg, layer, 8, torch.bfloat16)
        assert [t.shape for t in s] == [(8, 197, 768)]


def test_input_output_specs():
    cfg = get_config("google/vit-base-patch16-224")
    (inp,) = spec.input_spec(cfg, 8, torch.bfloat16)
    assert inp.shape == (8, 3, 224, 224)
    (out,) = spec.output_spec(cfg, 8, torch.bfloat16)
    assert out.shape == (8, 1000)
    bcfg = get_config("textattack/bert-base-uncased-CoLA")
    (binp,) = spec.input_spec(bcfg, 4, torch.bfloat16)
    assert binp.shape == (4, 512) and binp.dtype == torch.int64
    (bout,) = spec.output_spec(bcfg, 4, torch.bfloat16)
    assert bout.shape == (4, 2)
    # BertModelShard (no classifier): pooled hidden
    mcfg = get_config("bert-base-uncased")
    (mout,) = spec.output_spec(mcfg, 4, torch.bfloat16)
    assert mout.shape == (4, 768)


def test_final_stage_boundary_is_output():
    cfg = get_config("google/vit-base-patch16-224")
    s = spec.stage_boundary_spec(cfg, 48, 48, 8, torch.bfloat16)
    assert [t.shape for t in s] == [(8, 1000)]


def test_quant_wire_spec_five_tensors_per_source():
    cfg = get_config("google/vit-base-patch16-224")
    base = spec.boundary_spec(cfg, 1, 4, torch.bfloat16)  # tuple boundary
    wire = spec.quant_wire_spec(base, 8)
    assert len(wire) == 10  # 5 per source tensor
    assert wire[0].dtype == torch.uint8
    assert wire[0].shape == (4, (197 * 768 // 4) * 4)
    assert wire[1].shape == (4, 2)  # per-image shape dims
    assert wire[4].dtype == torch.int8


def test_stage_wire_schemas_cover_all_edges():
    edges = model_cfg.stage_wire_schemas(
        "google/vit-base-patch16-224", [(1, 24), (25, 48)], 8, [8, 0],
        torch.bfloat16)
    assert len(edges) == 3  # input, stage0->1 (quantized), results
    assert edges[0][0].shape == (8, 3, 224, 224)
    assert len(edges[1]) == 5 and edges[1][0].dtype == torch.uint8
    assert edges[2][0].shape == (8, 1000)


def test_schemas_match_actual_shard_outputs():
    """The schema must agree with what the shards actually emit."""
    from pipeedge_amd.models import ModuleShardConfig, vit
    cfg = get_config("google/vit-base-patch16-224")
    for a, b in [(1, 1), (1, 3), (1, 24)]:
        sc = ModuleShardConfig(layer_start=a, layer_end=b, is_first=True,
                               is_last=False)
        shard = vit.ViTShardForImageClassification(cfg, sc, None)
        out = shard(torch.randn(2, 3, 224, 224))
        predicted = spec.boundary_spec(cfg, b, 2, torch.float32)
        tensors = (out,) if isinstance(out, torch.Tensor) else out
        assert [tuple(t.shape) for t in tensors] == \
            [tuple(s.shape) for s in predicted]
