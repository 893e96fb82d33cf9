"""pipeedge-amd: an MI355X-native pipelined transformer inference framework.

A from-scratch rebuild of the capabilities of usc-isi/PipeEdge (reference:
/root/reference) designed MI355X-first:

- model shards (ViT / DeiT / BERT, 4 schedulable sub-layers per transformer
  block) are placed across the GPUs of one node by a throughput-optimal DP
  scheduler (``sched-pipeline`` C++ binary, re-implemented);
- stage-to-stage activation hand-off uses RCCL point-to-point send/recv over
  xGMI (torch.distributed "nccl" backend == RCCL on ROCm) between GPU-resident
  buffers with a shape schema negotiated once at schedule time (the reference
  instead sends per-message dtype/shape headers over gloo/TCP with CPU staging
  - see reference comm/p2p/__init__.py:96-121 and devices.py:8-24);
- the per-stage hot path (patch-embed, attention, MLP GEMMs, LayerNorm,
  QuantPipe quant/dequant) is hand-written CDNA4 HIP (gfx950 MFMA/LDS tiling)
  in ``pipeedge_amd/ops/hip``.

Reference capability map: see SURVEY.md at the repo root.
"""

__version__ = "0.1.0"
