"""bench.py driver-contract tests on CPU: JSON line shape, and the
distributed launch path (torchrun, 2 ranks, gloo) the driver uses."""
import json
import os
import subprocess
import sys

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def _check_json(line):
    out = json.loads(line)
    assert out["metric"].startswith("images/sec")
    assert out["unit"] == "images/s"
    assert out["value"] > 0
    assert out["higher_is_better"] is True
    assert out["scaling"] == "strong"
    assert out["data"] == "synthetic"
    assert out["config"]["global_batch"] == out["config"]["ubatch_size"] * \
        (out["config"]["global_batch"] // out["config"]["ubatch_size"])
    assert out["vs_baseline"] is not None
    return out


def test_bench_single_rank_cpu():
    env = dict(os.environ)
    env.pop("RANK", None)
    env.pop("WORLD_SIZE", None)
    proc = subprocess.run(
        [sys.executable, os.path.join(REPO, "bench.py"), "--steps", "2",
         "--warmup", "1", "--batch-size", "8", "--ubatch-size", "4",
         "--model", "facebook/deit-tiny-distilled-patch16-224"],
        capture_output=True, timeout=600, env=env, cwd=REPO, check=True)
    line = proc.stdout.decode().strip().splitlines()[-1]
    out = _check_json(line)
    assert out["n_gpus"] == 1
    assert out["steps"] == 2


def test_bench_torchrun_two_ranks_cpu():
    """The exact launch form the driver uses for N > 1."""
    env = dict(os.environ)
    env.pop("RANK", None)
    env.pop("WORLD_SIZE", None)
    proc = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
         "--master-port", "29641", os.path.join(REPO, "bench.py"),
         "--gpus", "2", "--steps", "2", "--warmup", "1",
         "--batch-size", "8", "--ubatch-size", "4",
         "--model", "facebook/deit-tiny-distilled-patch16-224"],
        capture_output=True, timeout=600, env=env, cwd=REPO, check=False)
    assert proc.returncode == 0, proc.stderr.decode()[-2000:]
    lines = [l for l in proc.stdout.decode().splitlines()
             if l.startswith("{")]
    out = _check_json(lines[-1])
    assert out["n_gpus"] == 2
    assert out["config"]["parallelism"] == "pp2"
    assert out["config"]["partition"] == [[1, 24], [25, 48]]


def test_bench_torchrun_eight_ranks_cpu():
    """Rehearsal of the driver's 8-rank launch (gloo on CPU)."""
    env = dict(os.environ, OMP_NUM_THREADS="2")
    env.pop("RANK", None)
    env.pop("WORLD_SIZE", None)
    proc = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", "8", "--master-addr", "127.0.0.1",
         "--master-port", "29643", os.path.join(REPO, "bench.py"),
         "--gpus", "8", "--steps", "2", "--warmup", "1",
         "--batch-size", "8", "--ubatch-size", "4",
         "--model", "facebook/deit-tiny-distilled-patch16-224"],
        capture_output=True, timeout=600, env=env, cwd=REPO, check=False)
    assert proc.returncode == 0, proc.stderr.decode()[-2000:]
    lines = [l for l in proc.stdout.decode().splitlines()
             if l.startswith("{")]
    out = _check_json(lines[-1])
    assert out["n_gpus"] == 8
    assert out["config"]["partition"][0] == [1, 6]
    assert out["config"]["partition"][-1] == [43, 48]


def test_bench_bert_four_stage_cpu():
    """BASELINE config #4 shape: BERT-CoLA 4-stage, seq_len=128."""
    env = dict(os.environ, OMP_NUM_THREADS="2")
    env.pop("RANK", None)
    env.pop("WORLD_SIZE", None)
    proc = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", "4", "--master-addr", "127.0.0.1",
         "--master-port", "29645", os.path.join(REPO, "bench.py"),
         "--gpus", "4", "--steps", "2", "--warmup", "1",
         "--batch-size", "8", "--ubatch-size", "4", "--seq-len", "128",
         "--model", "textattack/bert-base-uncased-CoLA"],
        capture_output=True, timeout=600, env=env, cwd=REPO, check=False)
    assert proc.returncode == 0, proc.stderr.decode()[-2000:]
    lines = [l for l in proc.stdout.decode().splitlines()
             if l.startswith("{")]
    out = json.loads(lines[-1])
    assert out["n_gpus"] == 4
    assert out["config"]["seq_len"] == 128


def test_bench_two_rank_deep_stream():
    """Driver-shaped torchrun bench at a DEPTH that exceeds the 2-stage
    chain's buffer capacity (~20 in-flight): regression for the results
    back-edge deadlock the admission control fixes - the real SCALE run
    streams steps x n_ubatch microbatches continuously."""
    env = dict(os.environ, OMP_NUM_THREADS="2")
    proc = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
         "--master-port", "29671", os.path.join(REPO, "bench.py"),
         "--gpus", "2", "--steps", "40", "--warmup", "5",
         "--batch-size", "8", "--ubatch-size", "2",
         "--model", "facebook/deit-tiny-distilled-patch16-224"],
        capture_output=True, timeout=600, env=env, cwd=REPO, check=False)
    assert proc.returncode == 0, proc.stderr.decode()[-2000:]
    line = [l for l in proc.stdout.decode().splitlines()
            if l.startswith("{")][-1]
    out = json.loads(line)
    assert out["n_gpus"] == 2 and out["steps"] == 40 and out["value"] > 0
