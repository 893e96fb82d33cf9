"""End-to-end runtime.py mode tests: RPC transport, adaptive quantization,
auto-scheduling from YAML files (all 2-rank CPU/gloo subprocess runs)."""
import multiprocessing as mp
import os
import subprocess
import sys

import yaml

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
MODEL = "facebook/deit-tiny-distilled-patch16-224"


def _worker(rank, port, extra_args, env_extra, result_q, cwd):
    env = dict(os.environ, MASTER_ADDR="127.0.0.1", MASTER_PORT=str(port),
               PYTHONPATH=REPO, **env_extra)
    cmd = [sys.executable, os.path.join(REPO, "runtime.py"), str(rank), "2",
           "-d", "cpu", "--port", str(port)] + extra_args
    proc = subprocess.run(cmd, capture_output=True, timeout=300, env=env,
                          cwd=cwd or REPO, check=False)
    result_q.put((rank, proc.returncode, proc.stdout.decode(),
                  proc.stderr.decode()[-1200:]))


def _run_two(port, extra_args, env_extra=None, cwd=None):
    ctx = mp.get_context("spawn")
    result_q = ctx.Queue()
    procs = [ctx.Process(target=_worker,
                         args=(r, port, extra_args, env_extra or {},
                               result_q, cwd))
             for r in range(2)]
    for p in procs:
        p.start()
    outs = {}
    for _ in range(2):
        rank, rc, out, err = result_q.get(timeout=300)
        outs[rank] = (rc, out, err)
    for p in procs:
        p.join(timeout=30)
    return outs


def test_runtime_rpc_two_rank():
    outs = _run_two(29631, ["-c", "rpc", "-b", "4", "-u", "2", "-m", MODEL,
                            "-pt", "1,24,25,48"])
    assert outs[0][0] == 0, outs[0][2]
    assert outs[1][0] == 0, outs[1][2]
    assert "throughput is" in outs[0][1]


def test_runtime_adaptive_quant_controller(tmp_path):
    env = {"ADAPTIVE_QUANT": "CONTROLLER", "SEND_CONSTRAINT": "100",
           "WINDOW_SIZE": "2"}
    outs = _run_two(29633, ["-b", "8", "-u", "2", "-m", MODEL,
                            "-pt", "1,24,25,48", "-q", "8,0"],
                    env_extra=env, cwd=str(tmp_path))
    assert outs[0][0] == 0, outs[0][2]
    assert outs[1][0] == 0, outs[1][2]
    assert "throughput is" in outs[0][1]
    # the adaptive controller fired at the window boundary on rank 0
    assert "Adaptive quantization (controller)" in outs[0][1], outs[0][1]


def test_runtime_rpc_quantized_adaptive(tmp_path):
    """RPC-path quant parity (reference runtime.py:558-565): `-c rpc -q 8,0`
    must quantize the stage-0 -> stage-1 hop and run the adaptive policy."""
    env = {"ADAPTIVE_QUANT": "CONTROLLER", "SEND_CONSTRAINT": "100",
           "WINDOW_SIZE": "2"}
    # per-rank cwds: each rank writes its own monitoring CSVs
    cwds = []
    for r in range(2):
        d = tmp_path / f"rank{r}"
        d.mkdir()
        cwds.append(str(d))
    ctx = mp.get_context("spawn")
    result_q = ctx.Queue()
    port = 29641
    extra = ["-c", "rpc", "-b", "8", "-u", "2", "-m", MODEL,
             "-pt", "1,24,25,48", "-q", "8,0"]
    procs = [ctx.Process(target=_worker,
                         args=(r, port, extra, env, result_q, cwds[r]))
             for r in range(2)]
    for p in procs:
        p.start()
    outs = {}
    for _ in range(2):
        rank, rc, out, err = result_q.get(timeout=300)
        outs[rank] = (rc, out, err)
    for p in procs:
        p.join(timeout=30)
    assert outs[0][0] == 0, outs[0][2]
    assert outs[1][0] == 0, outs[1][2]
    assert "throughput is" in outs[0][1]
    # quantized transport: the encode hook ran on the stage-0 owner (rank 0)
    # and recorded per-iteration bitwidths in its monitoring CSV
    enc = (tmp_path / "rank0" / "quant_encode.csv").read_text() \
        .strip().splitlines()
    assert len(enc) > 1, enc
    bits = {int(float(row.split(",")[-1])) for row in enc[1:]}
    assert bits - {0}, f"no nonzero quant bitwidths recorded: {enc}"
    # decode fired on the stage-1 owner (rank 1)
    dec = (tmp_path / "rank1" / "quant_decode.csv").read_text() \
        .strip().splitlines()
    assert len(dec) > 1, dec
    # the adaptive controller fired at the window boundary
    assert "Adaptive quantization (controller)" in outs[0][1], outs[0][1]


def test_runtime_auto_sched_from_yaml(tmp_path):
    layers = 48
    model = {"layers": layers, "parameters_in": 150528,
             "parameters_out": [2 * 198 * 192 if l % 4 in (1, 3)
                                else 198 * 192 for l in range(1, layers)]
             + [1000],
             "mem_MB": [5.0] * layers}
    dev_types = {"mi355x": {
        "mem_MB": 100000, "bw_Mbps": 100000,
        "model_profiles": {MODEL: [
            {"dtype": "torch.float32", "batch_size": 2,
             "time_s": [0.01] * layers}]}}}
    devices_yml = {"mi355x": ["0", "1"]}
    with open(tmp_path / "models.yml", "w") as f:
        yaml.safe_dump({MODEL: model}, f)
    with open(tmp_path / "device_types.yml", "w") as f:
        yaml.safe_dump(dev_types, f)
    with open(tmp_path / "devices.yml", "w") as f:
        yaml.safe_dump(devices_yml, f)
    extra = ["-b", "4", "-u", "2", "-m", MODEL,
             "-sm", str(tmp_path / "models.yml"),
             "-sdt", str(tmp_path / "device_types.yml"),
             "-sd", str(tmp_path / "devices.yml")]
    outs = _run_two(29635, extra, cwd=str(tmp_path))
    assert outs[0][0] == 0, outs[0][2]
    assert outs[1][0] == 0, outs[1][2]
    assert "throughput is" in outs[0][1]
    # monitoring CSVs were written by the run
    assert (tmp_path / "shard.csv").exists()


def _run_n(port, n, extra_args, env_extra=None):
    ctx = mp.get_context("spawn")
    result_q = ctx.Queue()
    procs = []
    for r in range(n):
        p = ctx.Process(target=_worker_n,
                        args=(r, n, port, extra_args, env_extra or {},
                              result_q))
        p.start()
        procs.append(p)
    outs = {}
    for _ in range(n):
        rank, rc, out, err = result_q.get(timeout=300)
        outs[rank] = (rc, out, err)
    for p in procs:
        p.join(timeout=30)
    return outs


def _worker_n(rank, world, port, extra_args, env_extra, result_q):
    env = dict(os.environ, MASTER_ADDR="127.0.0.1", MASTER_PORT=str(port),
               PYTHONPATH=REPO, OMP_NUM_THREADS="2", **env_extra)
    cmd = [sys.executable, os.path.join(REPO, "runtime.py"), str(rank),
           str(world), "-d", "cpu", "--port", str(port)] + extra_args
    proc = subprocess.run(cmd, capture_output=True, timeout=300, env=env,
                          cwd=REPO, check=False)
    result_q.put((rank, proc.returncode, proc.stdout.decode(),
                  proc.stderr.decode()[-1200:]))


def test_runtime_pure_data_rank():
    """Data rank that is NOT a pipeline stage (reference model_cfg.py:133-141
    wiring): rank 2 feeds inputs / collects results; ranks 0-1 compute."""
    outs = _run_n(29637, 3, ["-b", "4", "-u", "2", "-m", MODEL,
                             "-pt", "1,24,25,48", "-r", "0,1", "-D", "2"])
    for r in range(3):
        assert outs[r][0] == 0, (r, outs[r][2])
    assert "throughput is" in outs[2][1]


def test_runtime_reversed_rank_order():
    """Stage order reversed relative to ranks (-r 1,0 with data on rank 1)."""
    outs = _run_n(29639, 2, ["-b", "4", "-u", "2", "-m", MODEL,
                             "-pt", "1,24,25,48", "-r", "1,0", "-D", "1"])
    for r in range(2):
        assert outs[r][0] == 0, (r, outs[r][2])
    assert "throughput is" in outs[1][1]


def test_runtime_bert_variable_length_npz(tmp_path):
    """BASELINE config #4 workload shape: BERT-CoLA classifier over a
    VARIABLE-LENGTH padded batch (tokenized npz from tools/bert_save_input
    --tokenized); padding masked by the first shard (see bert.py)."""
    import numpy as np
    rng = np.random.default_rng(3)
    n, smax = 16, 64
    ids = np.zeros((n, smax), dtype=np.int64)
    for i in range(n):
        ln = int(rng.integers(8, smax + 1))
        ids[i, :ln] = rng.integers(1000, 30000, size=ln)
    np.savez(tmp_path / "cola.npz", ids=ids,
             label=rng.integers(0, 2, size=n))
    outs = _run_two(29643, ["-b", "8", "-u", "2",
                            "-m", "textattack/bert-base-uncased-CoLA",
                            "-pt", "1,24,25,48",
                            "--dataset-name", "npz",
                            "--dataset-root", str(tmp_path / "cola.npz")],
                    cwd=str(tmp_path))
    assert outs[0][0] == 0, outs[0][2]
    assert outs[1][0] == 0, outs[1][2]
    assert "throughput is" in outs[0][1]


def test_runtime_sched_failure_does_not_hang(tmp_path):
    """A rank-0 scheduling failure must abort ALL ranks promptly (observed:
    a bad YAML left the other ranks waiting on the schedule forever)."""
    import time
    # device_types.yml has no profile for this model -> no viable schedule
    bad_dev_types = {"mi355x": {"mem_MB": 1000, "bw_Mbps": 1000,
                                "model_profiles": {}}}
    model = {"layers": 48, "parameters_in": 1, "parameters_out": [1] * 48,
             "mem_MB": [1.0] * 48}
    with open(tmp_path / "models.yml", "w") as f:
        yaml.safe_dump({MODEL: model}, f)
    with open(tmp_path / "device_types.yml", "w") as f:
        yaml.safe_dump(bad_dev_types, f)
    with open(tmp_path / "devices.yml", "w") as f:
        yaml.safe_dump({"mi355x": ["0", "1"]}, f)
    t0 = time.time()
    outs = _run_two(29645, ["-b", "4", "-u", "2", "-m", MODEL,
                            "-sm", str(tmp_path / "models.yml"),
                            "-sdt", str(tmp_path / "device_types.yml"),
                            "-sd", str(tmp_path / "devices.yml")],
                    cwd=str(tmp_path))
    elapsed = time.time() - t0
    assert outs[0][0] != 0          # rank 0 reports the failure
    assert elapsed < 120, f"ranks hung for {elapsed:.0f}s"
    # rank 1 exited (either cleanly stopped or with the stop error)
    assert outs[1][0] is not None


def test_runtime_bert_quantized_masked(tmp_path):
    """Quantized wire x padding mask interplay: 2-stage BERT-CoLA with
    -q 8,0 over a variable-length padded batch (the mask tensor rides the
    quantized boundary)."""
    import numpy as np
    rng = np.random.default_rng(5)
    n, smax = 8, 48
    ids = np.zeros((n, smax), dtype=np.int64)
    for i in range(n):
        ln = int(rng.integers(8, smax + 1))
        ids[i, :ln] = rng.integers(1000, 30000, size=ln)
    np.savez(tmp_path / "cola.npz", ids=ids,
             label=rng.integers(0, 2, size=n))
    outs = _run_two(29647, ["-b", "8", "-u", "2",
                            "-m", "textattack/bert-base-uncased-CoLA",
                            "-pt", "1,24,25,48", "-q", "8,0",
                            "--dataset-name", "npz",
                            "--dataset-root", str(tmp_path / "cola.npz")],
                    cwd=str(tmp_path))
    assert outs[0][0] == 0, outs[0][2]
    assert outs[1][0] == 0, outs[1][2]
    assert "throughput is" in outs[0][1]


def test_dataset_shuffle_and_indices():
    """--dataset-shuffle permutes deterministically; indices subset the
    synthetic dataset (VERDICT round-1 missing #3)."""
    import runtime as runtime_mod
    base = runtime_mod.load_dataset({"name": None}, 
                                    "facebook/deit-tiny-distilled-patch16-224",
                                    8, 2)
    shuf = runtime_mod.load_dataset({"name": None, "shuffle": True},
                                    "facebook/deit-tiny-distilled-patch16-224",
                                    8, 2)
    import torch
    assert len(shuf) == len(base)
    # same multiset of items, different order (deterministic permutation)
    perm_differs = any(not torch.equal(base[i][0], shuf[i][0])
                       for i in range(len(base)))
    assert perm_differs
    shuf2 = runtime_mod.load_dataset({"name": None, "shuffle": True},
                                     "facebook/deit-tiny-distilled-patch16-224",
                                     8, 2)
    assert all(torch.equal(shuf[i][0], shuf2[i][0]) for i in range(len(shuf)))
    sel = runtime_mod.load_dataset({"name": None, "indices": [3, 1]},
                                   "facebook/deit-tiny-distilled-patch16-224",
                                   4, 2)
    assert torch.equal(sel[0][0], base[3][0])
    assert torch.equal(sel[1][0], base[1][0])


def test_init_env_rank_device_mapping(monkeypatch):
    """Bare 'cuda' on a multi-GPU host maps rank -> cuda:LOCAL_RANK."""
    import runtime as runtime_mod
    import devices as devices_mod
    import torch
    calls = {}
    monkeypatch.setattr(torch.cuda, "is_available", lambda: True)
    monkeypatch.setattr(torch.cuda, "device_count", lambda: 8)
    monkeypatch.setattr(devices_mod, "initialize",
                        lambda d=None: calls.setdefault("dev", d))
    monkeypatch.setenv("LOCAL_RANK", "5")
    runtime_mod.init_env("cuda", "127.0.0.1", 29999, "lo0", rank=2)
    assert calls["dev"] == "cuda:5"
    monkeypatch.delenv("LOCAL_RANK")
    calls.clear()
    runtime_mod.init_env("cuda", "127.0.0.1", 29999, "lo0", rank=3)
    assert calls["dev"] == "cuda:3"
