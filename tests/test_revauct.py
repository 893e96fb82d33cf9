"""Reverse-auction scheduler tests (sched/revauct.py + the revauct.py app)."""
import multiprocessing as mp
import os
import subprocess
import sys

import pytest
import yaml

from pipeedge_amd.sched import revauct

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def _model(layers=8):
    return {"layers": layers, "parameters_in": 10,
            "parameters_out": [10] * layers, "mem_MB": [1] * layers}


def _dev_type(mem=10 ** 6, bw=1000):
    return {"mem_MB": mem, "bw_Mbps": bw, "model_profiles": {}}


def _profile(layers=8, t=0.1):
    return {"dtype": "torch.float32", "batch_size": 1,
            "time_s": [t] * layers}


def test_bid_latency_all_feasible():
    bids = revauct.bid_latency(_model(4), _dev_type(), _profile(4), 1)
    # 4+3+2+1 contiguous shards
    assert len(bids) == 10
    assert (((0, 3), pytest.approx(0.4)) in
            [(s, pytest.approx(c)) for s, c in bids])


def test_bid_latency_memory_gate():
    bids = revauct.bid_latency(_model(4), _dev_type(mem=2.5), _profile(4), 1)
    shards = [s for s, _ in bids]
    assert (0, 3) not in shards  # 4 MB of weights > 2.5 MB
    assert (0, 0) in shards


def test_filter_bids_chunk():
    bids = {s: c for s, c in revauct.bid_latency(_model(8), _dev_type(),
                                                 _profile(8), 1)}
    filt = revauct.filter_bids_chunk(_model(8), bids, chunk=4)
    assert set(filt) == {(0, 3), (0, 7), (4, 7)}


def test_filter_bids_largest():
    bids = {(0, 1): 0.2, (0, 3): 0.4, (2, 3): 0.2, (2, 5): 0.4}
    filt = revauct.filter_bids_largest(bids)
    assert filt == {(0, 3): 0.4, (2, 5): 0.4}


def _bids_for(hosts, layers=8, neighbors_bw=1000):
    model = _model(layers)
    out = {}
    for h in hosts:
        shard_bids = {s: c for s, c in revauct.bid_latency(
            model, _dev_type(), _profile(layers), 1)}
        nbrs = {o: {"bw_Mbps": neighbors_bw} for o in hosts if o != h}
        out[h] = (shard_bids, nbrs)
    return out


def test_greedy_host_count_single_host_takes_all():
    model = _model(8)
    bids = _bids_for(["a", "b"])
    sched = revauct.sched_greedy_host_count(model, 1, "torch.float32", bids,
                                            "a", "a")
    assert sched[0] == {"a": [0, 7]}


def test_optimal_throughput_balances():
    model = _model(8)
    bids = _bids_for(["a", "b"])
    # limit each host to at most 4 layers via memory
    for h in bids:
        bids[h] = ({s: c for s, c in bids[h][0] .items()
                    if s[1] - s[0] < 4}, bids[h][1])
    sched, thr = revauct.sched_optimal_throughput_dev_order(
        model, 1, "torch.float32", bids, "a", "a", ["a", "b"])
    layers = [list(s.values())[0] for s in sched if list(s.values())[0]]
    assert layers[0][0] == 0 and layers[-1][1] == 7
    assert thr > 0


def test_optimal_latency_prefers_one_stage():
    model = _model(8)
    bids = _bids_for(["a", "b"], neighbors_bw=0.001)  # comm very costly
    sched, cost = revauct.sched_optimal_latency_dev_order(
        model, 1, "torch.float32", bids, "a", "a", ["a", "b"],
        strict_order=False, strict_first=False, strict_last=False)
    stages = [s for s in sched if list(s.values())[0]]
    assert len(stages) == 1
    assert cost == pytest.approx(0.8)


def test_no_path_returns_empty():
    model = _model(8)
    bids = _bids_for(["a", "b"])
    # remove all head shards so no path exists
    for h in bids:
        bids[h] = ({s: c for s, c in bids[h][0].items() if s[0] != 0},
                   bids[h][1])
    sched, cost = revauct.sched_optimal_latency_dev_order(
        model, 1, "torch.float32", bids, "a", "a", ["a", "b"])
    assert sched == []
    assert cost == float("inf")


def _revauct_worker(rank, port, tmpdir, result_q):
    env = dict(os.environ, MASTER_ADDR="127.0.0.1", MASTER_PORT=str(port),
               PYTHONPATH=REPO)
    host = f"h{rank}"
    cmd = [sys.executable, os.path.join(REPO, "revauct.py"), str(rank), "2",
           "--host", host, "--dev-type", "mi355x",
           "-m", "google/vit-base-patch16-224", "-u", "1",
           "-sm", os.path.join(tmpdir, "models.yml"),
           "-sdt", os.path.join(tmpdir, "device_types.yml"),
           "-sdnw", os.path.join(tmpdir, "neighbors.yml"),
           "--port", str(port), "--seed", "0"]
    proc = subprocess.run(cmd, capture_output=True, timeout=240, check=False,
                          env=env, cwd=REPO)
    result_q.put((rank, proc.returncode, proc.stdout.decode(),
                  proc.stderr.decode()[-800:]))


def test_revauct_app_two_ranks(tmp_path):
    layers = 48
    model = {"layers": layers, "parameters_in": 150528,
             "parameters_out": [151296] * (layers - 1) + [1000],
             "mem_MB": [25.0] * layers}
    dev_types = {"mi355x": {
        "mem_MB": 1000, "bw_Mbps": 100000,
        "model_profiles": {"google/vit-base-patch16-224": [
            {"dtype": "torch.float32", "batch_size": 1,
             "time_s": [0.001] * layers}]}}}
    neighbors = {"h0": {"h1": {"bw_Mbps": 100000}},
                 "h1": {"h0": {"bw_Mbps": 100000}}}
    with open(tmp_path / "models.yml", "w") as f:
        yaml.safe_dump({"google/vit-base-patch16-224": model}, f)
    with open(tmp_path / "device_types.yml", "w") as f:
        yaml.safe_dump(dev_types, f)
    with open(tmp_path / "neighbors.yml", "w") as f:
        yaml.safe_dump(neighbors, f)

    ctx = mp.get_context("spawn")
    result_q = ctx.Queue()
    procs = [ctx.Process(target=_revauct_worker,
                         args=(r, 29651, str(tmp_path), result_q))
             for r in range(2)]
    for p in procs:
        p.start()
    outs = {}
    for _ in range(2):
        rank, rc, out, err = result_q.get(timeout=240)
        outs[rank] = (rc, out, err)
    for p in procs:
        p.join(timeout=30)
    assert outs[0][0] == 0, outs[0][2]
    assert outs[1][0] == 0, outs[1][2]
    # rank 0 printed a schedule covering layers 1..48 (stdout only)
    stdout = outs[0][1]
    sched = yaml.safe_load(stdout[stdout.index("- h"):])
    layers_covered = [list(s.values())[0] for s in sched
                      if list(s.values())[0]]
    assert layers_covered[0][0] == 1
    assert layers_covered[-1][1] == 48
