"""P2P comm tests: context lifecycle (single-rank, like the reference's
test/comm/p2p/test_context.py) and a real 2-process gloo pipeline exercising
the full engine (schedule broadcast, batch announcement, static-schema
transfers, results collection) - coverage the reference lacks."""
import multiprocessing as mp
import os
import queue
import threading

import pytest
import torch

from pipeedge_amd.comm import CMD_STOP, TensorSpec


def _set_master(port):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)


def test_context_lifecycle_single_rank():
    from pipeedge_amd.comm.p2p import DistP2pContext
    _set_master(29611)
    ctx = DistP2pContext(1, 0, lambda c, t: None, backend="gloo")
    ctx.init()
    assert ctx.rank == 0 and ctx.world_size == 1
    ctx.shutdown()


def test_context_manager_single_rank():
    from pipeedge_amd.comm.p2p import DistP2pContext
    _set_master(29612)
    with DistP2pContext(1, 0, lambda c, t: None, backend="gloo") as ctx:
        assert ctx.rank == 0


def _pipeline_worker(rank, world_size, port, result_q):
    import logging
    logging.disable(logging.WARNING)
    _set_master(port)
    from pipeedge_amd.comm import CMD_BATCH
    from pipeedge_amd.comm.p2p import DistP2pContext, DistP2pPipelineStage

    batch_q = queue.Queue()
    got_cmds = []
    stop_event = threading.Event()

    def handle_cmd(cmd, tensors):
        got_cmds.append(cmd)
        if cmd == CMD_BATCH:
            batch_q.put(int(tensors[0].item()))
        elif cmd == CMD_STOP:
            stop_event.set()

    spec_in = [TensorSpec((2, 4), torch.float32)]
    spec_mid = [TensorSpec((2, 4), torch.float32),
                TensorSpec((2, 3), torch.float32)]
    spec_out = [TensorSpec((2, 4), torch.float32)]

    with DistP2pContext(world_size, rank, handle_cmd, backend="gloo") as ctx:
        if rank == 0:
            results = []
            # data rank AND stage 0: work doubles, sends (y, aux) tuple
            def work0(x):
                return (x * 2, torch.ones(2, 3))
            stage = DistP2pPipelineStage(1, 1, spec_out, spec_mid, work0,
                                         results.append, device="cpu")
            ubatches = [torch.full((2, 4), float(i)) for i in range(5)]
            ctx.cmd_broadcast(CMD_BATCH, (torch.tensor(len(ubatches)),))
            stage.data_run_batch(ubatches)
            ctx.cmd_broadcast(CMD_STOP)
            stop_event.set()
            ok = len(results) == 5 and all(
                torch.allclose(results[i], torch.full((2, 4), i * 2 + 1.0))
                for i in range(5))
            result_q.put(("rank0", ok))
        else:
            # stage 1: adds the aux tensor's first column value (1.0)
            def work1(data):
                y, aux = data
                return y + aux[:, :1].expand(2, 4) * torch.ones(2, 4)
            stage = DistP2pPipelineStage(0, 0, spec_mid, spec_out, work1,
                                         None, device="cpu")
            n = batch_q.get(timeout=30)
            stage.stage_run_batch(n)
            stop_event.wait(timeout=30)
            result_q.put(("rank1", n == 5))


def test_two_rank_pipeline():
    ctx = mp.get_context("spawn")
    result_q = ctx.Queue()
    procs = [ctx.Process(target=_pipeline_worker, args=(r, 2, 29613, result_q))
             for r in range(2)]
    for p in procs:
        p.start()
    results = {}
    for _ in range(2):
        k, v = result_q.get(timeout=120)
        results[k] = v
    for p in procs:
        p.join(timeout=60)
        assert p.exitcode == 0
    assert results == {"rank0": True, "rank1": True}


def _runtime_worker(rank, world_size, port, result_q, extra_args):
    import logging
    import subprocess
    import sys
    env = dict(os.environ, MASTER_ADDR="127.0.0.1", MASTER_PORT=str(port))
    repo = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    cmd = [sys.executable, os.path.join(repo, "runtime.py"), str(rank),
           str(world_size), "-d", "cpu"] + extra_args
    proc = subprocess.run(cmd, capture_output=True, timeout=300, env=env,
                          cwd=repo, check=False)
    out = proc.stdout.decode()
    result_q.put((rank, proc.returncode, out))


@pytest.mark.parametrize("extra", [
    ["-b", "4", "-u", "2", "-m", "facebook/deit-tiny-distilled-patch16-224",
     "-pt", "1,20,21,48"],
    ["-b", "4", "-u", "2", "-m", "facebook/deit-tiny-distilled-patch16-224",
     "-pt", "1,22,23,48", "-q", "8,0"],
])
def test_runtime_two_rank_cpu(extra):
    """End-to-end runtime.py over 2 local processes (gloo/CPU)."""
    ctx = mp.get_context("spawn")
    result_q = ctx.Queue()
    port = 29620 + len(extra)
    procs = [ctx.Process(target=_runtime_worker,
                         args=(r, 2, port, result_q, extra))
             for r in range(2)]
    for p in procs:
        p.start()
    outs = {}
    for _ in range(2):
        rank, rc, out = result_q.get(timeout=300)
        outs[rank] = (rc, out)
    for p in procs:
        p.join(timeout=30)
    assert outs[0][0] == 0, outs[0][1][-2000:]
    assert outs[1][0] == 0, outs[1][1][-2000:]
    assert "throughput is" in outs[0][1]


def _deep_stream_worker(rank, world_size, port, result_q):
    """Stream FAR more microbatches than the chain can buffer: regression
    for the results-back-edge deadlock (the cycle data rank -> stages ->
    data rank saturated every send window at ~stages*(8+2) in-flight and
    froze an 8-stage run; data-rank admission control bounds in-flight)."""
    import logging
    logging.disable(logging.WARNING)
    _set_master(port)
    from pipeedge_amd.comm.p2p import DistP2pContext, DistP2pPipelineStage

    spec = [TensorSpec((1, 4), torch.float32)]
    n_ub = 300   # >> capacity of a 4-stage chain (~40)
    with DistP2pContext(world_size, rank, lambda c, t: None,
                        backend="gloo"):
        last = world_size - 1
        if rank == 0:
            results = []
            stage = DistP2pPipelineStage(last, 1, spec, spec,
                                         lambda x: x + 1.0, results.append,
                                         device="cpu")
            stage.pipeline_depth = world_size
            ubatches = [torch.full((1, 4), float(i)) for i in range(n_ub)]
            stage.data_run_batch(ubatches)
            ok = len(results) == n_ub and all(
                torch.allclose(results[i],
                               torch.full((1, 4), i + float(world_size)))
                for i in range(0, n_ub, 37))
            result_q.put(("rank0", ok))
        else:
            nxt = 0 if rank == last else rank + 1
            stage = DistP2pPipelineStage(rank - 1, nxt, spec, spec,
                                         lambda x: x + 1.0, None,
                                         device="cpu")
            stage.stage_run_batch(n_ub)
            result_q.put((f"rank{rank}", True))


def test_deep_stream_no_deadlock():
    ctx = mp.get_context("spawn")
    result_q = ctx.Queue()
    world = 4
    procs = [ctx.Process(target=_deep_stream_worker,
                         args=(r, world, 29617, result_q))
             for r in range(world)]
    for p in procs:
        p.start()
    results = {}
    for _ in range(world):
        k, v = result_q.get(timeout=180)
        results[k] = v
    for p in procs:
        p.join(timeout=60)
        assert p.exitcode == 0
    assert all(results.values()), results
