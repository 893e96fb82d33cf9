"""Monitoring tests: heartbeat accounting across scopes, CSV output, and the
thread-safe global wrapper (reference parity: instant/window/global getters,
per-thread iteration contexts)."""
import os
import threading
import time

from pipeedge_amd.monitoring import MonitorContext


def test_heartbeat_scopes(tmp_path):
    os.chdir(tmp_path)
    ctx = MonitorContext("k", window_size=3, log_name=str(tmp_path / "k.csv"),
                         use_energy=False)
    ctx.open()
    for i in range(6):
        ctx.iteration_start("k")
        time.sleep(0.002)
        ctx.iteration("k", work=2, accuracy=1)
    assert ctx.get_count("k") == 6
    assert ctx.get_global_work("k") == 12
    assert ctx.get_window_work("k") == 6  # last 3
    assert ctx.get_instant_work("k") == 2
    assert ctx.get_global_perf("k") > 0
    assert ctx.get_window_heart_rate("k") > 0
    assert ctx.get_window_heartrate("k") == ctx.get_window_heart_rate("k")
    assert ctx.get_global_accuracy("k") == 6
    assert ctx.get_tag("k") == 6
    ctx.close()
    lines = open(tmp_path / "k.csv").read().strip().splitlines()
    assert len(lines) == 7  # header + 6 records


def test_monitor_context_multiple_keys(tmp_path):
    ctx = MonitorContext("a", window_size=2, log_name=None, use_energy=False)
    ctx.add_heartbeat("b")
    ctx.iteration("a", work=1)
    ctx.iteration("b", work=5)
    assert set(ctx.keys()) == {"a", "b"}
    assert ctx.get_global_work("b") == 5


def test_global_wrapper_threads(tmp_path, monkeypatch):
    monkeypatch.chdir(tmp_path)
    import importlib
    import monitoring
    importlib.reload(monitoring)
    monitoring.init("shard", 4, work_type="tensors")
    monitoring.add_key("send", work_type="Mbits")

    def worker():
        for _ in range(5):
            monitoring.iteration_start("send")
            monitoring.iteration("send", work=1.5)

    threads = [threading.Thread(target=worker) for _ in range(4)]
    for t in threads:
        t.start()
    for t in threads:
        t.join()
    with monitoring.get_locked_context("send") as mctx:
        assert mctx.get_count("send") == 20
        assert abs(mctx.get_global_work("send") - 30.0) < 1e-9
    monitoring.finish()
    assert (tmp_path / "send.csv").exists()


def test_energy_source_graceful():
    from pipeedge_amd.monitoring import EnergySource
    src = EnergySource()
    assert src.name in ("hwmon", "none")
    assert src.read_uj() >= 0
