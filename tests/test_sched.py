"""Scheduler tests: YAML type contracts (mirrors reference
test/sched/test_yaml_types.py), cost models, and golden-output tests of the
optimal-throughput DP on hand-computable inputs."""

from pipeedge_amd import sched
from pipeedge_amd.sched import yaml_types


def test_yaml_model():
    ym = yaml_types.yaml_model(2, 10, [20, 30], [1.5, 2.5])
    assert ym["layers"] == 2
    assert ym["parameters_in"] == 10
    assert ym["parameters_out"] == [20, 30]
    assert ym["mem_MB"] == [1.5, 2.5]


def test_yaml_model_profile():
    prof = yaml_types.yaml_model_profile("torch.float32", 8, [0.1, 0.2])
    assert prof["dtype"] == "torch.float32"
    assert prof["batch_size"] == 8
    assert prof["time_s"] == [0.1, 0.2]


def test_yaml_device_type():
    dt = yaml_types.yaml_device_type(1024, 1000, None)
    assert dt["mem_MB"] == 1024
    assert dt["bw_Mbps"] == 1000
    assert dt["model_profiles"] == {}


def test_yaml_device_neighbors():
    dn = yaml_types.yaml_device_neighbors(["a", "b"], [100, 200])
    assert dn == {"a": {"bw_Mbps": 100}, "b": {"bw_Mbps": 200}}


def test_ubatch_bytes():
    assert sched.ubatch_bytes(10, 8, "torch.float32") == 320
    assert sched.ubatch_bytes(10, 8, "torch.bfloat16") == 160


def test_mem_bytes_first_vs_middle():
    ym = {"layers": 3, "parameters_in": 4, "parameters_out": [6, 8, 2],
          "mem_MB": [1.0, 1.0, 1.0]}
    first = sched.mem_bytes(ym, 0, 0, "torch.float32", 1)
    mid = sched.mem_bytes(ym, 1, 1, "torch.float32", 1)
    # first stage has no recv buffers; middle stage includes them
    assert first == 1024 * 1024 + (6 * 4) * 2 + (4 * 4 + 6 * 4)
    assert mid == 1024 * 1024 + (6 * 4) * 2 + (8 * 4) * 2 + (6 * 4 + 8 * 4)


def _simple_setup(bw=1e9, mem=10**6):
    ym = {"layers": 4, "parameters_in": 10,
          "parameters_out": [10, 10, 10, 10],
          "mem_MB": [1, 1, 1, 1]}
    prof = {"dtype": "torch.float32", "batch_size": 1,
            "time_s": [1.0, 1.0, 1.0, 1.0]}
    dev = ("gpu", {"mem_MB": mem, "bw_Mbps": bw,
                   "model_profiles": {"m": prof}})
    return ym, dev


def test_dp_balances_two_devices():
    ym, dev = _simple_setup()
    sched_out = sched.schedule_optimal_throughput(
        ym, "torch.float32", 1, [dev], {"gpu": ["h0", "h1"]}, model_name="m")
    # 4 equal layers over 2 hosts -> even 2+2 split, max stage cost 2.0
    assert len(sched_out) == 2
    layers = [list(s.values())[0] for s in sched_out]
    assert layers == [[1, 2], [3, 4]]


def test_dp_single_device_when_comm_dominates():
    # tiny bandwidth makes any split worse than one device
    ym, dev = _simple_setup(bw=1e-6)
    out = sched.schedule_optimal_throughput(
        ym, "torch.float32", 1, [dev], {"gpu": ["h0", "h1"]}, model_name="m")
    assert len(out) == 1
    assert list(out[0].values())[0] == [1, 4]


def test_dp_memory_gate_forces_split():
    # each device can hold ~2 layers of 1 MB plus buffers, not all 4
    ym, dev = _simple_setup(mem=2.5)
    out = sched.schedule_optimal_throughput(
        ym, "torch.float32", 1, [dev], {"gpu": ["h0", "h1"]}, model_name="m")
    assert len(out) == 2


def test_dp_heterogeneous_prefers_fast_device():
    ym = {"layers": 2, "parameters_in": 1, "parameters_out": [1, 1],
          "mem_MB": [1, 1]}
    fast = ("fast", {"mem_MB": 10**6, "bw_Mbps": 1e9,
                     "model_profiles": {"m": {"dtype": "torch.float32",
                                              "batch_size": 1,
                                              "time_s": [0.1, 0.1]}}})
    slow = ("slow", {"mem_MB": 10**6, "bw_Mbps": 1e9,
                     "model_profiles": {"m": {"dtype": "torch.float32",
                                              "batch_size": 1,
                                              "time_s": [10.0, 10.0]}}})
    out = sched.schedule_optimal_throughput(
        ym, "torch.float32", 1, [fast, slow],
        {"fast": ["f0"], "slow": ["s0"]}, model_name="m")
    # everything on the fast device
    assert out == [{"f0": [1, 2]}]


def test_dp_infeasible_returns_empty():
    ym, dev = _simple_setup(mem=0.5)  # nothing fits
    out = sched.schedule_optimal_throughput(
        ym, "torch.float32", 1, [dev], {"gpu": ["h0"]}, model_name="m")
    assert out == []
