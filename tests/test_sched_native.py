"""Golden tests for the native sched-pipeline binary: built on demand, its
YAML output must match the pure-Python DP on the same inputs (the coverage
the reference lacks - it has no C++ tests at all)."""
import os
import subprocess

import pytest
import yaml

from pipeedge_amd.sched import schedule_optimal_throughput
from pipeedge_amd.sched.scheduler import sched_pipeline
from pipeedge_amd.sched.yaml_files import yaml_save

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
BINARY = os.path.join(REPO, "src-native", "sched-pipeline")


@pytest.fixture(scope="module")
def binary():
    rc = subprocess.run(["make", "-C", os.path.join(REPO, "src-native")],
                        capture_output=True, check=False)
    if rc.returncode != 0 or not os.path.isfile(BINARY):
        pytest.skip("sched-pipeline binary unavailable: "
                    + rc.stderr.decode()[-500:])
    return BINARY


def _write_inputs(tmp_path, yml_model, dev_types, dev_hosts, model="m"):
    models_file = str(tmp_path / "models.yml")
    types_file = str(tmp_path / "device_types.yml")
    devs_file = str(tmp_path / "devices.yml")
    yaml_save({model: yml_model}, models_file)
    types_out = {}
    for name, props in dev_types:
        entry = {"mem_MB": props["mem_MB"], "bw_Mbps": props["bw_Mbps"],
                 "model_profiles": {}}
        prof = props.get("model_profiles", {}).get(model)
        if prof:
            entry["model_profiles"] = {model: [prof]}
        types_out[name] = entry
    yaml_save(types_out, types_file)
    yaml_save(dev_hosts, devs_file)
    return models_file, types_file, devs_file


def _run_binary(binary, models_file, types_file, devs_file, batch=1,
                dtype="torch.float32", model="m"):
    proc = subprocess.run(
        [binary, "-m", model, "-b", str(batch), "-d", dtype,
         "-M", models_file, "-T", types_file, "-D", devs_file],
        capture_output=True, check=False)
    if proc.returncode == 2:
        return []
    assert proc.returncode == 0, proc.stderr.decode()
    return yaml.safe_load(proc.stdout.decode())


def _case_uniform(n_layers, n_hosts, bw=1e9, mem=10 ** 6, times=None):
    yml_model = {"layers": n_layers, "parameters_in": 10,
                 "parameters_out": [10] * n_layers,
                 "mem_MB": [1] * n_layers}
    prof = {"dtype": "torch.float32", "batch_size": 1,
            "time_s": times or [1.0] * n_layers}
    dev = ("gpu", {"mem_MB": mem, "bw_Mbps": bw,
                   "model_profiles": {"m": prof}})
    hosts = {"gpu": [f"h{i}" for i in range(n_hosts)]}
    return yml_model, [dev], hosts


@pytest.mark.parametrize("n_layers,n_hosts", [(4, 2), (8, 4), (6, 3), (5, 8)])
def test_native_matches_python_uniform(binary, tmp_path, n_layers, n_hosts):
    yml_model, dev_types, hosts = _case_uniform(n_layers, n_hosts)
    files = _write_inputs(tmp_path, yml_model, dev_types, hosts)
    native = _run_binary(binary, *files)
    py = schedule_optimal_throughput(yml_model, "torch.float32", 1, dev_types,
                                     hosts, model_name="m")
    assert native == py


def test_native_uneven_times(binary, tmp_path):
    # layer 3 is huge: it should get its own stage
    yml_model, dev_types, hosts = _case_uniform(
        6, 3, times=[1, 1, 10, 1, 1, 1])
    files = _write_inputs(tmp_path, yml_model, dev_types, hosts)
    native = _run_binary(binary, *files)
    py = schedule_optimal_throughput(yml_model, "torch.float32", 1, dev_types,
                                     hosts, model_name="m")
    assert native == py
    # the heavy layer's stage should be alone
    heavy = [s for s in native if list(s.values())[0][0] <= 3
             <= list(s.values())[0][1]][0]
    assert list(heavy.values())[0] == [3, 3]


def test_native_low_bandwidth_single_stage(binary, tmp_path):
    yml_model, dev_types, hosts = _case_uniform(4, 2, bw=1e-6)
    files = _write_inputs(tmp_path, yml_model, dev_types, hosts)
    native = _run_binary(binary, *files)
    assert len(native) == 1
    assert list(native[0].values())[0] == [1, 4]


def test_native_heterogeneous(binary, tmp_path):
    yml_model = {"layers": 4, "parameters_in": 1,
                 "parameters_out": [1, 1, 1, 1], "mem_MB": [1, 1, 1, 1]}
    fast = ("fast", {"mem_MB": 10 ** 6, "bw_Mbps": 1e9, "model_profiles":
                     {"m": {"dtype": "torch.float32", "batch_size": 1,
                            "time_s": [0.1] * 4}}})
    slow = ("slow", {"mem_MB": 10 ** 6, "bw_Mbps": 1e9, "model_profiles":
                     {"m": {"dtype": "torch.float32", "batch_size": 1,
                            "time_s": [10.0] * 4}}})
    hosts = {"fast": ["f0"], "slow": ["s0"]}
    files = _write_inputs(tmp_path, yml_model, [fast, slow], hosts)
    native = _run_binary(binary, *files)
    py = schedule_optimal_throughput(yml_model, "torch.float32", 1,
                                     [fast, slow], hosts, model_name="m")
    assert native == py == [{"f0": [1, 4]}]


def test_native_memory_infeasible(binary, tmp_path):
    yml_model, dev_types, hosts = _case_uniform(4, 1, mem=0.5)
    files = _write_inputs(tmp_path, yml_model, dev_types, hosts)
    assert _run_binary(binary, *files) == []


def test_native_bf16_dtype(binary, tmp_path):
    """The reference rejects non-fp32 dtypes (sched-pipeline.cpp:227-232);
    this build supports bf16."""
    yml_model, dev_types, hosts = _case_uniform(4, 2)
    dev_types[0][1]["model_profiles"]["m"]["dtype"] = "torch.bfloat16"
    files = _write_inputs(tmp_path, yml_model, dev_types, hosts)
    native = _run_binary(binary, *files, dtype="torch.bfloat16")
    assert len(native) == 2


def test_sched_pipeline_wrapper_uses_binary(binary, tmp_path):
    yml_model, dev_types, hosts = _case_uniform(4, 2)
    models_file, types_file, devs_file = _write_inputs(
        tmp_path, yml_model, dev_types, hosts)
    out = sched_pipeline("m", 2, 2, 1, models_file=models_file,
                         dev_types_file=types_file, dev_file=devs_file,
                         app_paths=[binary], require_native=True)
    assert out == [{"h0": [1, 2]}, {"h1": [3, 4]}]
