"""Profiler -> YAML converters -> scheduler toolchain tests (CPU)."""
import subprocess
import sys
import os

import pytest
import torch
import yaml

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, REPO)


def test_profile_module_shard_inprocess():
    from profiler import profile_module_shard, get_shapes
    inputs = torch.randn(2, 3, 224, 224)
    module_cfg = {"device": "cpu",
                  "name": "facebook/deit-tiny-distilled-patch16-224",
                  "file": "/nonexistent.npz"}
    res = profile_module_shard(module_cfg, inputs, 1, 1, warmup=0,
                               iterations=1)
    assert res["layer"] == 1
    assert res["time"] > 0
    assert res["shape_in"] == [[3, 224, 224]]
    assert res["shape_out"] == [[198, 192], [198, 192]]
    out = res["_outputs"]
    # chain into layer 2
    res2 = profile_module_shard(module_cfg, out, 2, 3, warmup=0, iterations=1)
    assert res2["shape_in"] == [[198, 192], [198, 192]]


def _fake_profile(layers=8, model="facebook/deit-tiny-distilled-patch16-224"):
    data = []
    shape = [[3, 224, 224]]
    for l in range(1, layers + 1):
        out_shape = [[198, 192], [198, 192]] if l % 4 in (1, 3) \
            else [[198, 192]]
        data.append({"layer": l, "memory": 2.0 + l * 0.1,
                     "shape_in": shape, "shape_out": out_shape,
                     "time": 0.01 * (1 + l % 3)})
        shape = out_shape
    return {"model_name": model, "dtype": "torch.float32", "batch_size": 2,
            "layers": layers, "profile_data": data}


def test_converters_and_scheduler_end_to_end(tmp_path):
    """profiler_results.yml -> models.yml + device_types.yml ->
    sched-pipeline (python DP) -> valid partition."""
    results_yml = tmp_path / "profiler_results.yml"
    models_yml = tmp_path / "models.yml"
    types_yml = tmp_path / "device_types.yml"
    devs_yml = tmp_path / "devices.yml"
    prof = _fake_profile()
    with open(results_yml, "w") as f:
        yaml.safe_dump(prof, f)

    env = dict(os.environ, PYTHONPATH=REPO)
    subprocess.run([sys.executable,
                    os.path.join(REPO, "profiler_results_to_models.py"),
                    "-r", str(results_yml), "-o", str(models_yml)],
                   check=True, env=env, cwd=REPO)
    subprocess.run([sys.executable,
                    os.path.join(REPO, "profiler_results_to_device_types.py"),
                    "-r", str(results_yml), "-o", str(types_yml),
                    "-t", "mi355x"], check=True, env=env, cwd=REPO)
    with open(devs_yml, "w") as f:
        yaml.safe_dump({"mi355x": ["0", "1"]}, f)

    with open(models_yml) as f:
        models = yaml.safe_load(f)
    entry = models[prof["model_name"]]
    assert entry["layers"] == 8
    assert entry["parameters_in"] == 3 * 224 * 224
    assert entry["parameters_out"][0] == 2 * 198 * 192
    assert entry["parameters_out"][1] == 198 * 192

    with open(types_yml) as f:
        types = yaml.safe_load(f)
    profs = types["mi355x"]["model_profiles"][prof["model_name"]]
    assert profs[0]["batch_size"] == 2
    assert len(profs[0]["time_s"]) == 8

    from pipeedge_amd.sched.scheduler import sched_pipeline_py
    sched = sched_pipeline_py(prof["model_name"], 2, 2, 2,
                              dtype="torch.float32",
                              models_file=str(models_yml),
                              dev_types_file=str(types_yml),
                              dev_file=str(devs_yml))
    assert len(sched) >= 1
    # the partition must cover all layers contiguously
    layers = sorted((list(s.values())[0] for s in sched))
    assert layers[0][0] == 1 and layers[-1][1] == 8
    for (a, b), (c, d) in zip(layers, layers[1:]):
        assert c == b + 1


def test_device_types_merge_overwrite(tmp_path):
    results_yml = tmp_path / "r.yml"
    types_yml = tmp_path / "t.yml"
    with open(results_yml, "w") as f:
        yaml.safe_dump(_fake_profile(), f)
    env = dict(os.environ, PYTHONPATH=REPO)
    cmd = [sys.executable,
           os.path.join(REPO, "profiler_results_to_device_types.py"),
           "-r", str(results_yml), "-o", str(types_yml)]
    subprocess.run(cmd, check=True, env=env, cwd=REPO)
    # same profile again without overwrite: must fail
    rc = subprocess.run(cmd, capture_output=True, env=env, cwd=REPO)
    assert rc.returncode != 0
    subprocess.run(cmd + ["--overwrite-profile"], check=True, env=env,
                   cwd=REPO)


def test_save_model_weights_cli(tmp_path):
    out = tmp_path / "deit-t.npz"
    subprocess.run([sys.executable, os.path.join(REPO, "save_model_weights.py"),
                    "-m", "facebook/deit-tiny-distilled-patch16-224",
                    "-M", str(out)], check=True,
                   env=dict(os.environ, PYTHONPATH=REPO), cwd=REPO)
    assert out.exists()


def test_profiler_cli_subprocess_mode(tmp_path):
    """The real profiler CLI over a few layers - exercises the fresh-
    subprocess measurement path (regression: torch shared-storage tensors
    on the mp queue raced the child's exit -> ConnectionResetError)."""
    env = dict(os.environ, PYTHONPATH=REPO)
    out_yml = tmp_path / "profiler_results.yml"
    subprocess.run(
        [sys.executable, os.path.join(REPO, "profiler.py"),
         "-m", "google/vit-base-patch16-224", "-b", "2", "-i", "1",
         "-w", "0", "-l", "1", "-L", "3", "-o", str(out_yml)],
        check=True, env=env, cwd=str(tmp_path), timeout=600)
    with open(out_yml) as f:
        res = yaml.safe_load(f)
    assert res["layers"] == 48 and len(res["profile_data"]) == 3
    assert res["profile_data"][0]["shape_out"] == [[197, 768], [197, 768]]
    assert all(d["time"] > 0 for d in res["profile_data"])
