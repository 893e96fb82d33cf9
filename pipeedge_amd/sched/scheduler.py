"""`sched-pipeline` driver (parity: reference sched/scheduler.py).

Runs the native C++ ``sched-pipeline`` binary (this repo's ``src-native/``)
with the same CLI flags as the reference (sched-pipeline.cpp:132-172) and
parses its YAML stdout. Falls back to the in-process Python DP
(`pipeedge_amd.sched.schedule_optimal_throughput`) when the binary is absent
unless ``require_native=True``.
"""
import logging
import os
import subprocess
from typing import Dict, List, Optional

import yaml

from . import schedule_optimal_throughput
from .yaml_files import yaml_load

logger = logging.getLogger(__name__)

#: default search locations for the native binary (repo-local builds first)
_REPO_ROOT = os.path.dirname(os.path.dirname(os.path.dirname(
    os.path.abspath(__file__))))
DEFAULT_APP_PATHS = [
    os.path.join(_REPO_ROOT, "src-native", "sched-pipeline"),
    os.path.join(_REPO_ROOT, "src-native", "build", "sched-pipeline"),
]


def _run_native(app_path: str, args: List[str]) -> Optional[str]:
    try:
        proc = subprocess.run([app_path] + args, capture_output=True, check=True)
    except FileNotFoundError:
        return None
    except subprocess.CalledProcessError as exc:
        logger.error("sched-pipeline failed rc=%d stderr=%s", exc.returncode,
                     exc.stderr.decode().strip())
        raise
    stderr = proc.stderr.decode().strip()
    if stderr:
        logger.warning(stderr)
    return proc.stdout.decode()


def sched_pipeline(model_name: str, buffers_in: int, buffers_out: int,
                   batch_size: int, dtype: str = "torch.float32",
                   models_file: Optional[str] = None,
                   dev_types_file: Optional[str] = None,
                   dev_file: Optional[str] = None,
                   app_paths: Optional[List[str]] = None,
                   require_native: bool = False) -> List[Dict[str, List[int]]]:
    """Schedule the pipeline; returns ``[{host: [layer_l, layer_r]}, ...]``
    (1-indexed inclusive layers)."""
    args = ["-i", str(buffers_in), "-o", str(buffers_out),
            "-b", str(batch_size), "-d", dtype, "-m", model_name]
    if models_file:
        args += ["-M", models_file]
    if dev_types_file:
        args += ["-T", dev_types_file]
    if dev_file:
        args += ["-D", dev_file]

    paths = list(app_paths) if app_paths else []
    paths += DEFAULT_APP_PATHS + ["sched-pipeline"]
    for app_path in paths:
        out = _run_native(app_path, args)
        if out is not None:
            sched = yaml.safe_load(out)
            assert isinstance(sched, list)
            return sched

    if require_native:
        raise FileNotFoundError("sched-pipeline binary not found in "
                                + ", ".join(paths))
    logger.warning("sched-pipeline binary not found; using Python DP fallback")
    return sched_pipeline_py(model_name, buffers_in, buffers_out, batch_size,
                             dtype, models_file, dev_types_file, dev_file)


def sched_pipeline_py(model_name: str, buffers_in: int, buffers_out: int,
                      batch_size: int, dtype: str = "torch.float32",
                      models_file: str = "models.yml",
                      dev_types_file: str = "device_types.yml",
                      dev_file: str = "devices.yml") -> List[Dict[str, List[int]]]:
    """Pure-Python scheduling path (same algorithm as the native binary)."""
    yml_model = yaml_load(models_file)[model_name]
    dev_types_raw = yaml_load(dev_types_file)
    dev_hosts = yaml_load(dev_file)
    dev_types = []
    for name, props in dev_types_raw.items():
        profs = props.get("model_profiles", {})
        prof = None
        for candidate in profs.get(model_name, []):
            if candidate["dtype"] == dtype and candidate["batch_size"] == batch_size:
                prof = candidate
                break
        entry = dict(props)
        entry["model_profiles"] = {model_name: prof} if prof else {}
        dev_types.append((name, entry))
    return schedule_optimal_throughput(
        yml_model, dtype, batch_size, dev_types, dev_hosts,
        data_buffers_in=buffers_in, data_buffers_out=buffers_out,
        model_name=model_name)
