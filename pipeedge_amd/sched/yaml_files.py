"""YAML file load/save helpers (parity: reference sched/yaml_files.py)."""
from typing import Optional

import yaml


def yaml_load(file_path: str) -> dict:
    with open(file_path, encoding="utf-8") as f:
        return yaml.safe_load(f)


def yaml_save(data, file_path: str) -> None:
    with open(file_path, "w", encoding="utf-8") as f:
        yaml.safe_dump(data, f, default_flow_style=None)


def load_models_file(file_path: str, model_name: Optional[str] = None):
    """Load models.yml; optionally select one model entry."""
    data = yaml_load(file_path)
    if model_name is None:
        return data
    return data[model_name]


def load_device_types_file(file_path: str) -> dict:
    return yaml_load(file_path)


def load_devices_file(file_path: str) -> dict:
    """devices.yml: map of device type name -> list of host names."""
    return yaml_load(file_path)


# reference-compatible loader aliases (reference sched/yaml_files.py:15-49)
def yaml_models_load(file_path: str) -> dict:
    """Load a models.yml file."""
    return yaml_load(file_path) or {}


def yaml_device_types_load(file_path: str) -> dict:
    """Load a device_types.yml file."""
    return yaml_load(file_path) or {}


def yaml_device_neighbors_world_load(file_path: str) -> dict:
    """Load a device neighbors world file: {host: {neighbor: {bw_Mbps}}}."""
    return yaml_load(file_path) or {}
