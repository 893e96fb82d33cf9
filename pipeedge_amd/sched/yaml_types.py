"""YAML schema constructors/validators (parity: reference sched/yaml_types.py).

Schemas are unchanged from the reference (README_Scheduler.md:44-264) so
existing PipeEdge YAML files work as-is; ``bw_Mbps`` now typically carries
xGMI link bandwidth and ``mem_MB`` HBM capacity.
"""
from typing import List, Optional, Union


def _assert_list_type(lst, dtype):
    assert isinstance(lst, list)
    for var in lst:
        assert isinstance(var, dtype)


def yaml_model(num_layers: int, parameters_in: int, parameters_out: List[int],
               mem_MB: Union[List[int], List[float]]) -> dict:
    """Create a YAML model entry."""
    assert isinstance(num_layers, int)
    assert isinstance(parameters_in, int)
    _assert_list_type(parameters_out, int)
    _assert_list_type(mem_MB, (int, float))
    assert len(parameters_out) == num_layers
    assert len(mem_MB) == num_layers
    return {
        "layers": num_layers,
        "parameters_in": parameters_in,
        "parameters_out": parameters_out,
        "mem_MB": mem_MB,
    }


def yaml_model_profile(dtype: str, batch_size: int,
                       time_s: Union[List[int], List[float]]) -> dict:
    """Create a YAML model profile entry."""
    assert isinstance(dtype, str)
    assert isinstance(batch_size, int)
    _assert_list_type(time_s, (int, float))
    return {
        "dtype": dtype,
        "batch_size": batch_size,
        "time_s": time_s,
    }


def yaml_device_type(mem_MB: Union[int, float], bw_Mbps: Union[int, float],
                     model_profiles: Optional[dict]) -> dict:
    """Create a YAML device type entry."""
    assert isinstance(mem_MB, (int, float))
    assert isinstance(bw_Mbps, (int, float))
    if model_profiles is None:
        model_profiles = {}
    assert isinstance(model_profiles, dict)
    return {
        "mem_MB": mem_MB,
        "bw_Mbps": bw_Mbps,
        "model_profiles": model_profiles,
    }


def yaml_device_neighbors_type(bw_Mbps: Union[int, float]) -> dict:
    """Create a YAML device-neighbors bandwidth entry."""
    assert isinstance(bw_Mbps, (int, float))
    return {"bw_Mbps": bw_Mbps}


def yaml_device_neighbors(neighbors: List[str],
                          bws_Mbps: Union[List[int], List[float]]) -> dict:
    """Map neighbor device names to link properties."""
    _assert_list_type(neighbors, str)
    _assert_list_type(bws_Mbps, (int, float))
    return {neighbor: yaml_device_neighbors_type(bw)
            for neighbor, bw in zip(neighbors, bws_Mbps)}
