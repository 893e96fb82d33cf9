"""Partition scheduling: analytic cost models + optimal-throughput DP.

Capability parity with reference ``src/pipeedge/sched/__init__.py`` (cost
models; layer indexing here is [0, L-1] while the runtime and the native
scheduler use [1, L] - same convention split as the reference) and
``src-native/schedule.cpp`` (the DP, also re-implemented natively in this
repo's ``src-native/``; the Python version below doubles as the golden
reference for the C++ binary's tests).

MI355X deltas: the dtype table covers bf16/fp16 (the reference was
float32-only, sched-pipeline.cpp:227-232), and device-type YAML files model
GPUs (HBM capacity in ``mem_MB``, per-link xGMI bandwidth in ``bw_Mbps``).
"""
from typing import Dict, List, Optional, Tuple

_DTYPE_BYTES_SIZES = {
    "torch.float32": 4,
    "torch.float16": 2,
    "torch.bfloat16": 2,
    "torch.float64": 8,
}


def _dtype_bytes(dtype: str) -> int:
    return _DTYPE_BYTES_SIZES[dtype]


def ubatch_bytes(n_params: int, ubatch_size: int, dtype: str = "torch.float32") -> int:
    """Bytes required for a microbatch buffer."""
    return n_params * ubatch_size * _dtype_bytes(dtype)


def mem_bytes(yml_model: dict, layer_l: int, layer_r: int, dtype: str,
              ubatch_size: int, data_buffers_in: int = 2,
              data_buffers_out: int = 2) -> int:
    """Estimate memory for a complete stage: weights + in/out data buffers +
    processing copies (reference sched/__init__.py:22-48)."""
    assert 0 <= layer_l <= layer_r < len(yml_model["mem_MB"])
    mem_model = sum(yml_model["mem_MB"][layer_l:layer_r + 1]) * 1024 * 1024
    params_in = yml_model["parameters_in"] if layer_l == 0 else \
        yml_model["parameters_out"][layer_l - 1]
    dat_in = ubatch_bytes(params_in, ubatch_size, dtype=dtype)
    dat_out = ubatch_bytes(yml_model["parameters_out"][layer_r], ubatch_size,
                           dtype=dtype)
    buffers = 0
    if layer_l > 0:
        buffers += dat_in * data_buffers_in
    buffers += dat_out * data_buffers_out
    buffers += dat_in + dat_out
    return mem_model + buffers


def computation_time(yml_model_profile: dict, layer_l: int, layer_r: int) -> float:
    """Seconds to process a layer range (sum of profiled per-layer times)."""
    time_s = yml_model_profile["time_s"]
    assert 0 <= layer_l <= layer_r < len(time_s)
    return sum(time_s[layer_l:layer_r + 1])


def communication_time(yml_device_type: dict, data_bytes: int) -> float:
    return communication_time_bw(yml_device_type["bw_Mbps"], data_bytes)


def communication_time_bw(bw_mbits_sec: float, data_bytes: int) -> float:
    bytes_sec = bw_mbits_sec * 1024 * 1024 / 8
    return data_bytes / bytes_sec


# ---------------------------------------------------------------------------
# Optimal-throughput partitioner (Python reference implementation).
# The production path is the C++ `sched-pipeline` binary (src-native/), which
# implements the same algorithm; this version is the golden test oracle and
# the no-binary fallback.
# ---------------------------------------------------------------------------

def schedule_optimal_throughput(
        yml_model: dict, dtype: str, ubatch_size: int,
        dev_types: List[Tuple[str, dict]],
        dev_type_hosts: Dict[str, List[str]],
        data_buffers_in: int = 2, data_buffers_out: int = 2,
        model_name: Optional[str] = None) -> List[Dict[str, List[int]]]:
    """Minimize the max stage cost (compute or communication time) over all
    feasible pipeline partitions and device-type assignments.

    DP over (last layer i, multiset of used device types S as a mixed-radix
    int, current type u); transitions assign layers [i+1, j] to an instance of
    type u when they fit its memory. Returns the PipeEdge schedule format:
    ``[{host: [layer_l, layer_r]}, ...]`` with 1-indexed inclusive layers.
    """
    num_layers = yml_model["layers"]
    type_names = [name for name, _ in dev_types]
    counts = [len(dev_type_hosts.get(name, [])) for name in type_names]
    n_types = len(type_names)

    def profile(ti: int) -> dict:
        props = dev_types[ti][1]
        profs = props.get("model_profiles", {})
        prof = profs.get(model_name) if model_name else None
        if prof is None:
            raise ValueError(
                f"device type {type_names[ti]} has no profile for {model_name}")
        return prof

    # precompute per-type per-range compute time and memory feasibility
    comp = [[[0.0] * (num_layers + 1) for _ in range(num_layers + 1)]
            for _ in range(n_types)]
    fits = [[[False] * (num_layers + 1) for _ in range(num_layers + 1)]
            for _ in range(n_types)]
    for ti in range(n_types):
        if counts[ti] == 0:
            continue
        prof = profile(ti)
        cap = dev_types[ti][1]["mem_MB"] * 1024 * 1024
        for l in range(1, num_layers + 1):
            for r in range(l, num_layers + 1):
                comp[ti][l][r] = computation_time(prof, l - 1, r - 1)
                need = mem_bytes(yml_model, l - 1, r - 1, dtype, ubatch_size,
                                 data_buffers_in, data_buffers_out)
                fits[ti][l][r] = cap > need

    prefix = [1] * (n_types + 1)
    for i in range(n_types):
        prefix[i + 1] = prefix[i] * (counts[i] + 1)
    mask = prefix[n_types]

    INF = float("inf")
    # h[i][S][u]: best max-stage-cost covering layers [1, i], used multiset S,
    # last stage on an instance of type u
    h = [[[INF] * n_types for _ in range(mask)] for _ in range(num_layers + 1)]
    parent = [[[None] * n_types for _ in range(mask)]
              for _ in range(num_layers + 1)]
    for u in range(n_types):
        h[0][0][u] = 0.0

    def used(S, t):
        return (S // prefix[t]) % (counts[t] + 1)

    def comm_t(j, u, v):
        dat = ubatch_bytes(yml_model["parameters_out"][j - 1], ubatch_size, dtype)
        bw = min(dev_types[u][1]["bw_Mbps"], dev_types[v][1]["bw_Mbps"])
        return communication_time_bw(bw, dat)

    best = INF
    best_idx = None
    for i in range(num_layers):
        for S in range(mask):
            for u in range(n_types):
                if h[i][S][u] == INF or used(S, u) == counts[u]:
                    continue
                for j in range(i + 1, num_layers + 1):
                    if not fits[u][i + 1][j]:
                        continue
                    ct = comp[u][i + 1][j]
                    if j == num_layers:
                        cost = max(h[i][S][u], ct)
                        if cost < best:
                            best = cost
                            best_idx = (i, S, u)
                    for v in range(n_types):
                        if used(S, v) == counts[v]:
                            continue
                        cost = max(h[i][S][u], ct, comm_t(j, u, v))
                        S2 = S + prefix[u]
                        if cost < h[j][S2][v]:
                            h[j][S2][v] = cost
                            parent[j][S2][v] = (i, u)
    if best_idx is None:
        return []

    # reconstruct (type_idx, layer_l, layer_r) stages
    i, S, u = best_idx
    stages = [(u, i + 1, num_layers)]
    layer = i
    while layer > 0:
        last_l, last_u = parent[layer][S][u]
        stages.append((last_u, last_l + 1, layer))
        layer, S, u = last_l, S - prefix[last_u], last_u
    stages.sort(key=lambda s: s[1])

    # assign concrete hosts round-robin per type (schedule.cpp:248-267)
    next_host = {name: 0 for name in type_names}
    sched = []
    for ti, l, r in stages:
        name = type_names[ti]
        host = dev_type_hosts[name][next_host[name]]
        next_host[name] += 1
        sched.append({host: [l, r]})
    return sched
