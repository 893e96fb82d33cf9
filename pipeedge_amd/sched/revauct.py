"""Reverse-auction scheduling (FMEC'23 capability; parity: reference
sched/revauct.py).

Devices bid ``((layer_l, layer_r), latency_cost)`` for every memory-feasible
contiguous shard; the auctioneer filters bids and schedules with one of:

- ``sched_greedy_host_count``: minimize host count, largest-shard greedy;
- ``sched_optimal_latency_dev_order``: shortest end-to-end latency path
  through a DAG of (device, shard) nodes respecting a device order;
- ``sched_optimal_throughput_dev_order``: minimize the maximum stage latency
  (compute or comm) over the same DAG - implemented here as a label-setting
  Dijkstra over bottleneck costs (the reference drives networkx's Dijkstra
  with a stateful weight callback, sched/revauct.py:225-251; an explicit
  heap over (bottleneck, node) labels is equivalent and self-contained).

Layer indexing is [0, n_layers-1] like the rest of pipeedge_amd.sched; the
``revauct.py`` app shifts to the runtime's 1-indexed convention on output.
"""
import heapq
import logging
from typing import Dict, List, Mapping, Tuple

from . import communication_time_bw, computation_time, mem_bytes, ubatch_bytes

logger = logging.getLogger(__name__)

ShardBid = Tuple[Tuple[int, int], float]
#: A device's bids: (shard layer pairs -> cost, neighbor comm properties)
DeviceBidData = Tuple[Mapping[Tuple[int, int], float], Mapping[str, dict]]


def bid_latency(yml_model: dict, yml_dev_type: dict, yml_dtm_profile: dict,
                ubatch_size: int, dtype: str = "torch.float32") -> List[ShardBid]:
    """Bid for every memory-feasible shard, cost = profiled compute latency."""
    bids = []
    cap = yml_dev_type["mem_MB"] * 1024 * 1024
    for layer_l in range(yml_model["layers"]):
        for layer_r in range(layer_l, yml_model["layers"]):
            if cap > mem_bytes(yml_model, layer_l, layer_r, dtype, ubatch_size):
                cost = computation_time(yml_dtm_profile, layer_l, layer_r)
                bids.append(((layer_l, layer_r), cost))
    return bids


def filter_bids_chunk(yml_model: dict, bids: Mapping[Tuple[int, int], float],
                      chunk: int = 4) -> Dict[Tuple[int, int], float]:
    """Keep only shards aligned to ``chunk`` boundaries (tail may be short)."""
    n = yml_model["layers"]
    return {shard: cost for shard, cost in bids.items()
            if shard[0] % chunk == 0 and
            (shard[1] + 1 >= n or (shard[1] + 1) % chunk == 0)}


def filter_bids_largest(bids: Mapping[Tuple[int, int], float]) \
        -> Dict[Tuple[int, int], float]:
    """Keep only the largest shard per start layer."""
    best: Dict[int, ShardBid] = {}
    for shard, cost in bids.items():
        cur = best.get(shard[0])
        if cur is None or shard[1] > cur[0][1]:
            best[shard[0]] = (shard, cost)
    return {shard: cost for shard, cost in best.values()}


def sched_greedy_host_count(yml_model: dict, _ubatch_size: int, _dtype: str,
                            bids: Mapping[str, DeviceBidData], host_src: str,
                            host_dest: str) -> List[Mapping[str, List[int]]]:
    """Greedy minimum-host-count schedule (reference revauct.py:53-116):
    source host gets the largest head shard, dest host the largest tail
    shard, remaining layers filled with largest shards (ties by cost)."""
    max_lut: Dict[str, Dict[int, Tuple[int, float]]] = {h: {} for h in bids}
    for host, (shard_bids, _) in bids.items():
        for shard, cost in shard_bids.items():
            cur = max_lut[host].get(shard[0])
            if cur is None or shard[1] > cur[0]:
                max_lut[host][shard[0]] = (shard[1], cost)

    sched: List[Mapping[str, List[int]]] = []
    ins_off = 0
    lay_start = 0
    lay_end = yml_model["layers"] - 1
    used = set()
    if host_src in max_lut and lay_start in max_lut[host_src]:
        lay_max = max_lut[host_src][lay_start][0]
        sched.append({host_src: [lay_start, lay_max]})
        used.add(host_src)
        lay_start = lay_max + 1
    if host_dest in max_lut and host_src != host_dest:
        lay_min = lay_end + 1
        for l_s, (l_e, _) in max_lut[host_dest].items():
            if l_e == lay_end:
                lay_min = min(l_s, lay_min)
        if lay_min <= lay_end:
            sched.append({host_dest: [lay_min, lay_end]})
            used.add(host_dest)
            lay_end = lay_min - 1
            ins_off = 1
    while lay_start <= lay_end:
        best = (None, -1, float("inf"))
        for dev, lut in max_lut.items():
            if dev in used or lay_start not in lut:
                continue
            l_e, cost = lut[lay_start]
            if l_e > best[1] or (l_e == best[1] and cost < best[2]):
                best = (dev, l_e, cost)
        if best[0] is None:
            return []
        sched.insert(len(sched) - ins_off, {best[0]: [lay_start, best[1]]})
        used.add(best[0])
        lay_start = best[1] + 1
    if host_dest not in sched[-1]:
        sched.append({host_dest: []})
    return sched


# -- shard DAG ---------------------------------------------------------------

NodeID = Tuple[str, Tuple[int, int]]


def _link_bw(bids, dev_a: str, dev_b: str) -> float:
    """Link bandwidth = min of both devices' advertised neighbor bandwidth."""
    bw_ab = bids[dev_a][1].get(dev_b, {}).get("bw_Mbps", 0)
    bw_ba = bids[dev_b][1].get(dev_a, {}).get("bw_Mbps", 0)
    return min(bw_ab, bw_ba)


def _build_dag(bids: Mapping[str, DeviceBidData], yml_model: dict,
               ubatch_size: int, dtype: str, devices: List[str],
               host_src: str, host_dest: str, strict_order: bool,
               strict_first: bool, strict_last: bool):
    """Adjacency-list shard DAG: nodes (device, (l, r)), node weight =
    compute cost, edge weight = boundary comm time. Dummy source/dest carry
    the model input/output transfers."""
    n_layers = yml_model["layers"]
    node_w: Dict[NodeID, float] = {}
    adj: Dict[NodeID, List[Tuple[NodeID, float]]] = {}
    by_start: Dict[str, Dict[int, List[NodeID]]] = {}
    for dev in devices:
        by_start[dev] = {}
        for shard, cost in bids[dev][0].items():
            node = (dev, tuple(shard))
            node_w[node] = cost
            adj[node] = []
            by_start[dev].setdefault(shard[0], []).append(node)

    comm_bytes = [ubatch_bytes(p, ubatch_size, dtype=dtype)
                  for p in yml_model["parameters_out"]]
    for idx, dev_a in enumerate(devices[:-1]):
        dev_bs = devices[idx + 1: idx + 2] if strict_order else devices[idx + 1:]
        for dev_b in dev_bs:
            bw = _link_bw(bids, dev_a, dev_b)
            if bw <= 0:
                continue
            for nodes in by_start[dev_a].values():
                for node1 in nodes:
                    lay_end = node1[1][1]
                    if lay_end + 1 >= n_layers:
                        continue
                    ct = communication_time_bw(bw, comm_bytes[lay_end])
                    for node2 in by_start[dev_b].get(lay_end + 1, []):
                        adj[node1].append((node2, ct))

    node_src: NodeID = (host_src, (-1, -1))
    node_dest: NodeID = (host_dest, (n_layers, n_layers))
    node_w[node_src] = 0.0
    node_w[node_dest] = 0.0
    adj[node_src] = []
    adj.setdefault(node_dest, [])
    in_bytes = ubatch_bytes(yml_model["parameters_in"], ubatch_size, dtype=dtype)
    out_bytes = comm_bytes[-1]
    for node in list(node_w):
        dev, (l_s, l_e) = node
        if node in (node_src, node_dest):
            continue
        if l_s == 0 and (dev == devices[0] or not strict_first):
            if dev == host_src:
                adj[node_src].append((node, 0.0))
            else:
                bw = _link_bw(bids, host_src, dev) if host_src in bids else 0
                if bw > 0:
                    adj[node_src].append(
                        (node, communication_time_bw(bw, in_bytes)))
        if l_e == n_layers - 1 and (dev == devices[-1] or not strict_last):
            if dev == host_dest:
                adj[node].append((node_dest, 0.0))
            else:
                bw = _link_bw(bids, dev, host_dest) if host_dest in bids else 0
                if bw > 0:
                    adj[node].append(
                        (node_dest, communication_time_bw(bw, out_bytes)))
    return node_w, adj, node_src, node_dest


def _dijkstra(node_w, adj, src: NodeID, dst: NodeID, bottleneck: bool):
    """Shortest path by total latency (bottleneck=False) or by minimum
    maximum-stage-latency (bottleneck=True, compute/comm overlapped)."""
    dist: Dict[NodeID, float] = {src: node_w[src]}
    prev: Dict[NodeID, NodeID] = {}
    heap = [(node_w[src], src)]
    seen = set()
    while heap:
        d, node = heapq.heappop(heap)
        if node in seen:
            continue
        seen.add(node)
        if node == dst:
            path = [dst]
            while path[-1] != src:
                path.append(prev[path[-1]])
            return list(reversed(path)), d
        for nxt, edge_w in adj.get(node, ()):
            if bottleneck:
                cand = max(d, edge_w, node_w[nxt])
            else:
                cand = d + edge_w + node_w[nxt]
            if cand < dist.get(nxt, float("inf")):
                dist[nxt] = cand
                prev[nxt] = node
                heapq.heappush(heap, (cand, nxt))
    return [], float("inf")


def _dag_path_to_sched(path: List[NodeID], host_src: str, host_dest: str) \
        -> List[Mapping[str, List[int]]]:
    """Strip the dummy endpoints; keep a trailing empty entry when the data
    host isn't the last stage (reference revauct.py:254-273 semantics)."""
    if not path:
        return []
    assert len(path) > 2
    inner = path[1:-1]
    sched = [{node[0]: list(node[1])} for node in inner]
    if host_dest not in sched[-1]:
        sched.append({host_dest: []})
    if host_src not in sched[0]:
        sched.insert(0, {host_src: []})
    return sched


def _order_asserts(devices, host_src, host_dest):
    if host_src in devices:
        assert devices[0] == host_src
    if host_dest != host_src and host_dest in devices:
        assert devices[-1] == host_dest


def sched_optimal_latency_dev_order(yml_model: dict, ubatch_size: int,
                                    dtype: str,
                                    bids: Mapping[str, DeviceBidData],
                                    host_src: str, host_dest: str,
                                    devices: List[str],
                                    strict_order: bool = True,
                                    strict_first: bool = True,
                                    strict_last: bool = True):
    """Minimize end-to-end latency over the device order; returns
    (schedule, latency_seconds)."""
    _order_asserts(devices, host_src, host_dest)
    node_w, adj, src, dst = _build_dag(bids, yml_model, ubatch_size, dtype,
                                       devices, host_src, host_dest,
                                       strict_order, strict_first, strict_last)
    path, cost = _dijkstra(node_w, adj, src, dst, bottleneck=False)
    return _dag_path_to_sched(path, host_src, host_dest), cost


def sched_optimal_throughput_dev_order(yml_model: dict, ubatch_size: int,
                                       dtype: str,
                                       bids: Mapping[str, DeviceBidData],
                                       host_src: str, host_dest: str,
                                       devices: List[str],
                                       strict_order: bool = True,
                                       strict_first: bool = True,
                                       strict_last: bool = True):
    """Minimize the maximum stage latency; returns (schedule, items/sec
    throughput prediction = ubatch_size / bottleneck)."""
    _order_asserts(devices, host_src, host_dest)
    node_w, adj, src, dst = _build_dag(bids, yml_model, ubatch_size, dtype,
                                       devices, host_src, host_dest,
                                       strict_order, strict_first, strict_last)
    path, cost = _dijkstra(node_w, adj, src, dst, bottleneck=True)
    if not path:
        return [], 0.0
    return _dag_path_to_sched(path, host_src, host_dest), \
        (ubatch_size / cost if cost > 0 else float("inf"))
