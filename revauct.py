"""Distributed reverse-auction scheduling app (parity: reference revauct.py).

Rank 0 is the auctioneer: it broadcasts a bid request over RPC, every rank
bids for the shards its device type can hold (latency-costed from its local
YAML profiles), and the auctioneer filters/schedules with one of the three
FMEC'23 schedulers, printing a PipeEdge-compatible 1-indexed schedule.
"""
import argparse
import logging
import random
import time
from typing import List, Tuple

import torch
import yaml
from torch.distributed import rpc

import runtime
from pipeedge_amd.comm.rpc import DistRpcContext
from pipeedge_amd.sched import revauct
from pipeedge_amd.sched import yaml_files

logger = logging.getLogger(__name__)

Bid = Tuple[List[Tuple[int, int]], List[float], dict]

# populated locally on each rank before bidding (reference revauct.py:29-64)
_DEVICE_CFG = {}


def _find_profiles(model: str, ubatch_size: int, dtype: str):
    """Locate (model, device type, profile) entries in the local YAMLs."""
    yml_model = _DEVICE_CFG["yml_models"].get(model)
    yml_dev_type = _DEVICE_CFG["yml_dev_types"].get(_DEVICE_CFG["dev_type"])
    yml_dtm_profile = None
    if yml_dev_type is not None:
        for prof in yml_dev_type.get("model_profiles", {}).get(model, []):
            if prof["dtype"] == dtype and prof["batch_size"] == ubatch_size:
                yml_dtm_profile = prof
                break
    return yml_model, yml_dev_type, yml_dtm_profile


def revauct_bid_latency(model: str, ubatch_size: int,
                        dtype: str = "torch.float32") -> Tuple[str, Bid]:
    """RPC target: respond with this host's bid (reference revauct.py:68-87)."""
    yml_model, yml_dev_type, yml_dtm_profile = _find_profiles(model,
                                                             ubatch_size,
                                                             dtype)
    shards: List[Tuple[int, int]] = []
    costs: List[float] = []
    if yml_model and yml_dev_type and yml_dtm_profile:
        for shard, cost in revauct.bid_latency(yml_model, yml_dev_type,
                                               yml_dtm_profile, ubatch_size,
                                               dtype=dtype):
            shards.append(shard)
            costs.append(cost)
    host = _DEVICE_CFG["host"]
    neighbors = _DEVICE_CFG["yml_dev_neighbors_world"].get(host, {})
    return (host, (shards, costs, neighbors))


def main() -> None:
    parser = argparse.ArgumentParser(
        description="Reverse-auction distributed scheduler",
        formatter_class=argparse.ArgumentDefaultsHelpFormatter)
    parser.add_argument("rank", type=int)
    parser.add_argument("worldsize", type=int)
    parser.add_argument("--host", type=str, required=True,
                        help="this device's host name (in the YAML files)")
    parser.add_argument("--dev-type", type=str, required=True,
                        help="this device's type (in device_types.yml)")
    parser.add_argument("-s", "--socket-ifname", type=str, default="lo0")
    parser.add_argument("--addr", type=str, default="127.0.0.1")
    parser.add_argument("--port", type=int, default=29501)
    parser.add_argument("-m", "--model-name", type=str,
                        default="google/vit-base-patch16-224")
    parser.add_argument("-u", "--ubatch-size", default=8, type=int)
    parser.add_argument("-sm", "--sched-models-file", default="models.yml")
    parser.add_argument("-sdt", "--sched-dev-types-file",
                        default="device_types.yml")
    parser.add_argument("-sdnw", "--sched-dev-neighbors-world",
                        default="device_neighbors_world.yml")
    parser.add_argument("--data-host", type=str, default=None,
                        help="host where inputs/outputs live (default: rank 0)")
    parser.add_argument("-S", "--scheduler", default="throughput_ordered",
                        choices=["latency_ordered", "throughput_ordered",
                                 "greedy_host_count"])
    parser.add_argument("--filter-bids-chunk", type=int, default=1)
    parser.add_argument("--filter-bids-largest", action="store_true")
    parser.add_argument("--no-strict-order", action="store_true")
    parser.add_argument("--strict-first", action="store_true")
    parser.add_argument("--strict-last", action="store_true")
    parser.add_argument("--dev-count", type=int, default=1000000)
    parser.add_argument("--seed", type=int, default=None,
                        help="device-order shuffle seed")
    args = parser.parse_args()

    _DEVICE_CFG["yml_models"] = yaml_files.yaml_models_load(
        args.sched_models_file)
    _DEVICE_CFG["yml_dev_types"] = yaml_files.yaml_device_types_load(
        args.sched_dev_types_file)
    _DEVICE_CFG["yml_dev_neighbors_world"] = \
        yaml_files.yaml_device_neighbors_world_load(
            args.sched_dev_neighbors_world)
    _DEVICE_CFG["dev_type"] = args.dev_type
    _DEVICE_CFG["host"] = args.host

    runtime.init_env(None, args.addr, args.port, args.socket_ifname)
    with DistRpcContext(args.worldsize, args.rank):
        if args.rank == 0:
            yml_model = _DEVICE_CFG["yml_models"][args.model_name]
            dtype = "torch.float32"
            futs = [rpc.rpc_async(r, revauct_bid_latency,
                                  args=(args.model_name, args.ubatch_size))
                    for r in range(args.worldsize)]
            bids_in_rank_order = torch.futures.wait_all(futs)
            bid_data_by_host = {
                b[0]: ({tuple(l): c for l, c in zip(b[1][0], b[1][1])},
                       b[1][2])
                for b in bids_in_rank_order}
            logger.info("Received bids from %d hosts", len(bid_data_by_host))

            if args.filter_bids_chunk > 1:
                bid_data_by_host = {
                    h: (revauct.filter_bids_chunk(yml_model, b[0],
                                                  chunk=args.filter_bids_chunk),
                        b[1]) for h, b in bid_data_by_host.items()}
            if args.filter_bids_largest:
                bid_data_by_host = {h: (revauct.filter_bids_largest(b[0]), b[1])
                                    for h, b in bid_data_by_host.items()}

            data_host = args.data_host if args.data_host else args.host
            dev_order = list(bid_data_by_host.keys())
            rng = random.Random(args.seed)
            rng.shuffle(dev_order)
            dev_order = dev_order[:args.dev_count]
            for idx, dev in enumerate(dev_order):
                if dev == data_host:
                    dev_order[0], dev_order[idx] = dev_order[idx], dev_order[0]
            logger.info("Device order: %s", dev_order)

            t_start = time.time()
            if args.scheduler == "latency_ordered":
                schedule, pred = revauct.sched_optimal_latency_dev_order(
                    yml_model, args.ubatch_size, dtype, bid_data_by_host,
                    data_host, data_host, dev_order,
                    strict_order=not args.no_strict_order,
                    strict_first=args.strict_first,
                    strict_last=args.strict_last)
                logger.info("Latency prediction (sec): %s", pred)
            elif args.scheduler == "throughput_ordered":
                schedule, pred = revauct.sched_optimal_throughput_dev_order(
                    yml_model, args.ubatch_size, dtype, bid_data_by_host,
                    data_host, data_host, dev_order,
                    strict_order=not args.no_strict_order,
                    strict_first=args.strict_first,
                    strict_last=args.strict_last)
                logger.info("Throughput prediction (items/sec): %s", pred)
            else:
                schedule = revauct.sched_greedy_host_count(
                    yml_model, args.ubatch_size, dtype, bid_data_by_host,
                    data_host, data_host)
            logger.info("Scheduler runtime (sec): %s", time.time() - t_start)
            logger.info("Schedule stages: %d", len(schedule))

            # shift to the runtime's 1-indexed layers
            sched_compat = [{host: [l + 1 for l in layers]
                             for host, layers in part.items()}
                            for part in schedule]
            logger.info("Schedule:")
            print(yaml.safe_dump(sched_compat, default_flow_style=None,
                                 sort_keys=False))


if __name__ == "__main__":
    logging.basicConfig(format="%(message)s", level=logging.INFO)
    main()
