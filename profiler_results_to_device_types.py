"""Convert profiler_results.yml -> scheduler device_types.yml
(parity: reference profiler_results_to_device_types.py).

Takes the measured per-layer times and attaches user-supplied device
properties: HBM budget (``--mem-mb``, default 272 GB = 288 GB HBM3E minus
runtime headroom) and link bandwidth (``--bw-mbps``, default one xGMI p2p
link at ~153 GB/s). Merge semantics mirror the reference
(profiler_results_to_device_types.py:29-75): a device type accumulates
profiles per (model, dtype, batch_size); re-profiling overwrites with
``--overwrite-profile``."""
import argparse

import yaml

from pipeedge_amd.sched.yaml_types import yaml_device_type, yaml_model_profile

# one xGMI link ~153 GB/s in Mbps
XGMI_LINK_MBPS = int(153e9 * 8 / (1024 * 1024))
# HBM3E capacity minus headroom, in MB
MI355X_MEM_MB = int(272e9 / (1024 * 1024))


def main() -> None:
    parser = argparse.ArgumentParser(
        description="Convert profiler results to a scheduler device types file",
        formatter_class=argparse.ArgumentDefaultsHelpFormatter)
    parser.add_argument("-r", "--results-yml", default="profiler_results.yml")
    parser.add_argument("-o", "--dev-types-yml", default="device_types.yml")
    parser.add_argument("-t", "--dev-type", default="mi355x",
                        help="device type name to create/update")
    parser.add_argument("--mem-mb", type=float, default=MI355X_MEM_MB)
    parser.add_argument("--bw-mbps", type=float, default=XGMI_LINK_MBPS)
    parser.add_argument("--overwrite-profile", action="store_true")
    args = parser.parse_args()

    with open(args.results_yml, encoding="utf-8") as yfile:
        profile_results = yaml.safe_load(yfile)
    model_name = profile_results["model_name"]
    data = sorted(profile_results["profile_data"], key=lambda d: d["layer"])
    times = [float(d["time"]) for d in data]
    profile = yaml_model_profile(profile_results["dtype"],
                                 profile_results["batch_size"], times)

    try:
        with open(args.dev_types_yml, encoding="utf-8") as yfile:
            dev_types = yaml.safe_load(yfile) or {}
    except FileNotFoundError:
        dev_types = {}

    if args.dev_type in dev_types:
        entry = dev_types[args.dev_type]
        entry["mem_MB"] = args.mem_mb
        entry["bw_Mbps"] = args.bw_mbps
    else:
        entry = yaml_device_type(args.mem_mb, args.bw_mbps, {})
        dev_types[args.dev_type] = entry
    profiles = entry.setdefault("model_profiles", {}).setdefault(model_name, [])
    for i, existing in enumerate(profiles):
        if existing["dtype"] == profile["dtype"] and \
                existing["batch_size"] == profile["batch_size"]:
            if not args.overwrite_profile:
                raise RuntimeError(
                    f"profile for {model_name} ({profile['dtype']}, "
                    f"b={profile['batch_size']}) exists; "
                    "use --overwrite-profile")
            profiles[i] = profile
            break
    else:
        profiles.append(profile)

    with open(args.dev_types_yml, "w", encoding="utf-8") as yfile:
        yaml.safe_dump(dev_types, yfile, default_flow_style=None)


if __name__ == "__main__":
    main()
