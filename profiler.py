"""Per-layer offline profiler (parity: reference profiler.py).

Builds a single-layer shard ``[l, l]`` for every schedulable layer, measures
its forward latency and memory, and appends to ``profiler_results.yml`` with
the reference's exact schema (README_Profiler.md:22-73), which feeds
``profiler_results_to_models.py`` / ``profiler_results_to_device_types.py``
and then the scheduler.

MI355X adaptations:
- on GPU, per-layer time comes from hipEvent timing around the HIP-kernel
  forward (the rocprof-measured path the scheduler optimizes for);
  on CPU, wall time like the reference (profiler.py:72-79);
- memory is measured as the GPU allocation delta around shard construction
  (HBM is what gates placement), RSS delta in a fresh subprocess on CPU
  (reference profiler.py:99-118 semantics).
"""
import argparse
import os
import time
from typing import List, Optional

import psutil
import torch
import torch.multiprocessing as mp
import yaml

import devices
import model_cfg


def get_shapes(tensors) -> List[List[int]]:
    """Shapes of a shard-data value, microbatch dim stripped
    (reference profiler.py:16-24)."""
    if isinstance(tensors, tuple):
        return [list(t.shape[1:]) for t in tensors]
    return [list(tensors.shape[1:])]


def profile_module_shard(module_cfg: dict, inputs, layer_start: int,
                         layer_end: int, warmup: int, iterations: int) -> dict:
    """Measure one shard: memory at construction + timed forward loop
    (reference profiler.py:37-97)."""
    device = module_cfg.get("device")
    devices.initialize(device)
    use_cuda = devices.DEVICE is not None and devices.DEVICE.type == "cuda"

    if use_cuda:
        torch.cuda.synchronize()
        torch.cuda.reset_peak_memory_stats()
        mem_before = torch.cuda.memory_allocated()
    else:
        proc = psutil.Process(os.getpid())
        mem_before = proc.memory_info().rss

    module = model_cfg.module_shard_factory(
        module_cfg["name"], module_cfg.get("file"), layer_start, layer_end, 0)

    if use_cuda:
        torch.cuda.synchronize()
        mem_after = torch.cuda.memory_allocated()
    else:
        mem_after = psutil.Process(os.getpid()).memory_info().rss
    memory_mb = (mem_after - mem_before) / 1e6

    dtype = getattr(module, "compute_dtype", torch.float32)
    if isinstance(inputs, torch.Tensor):
        inputs_dev = inputs.to(devices.DEVICE)
        if inputs_dev.is_floating_point():
            inputs_dev = inputs_dev.to(dtype)
    else:
        inputs_dev = tuple(
            t.to(devices.DEVICE).to(dtype) if t.is_floating_point()
            else t.to(devices.DEVICE) for t in inputs)

    with torch.no_grad():
        for _ in range(warmup):
            outputs = module(inputs_dev)
        if use_cuda:
            torch.cuda.synchronize()
            start_ev = torch.cuda.Event(enable_timing=True)
            end_ev = torch.cuda.Event(enable_timing=True)
            start_ev.record()
            for _ in range(iterations):
                outputs = module(inputs_dev)
            end_ev.record()
            torch.cuda.synchronize()
            time_s = start_ev.elapsed_time(end_ev) / 1e3 / iterations
        else:
            tik = time.perf_counter()
            for _ in range(iterations):
                outputs = module(inputs_dev)
            time_s = (time.perf_counter() - tik) / iterations

    return {
        "layer": layer_start,
        "memory": memory_mb,
        "shape_in": get_shapes(inputs_dev),
        "shape_out": get_shapes(outputs),
        "time": time_s,
        "_outputs": tuple(o.cpu() for o in outputs)
        if isinstance(outputs, tuple) else outputs.cpu(),
    }


def _np_pack(data):
    """Tensors -> by-value numpy for the mp queue: torch's shared-storage
    tensor pickling needs the CHILD alive while the parent materializes the
    storage, and the child exits right after put() (ConnectionResetError)."""
    if isinstance(data, tuple):
        return tuple(t.detach().float().cpu().numpy() for t in data)
    return data.detach().float().cpu().numpy()


def _np_unpack(data):
    if isinstance(data, tuple):
        return tuple(torch.from_numpy(a) for a in data)
    return torch.from_numpy(data)


def _profile_mp_queue(queue, module_cfg, inputs, layer_start, layer_end,
                      warmup, iterations):
    res = profile_module_shard(module_cfg, inputs, layer_start, layer_end,
                               warmup, iterations)
    res["_outputs"] = _np_pack(res["_outputs"])
    queue.put(res)


def profile_module_shard_mp(module_cfg, inputs, layer_start, layer_end,
                            warmup, iterations) -> dict:
    """Profile in a fresh subprocess for clean memory accounting
    (reference profiler.py:99-118)."""
    ctx = mp.get_context("spawn")
    queue = ctx.Queue()
    proc = ctx.Process(target=_profile_mp_queue,
                       args=(queue, module_cfg, inputs, layer_start,
                             layer_end, warmup, iterations))
    proc.start()
    result = queue.get()
    proc.join()
    result["_outputs"] = _np_unpack(result["_outputs"])
    return result


def profile_layers_individually(module_cfg, inputs, layer_start, layer_end,
                                warmup, iterations) -> List[dict]:
    """Each layer [l, l] on its own; outputs feed the next layer's inputs
    (reference profiler.py:121-144)."""
    results = []
    data = inputs
    for layer in range(layer_start, layer_end + 1):
        res = profile_module_shard_mp(module_cfg, data, layer, layer, warmup,
                                      iterations)
        data = res.pop("_outputs")
        results.append(res)
    return results


def profile_layers_cumulatively(module_cfg, inputs, layer_start, layer_end,
                                warmup, iterations) -> List[dict]:
    """Shards [start, l] of growing size (reference profiler.py:146-160)."""
    results = []
    for layer in range(layer_start, layer_end + 1):
        res = profile_module_shard_mp(module_cfg, inputs, layer_start, layer,
                                      warmup, iterations)
        res.pop("_outputs")
        res["layer"] = layer
        results.append(res)
    return results


def validate_profile_results(profile_results, args, inputs, model_layers,
                             layer_end):
    assert profile_results["model_name"] == args.model_name
    assert profile_results["batch_size"] == args.batch_size
    assert profile_results["layers"] == model_layers


def main() -> None:
    parser = argparse.ArgumentParser(
        description="Per-layer model profiler",
        formatter_class=argparse.ArgumentDefaultsHelpFormatter)
    parser.add_argument("-d", "--device", type=str, default=None,
                        help="compute device: 'cpu', 'cuda', ...")
    parser.add_argument("-m", "--model-name", type=str,
                        default="google/vit-base-patch16-224",
                        choices=model_cfg.get_model_names())
    parser.add_argument("-M", "--model-file", type=str)
    parser.add_argument("-b", "--batch-size", default=8, type=int,
                        help="microbatch size to profile")
    parser.add_argument("-w", "--warmup", default=2, type=int)
    parser.add_argument("-i", "--iterations", default=10, type=int)
    parser.add_argument("-l", "--layer-start", default=1, type=int)
    parser.add_argument("-L", "--layer-end", default=None, type=int)
    parser.add_argument("-o", "--results-yml", default="profiler_results.yml",
                        type=str)
    args = parser.parse_args()

    config = model_cfg.get_model_config(args.model_name)
    if config.model_type in ("vit", "deit"):
        inputs = torch.randn(args.batch_size, config.num_channels,
                             config.image_size, config.image_size)
    else:
        inputs = torch.randint(0, config.vocab_size,
                               (args.batch_size,
                                config.max_position_embeddings))

    model_layers = model_cfg.get_model_layers(args.model_name)
    layer_end = args.layer_end if args.layer_end is not None else model_layers

    if os.path.exists(args.results_yml):
        print("Using existing results file")
        with open(args.results_yml, encoding="utf-8") as yfile:
            profile_results = yaml.safe_load(yfile)
        validate_profile_results(profile_results, args, inputs, model_layers,
                                 layer_end)
    else:
        dtype = "torch.bfloat16" if torch.cuda.is_available() and \
            (args.device is None or "cuda" in args.device) else "torch.float32"
        profile_results = {
            "model_name": args.model_name,
            "dtype": dtype,
            "batch_size": args.batch_size,
            "layers": model_layers,
            "profile_data": [],
        }

    module_cfg = {
        "device": args.device,
        "name": args.model_name,
        "file": args.model_file or
        model_cfg.get_model_default_weights_file(args.model_name),
    }
    results = profile_layers_individually(module_cfg, inputs,
                                          args.layer_start, layer_end,
                                          args.warmup, args.iterations)
    profile_results["profile_data"].extend(results)
    profile_results["profile_data"].sort(key=lambda pd: pd["layer"])
    with open(args.results_yml, "w", encoding="utf-8") as yfile:
        yaml.safe_dump(profile_results, yfile, default_flow_style=None)


if __name__ == "__main__":
    main()
