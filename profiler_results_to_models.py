"""Convert profiler_results.yml -> scheduler models.yml
(parity: reference profiler_results_to_models.py).

``parameters_in`` / ``parameters_out`` come from the recorded per-layer
shapes (product over all tensors of a boundary), ``mem_MB`` from the measured
memory deltas."""
import argparse
from functools import reduce

import yaml

from pipeedge_amd.sched.yaml_types import yaml_model


def _shape_params(shapes) -> int:
    return sum(reduce(lambda a, b: a * b, shape, 1) for shape in shapes)


def profile_to_yaml_model(profile_results: dict) -> dict:
    data = sorted(profile_results["profile_data"], key=lambda d: d["layer"])
    layers = profile_results["layers"]
    assert len(data) == layers, \
        f"profile has {len(data)} layers, model has {layers}"
    parameters_in = _shape_params(data[0]["shape_in"])
    parameters_out = [_shape_params(d["shape_out"]) for d in data]
    mem_mb = [float(d["memory"]) for d in data]
    return yaml_model(layers, parameters_in, parameters_out, mem_mb)


def main() -> None:
    parser = argparse.ArgumentParser(
        description="Convert profiler results to a scheduler models file",
        formatter_class=argparse.ArgumentDefaultsHelpFormatter)
    parser.add_argument("-r", "--results-yml", default="profiler_results.yml")
    parser.add_argument("-o", "--models-yml", default="models.yml")
    parser.add_argument("--overwrite-model", action="store_true",
                        help="replace an existing entry for the same model")
    args = parser.parse_args()

    with open(args.results_yml, encoding="utf-8") as yfile:
        profile_results = yaml.safe_load(yfile)
    model_name = profile_results["model_name"]

    try:
        with open(args.models_yml, encoding="utf-8") as yfile:
            models = yaml.safe_load(yfile) or {}
    except FileNotFoundError:
        models = {}
    if model_name in models and not args.overwrite_model:
        raise RuntimeError(f"{model_name} already in {args.models_yml}; "
                           "use --overwrite-model")
    models[model_name] = profile_to_yaml_model(profile_results)
    with open(args.models_yml, "w", encoding="utf-8") as yfile:
        yaml.safe_dump(models, yfile, default_flow_style=None)


if __name__ == "__main__":
    main()
