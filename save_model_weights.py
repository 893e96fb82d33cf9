"""Save model weights files (parity: reference save_model_weights.py).

The reference downloads pretrained checkpoints; this environment has no
network, so each shard class synthesizes a deterministic random-init
checkpoint in its canonical npz key layout (Google ViT / timm DeiT / HF
BERT) - the same files the runtime, profiler, and tests consume."""
import argparse

import model_cfg


def main() -> None:
    parser = argparse.ArgumentParser(
        description="Save model weights files",
        formatter_class=argparse.ArgumentDefaultsHelpFormatter)
    parser.add_argument("-m", "--model-name", type=str, default=None,
                        choices=model_cfg.get_model_names(),
                        help="model to save (default: all)")
    parser.add_argument("-M", "--model-file", type=str, default=None,
                        help="output file (default: the model's canonical name)")
    args = parser.parse_args()
    names = [args.model_name] if args.model_name else model_cfg.get_model_names()
    for name in names:
        model_file = args.model_file if args.model_name else None
        print(f"saving weights: {name}")
        model_cfg.save_model_weights_file(name, model_file=model_file)


if __name__ == "__main__":
    main()
