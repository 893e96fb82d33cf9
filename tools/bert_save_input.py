"""Generate a BERT evaluation input file (parity: reference
tools/bert_save_input.py).

The reference samples GLUE/CoLA sentences (network required); offline we
synthesize deterministic token sequences of the same shape. With
``--use-glue`` and a local HF datasets cache, the CoLA path is used instead.
"""
import argparse

import numpy as np


def main():
    parser = argparse.ArgumentParser(
        description=__doc__,
        formatter_class=argparse.ArgumentDefaultsHelpFormatter)
    parser.add_argument("-o", "--output", default="bert_input.npz")
    parser.add_argument("-n", "--num-inputs", type=int, default=512)
    parser.add_argument("--use-glue", action="store_true")
    parser.add_argument("--tokenized", action="store_true",
                        help="write token IDS directly (variable-length "
                             "sequences padded with id 0) - usable offline "
                             "by `runtime.py --dataset-name npz`")
    parser.add_argument("--max-len", type=int, default=128)
    args = parser.parse_args()
    if args.tokenized:
        rng = np.random.default_rng(0)
        n, smax = args.num_inputs, args.max_len
        ids = np.zeros((n, smax), dtype=np.int64)  # 0 = [PAD]
        for i in range(n):
            ln = int(rng.integers(8, smax + 1))
            ids[i, :ln] = rng.integers(1000, 30000, size=ln)
        label = rng.integers(0, 2, size=n)
        np.savez(args.output, ids=ids, label=label)
        print(f"wrote {args.output} (tokenized, variable-length)")
        return
    if args.use_glue:
        import datasets
        ds_train = datasets.load_dataset("glue", name="cola", split="train")
        bert_input = ds_train[:args.num_inputs]["sentence"]
        bert_label = ds_train[:args.num_inputs]["label"]
        bert_input[0] = "hello " * 512  # force max-width tokenization
        bert_label[0] = 0
    else:
        rng = np.random.default_rng(0)
        words = ["alpha", "beta", "gamma", "delta", "epsilon", "zeta", "eta"]
        bert_input = [" ".join(rng.choice(words, size=rng.integers(4, 32)))
                      for _ in range(args.num_inputs)]
        bert_input[0] = "hello " * 512
        bert_label = rng.integers(0, 2, size=args.num_inputs).tolist()
    np.savez(args.output, input=bert_input, label=bert_label)
    print(f"wrote {args.output}")


if __name__ == "__main__":
    main()
