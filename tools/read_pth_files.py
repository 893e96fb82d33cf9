"""Print the keys/shapes of a torch .pth checkpoint
(parity: reference tools/read_pth_files.py)."""
import sys

import torch


def main():
    for path in sys.argv[1:]:
        state = torch.load(path, map_location="cpu", weights_only=True)
        if hasattr(state, "state_dict"):
            state = state.state_dict()
        print(f"== {path}")
        for key, val in state.items():
            shape = tuple(val.shape) if hasattr(val, "shape") else type(val)
            print(f"  {key}: {shape}")


if __name__ == "__main__":
    main()
