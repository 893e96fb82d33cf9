"""Loader for the in-tree gfx950 kernel extension.

The extension is built IN-TREE (``python setup.py build_ext --inplace``) so
the ``.so`` ships with the repo snapshot to GPU boxes; a JIT cache under
``~/.cache`` would not.
"""
import importlib
import os


def load():
    """Import the built extension module; raises ImportError if absent."""
    # ensure torch symbols are loaded first
    import torch  # noqa: F401
    try:
        return importlib.import_module("pipeedge_amd.ops._hip_ops")
    except ImportError as exc:
        here = os.path.dirname(os.path.abspath(__file__))
        raise ImportError(
            f"pipeedge_amd.ops._hip_ops not importable ({exc}); expected a "
            f"built .so in {here}. Run `python setup.py build_ext --inplace` "
            f"with PYTORCH_ROCM_ARCH=gfx950.") from exc
