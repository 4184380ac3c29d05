"""Loader for the in-tree HIP extension (_hip.*.so).

The extension is built in-tree by `python setup.py build_ext --inplace`
(or __graft_entry__.build()) so the .so travels with the repo snapshot.
On a machine WITH a GPU the extension is mandatory: a missing .so raises
instead of silently falling back to a CPU path.
"""

import importlib

_ops = None


def get_ops():
    global _ops
    if _ops is not None:
        return _ops
    import torch  # noqa: F401  — loads libc10/libtorch the extension links
    try:
        _ops = importlib.import_module("flake16_framework_amd.ops._hip")
    except ImportError as e:
        import torch
        if torch.cuda.is_available():
            raise RuntimeError(
                "flake16_framework_amd HIP extension (_hip) is not built but "
                "a GPU is present.  Build it in-tree with "
                "`python setup.py build_ext --inplace` "
                "(PYTORCH_ROCM_ARCH=gfx950).  Refusing to fall back to the "
                "CPU reference path on a GPU machine.") from e
        raise
    return _ops


def hip_available():
    try:
        get_ops()
        return True
    except (ImportError, RuntimeError):
        return False
