"""
Native op loading.

The gfx950 HIP extension is built IN-TREE (setup.py build_ext --inplace
-> dragnet_amd/ops/_dragnet_hip.*.so) so it ships with the source tree.
On a GPU machine a missing extension is a HARD error — the engine never
silently falls back to an eager path.
"""

_ops = None
_load_err = None


def load_ops(required=True):
    global _ops, _load_err
    if _ops is not None:
        return _ops
    try:
        import torch  # noqa: F401 — loads libc10/libtorch for the ext
        from . import _dragnet_hip as ops
        _ops = ops
        return ops
    except ImportError as e:
        _load_err = e
        if required:
            raise RuntimeError(
                "dragnet_amd HIP extension not built. Build it in-tree "
                "with: python setup.py build_ext --inplace "
                "(PYTORCH_ROCM_ARCH=gfx950). Underlying error: %s" % e)
        return None
