"""HIP kernel ops for the MI355X autoscaler.

Importing this module on a machine WITH a ROCm device requires the
in-tree extension (mi355x_autoscaler/_hip_ops*.so). There is no silent
eager fallback: if a GPU is visible and the extension is missing, ops
raise immediately — build with
``PYTORCH_ROCM_ARCH=gfx950 python setup.py build_ext --inplace``.
"""

from __future__ import annotations

_hip_ops = None
_import_error = None


def load_ops():
    """Return the compiled extension module, raising loudly on GPU
    machines where it is absent."""
    global _hip_ops, _import_error
    if _hip_ops is not None:
        return _hip_ops
    import torch  # noqa: F401  (libc10 must be loaded first)

    try:
        from mi355x_autoscaler import _hip_ops as ext  # type: ignore

        if hasattr(ext, "compiled_constants"):
            from ..snapshot.encode import MAX_SELECTOR_PAIRS

            cc = ext.compiled_constants()
            if cc["MAXSEL"] != MAX_SELECTOR_PAIRS:
                raise RuntimeError(
                    f"_hip_ops compiled with MAXSEL={cc['MAXSEL']} but the "
                    f"lattice uses {MAX_SELECTOR_PAIRS}: stale build — "
                    "rebuild with `python setup.py build_ext --inplace`"
                )
        _hip_ops = ext
        return ext
    except ImportError as e:  # pragma: no cover
        _import_error = e
        raise RuntimeError(
            "mi355x_autoscaler._hip_ops is not built. Run "
            "`PYTORCH_ROCM_ARCH=gfx950 python setup.py build_ext --inplace` "
            f"(original error: {e})"
        ) from e
