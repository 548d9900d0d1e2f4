"""Native CDNA4 kernel extension loader + device executors.

The ``_hip_ops`` extension (built from ops/csrc/*.hip for gfx950 by
setup.py / __graft_entry__.build) holds the hand-written HIP kernels:
K1 batched linear fit, K4 scoring, K2 kmeans, K5 CSR (SURVEY.md
section 2.5).  Policy (task requirement): on a GPU box the extension is
MANDATORY -- a missing .so raises ImportError with build instructions
rather than silently falling back to an eager path, so GPU test runs
always exercise the native kernels.
"""

from __future__ import annotations


_ext = None
_ext_err: Exception | None = None


def load_extension():
    """Import the compiled _hip_ops extension (in-tree .so). Cached."""
    global _ext, _ext_err
    if _ext is not None:
        return _ext
    if _ext_err is not None:
        raise ImportError(
            f"spark_sklearn_amd._hip_ops failed to import: {_ext_err!r}. "
            f"Build it with `python setup.py build_ext --inplace` "
            f"(PYTORCH_ROCM_ARCH=gfx950)."
        ) from _ext_err
    try:
        import torch  # noqa: F401  - extension links against libtorch

        from spark_sklearn_amd import _hip_ops  # type: ignore

        _ext = _hip_ops
        return _ext
    except Exception as exc:  # noqa: BLE001
        _ext_err = exc
        raise ImportError(
            f"spark_sklearn_amd._hip_ops is not available ({exc!r}). "
            f"On a GPU box this is a hard error: build the extension with "
            f"`python setup.py build_ext --inplace` (PYTORCH_ROCM_ARCH="
            f"gfx950); there is no eager fallback on device."
        ) from exc


def extension_available() -> bool:
    try:
        load_extension()
        return True
    except ImportError:
        return False


from .device_executor import DeviceLinearExecutor  # noqa: E402

__all__ = ["load_extension", "extension_available", "DeviceLinearExecutor"]
