"""Backend resolution for the search engine.

The reference routes everything through a user-supplied SparkContext; here
the leading ``ctx`` argument is optional and resolution is automatic
(SURVEY.md C9 -> context plan).
"""

from __future__ import annotations

from .backend import Backend, SequentialBackend, ThreadBackend
from .distributed import DistributedBackend

__all__ = ["Backend", "SequentialBackend", "ThreadBackend",
           "DistributedBackend", "resolve_backend"]


def resolve_backend(ctx=None) -> Backend:
    """Map a context handle to an execution backend.

    Accepted handles:

    * ``None`` / ``"auto"`` -- torch.distributed group if initialized
      (one process per GPU over RCCL, or gloo on CPU), else a local
      thread pool.
    * a :class:`Backend` instance -- used as-is.
    * a :class:`spark_sklearn_amd.HipContext` -- its backend.
    * strings: ``"sequential"``, ``"threads"`` / ``"threads:N"``,
      ``"gpu"`` (single-process sequential dispatch on the default
      device), ``"dist"`` / ``"distributed"``.
    """
    if isinstance(ctx, Backend):
        return ctx
    if hasattr(ctx, "backend") and isinstance(getattr(ctx, "backend"), Backend):
        return ctx.backend
    if isinstance(ctx, str):
        name = ctx.lower()
        if name == "sequential":
            return SequentialBackend()
        if name.startswith("threads"):
            n = int(name.split(":", 1)[1]) if ":" in name else None
            return ThreadBackend(n)
        if name in ("dist", "distributed"):
            return DistributedBackend()
        if name == "gpu":
            # single-process GPU execution: sequential dispatch, device
            # resolved by the context (cuda:0 by default when visible)
            return SequentialBackend()
        if name == "auto":
            ctx = None
        else:
            raise ValueError(f"unknown backend name {ctx!r}")
    if ctx is None:
        try:
            import torch.distributed as dist

            if dist.is_available() and dist.is_initialized():
                return DistributedBackend()
        except Exception:
            pass
        return ThreadBackend()
    raise TypeError(f"cannot resolve a backend from {ctx!r}")
