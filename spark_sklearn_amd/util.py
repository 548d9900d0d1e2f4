"""Context and convenience helpers.

Behavioral counterpart of the reference's ``util.py``
(``createLocalSparkSession(appName=...)`` ~L10 [R]): where the
reference spins up a local SparkSession, this module discovers HIP devices,
wires up torch.distributed (RCCL over xGMI when launched one-process-per-GPU
with torchrun; gloo on CPU) and hands back a context object that the search
classes accept as their leading argument.
"""

from __future__ import annotations

import os
from typing import Optional

__all__ = ["HipContext", "createLocalContext", "init_distributed",
           "enable_prometheus"]


class HipContext:
    """Execution context: device discovery + backend + engine knobs.

    Engine-level knobs live here rather than on the search objects so the
    search constructor surface stays sklearn-compatible (SURVEY.md
    section 5.6): ``candidate_batch`` (candidates folded into one kernel
    launch), ``n_streams`` (HIP streams per GPU), ``dtype`` for the
    resident matrix.
    """

    def __init__(self, backend=None, device=None, candidate_batch: int = 256,
                 n_streams: int = 1, dtype: str = "bfloat16",
                 n_threads: Optional[int] = None, score_log: str = None,
                 resume_from_log: str = None, progress=None,
                 partition: str = "static", dynamic_block: int = 8):
        from .parallel import resolve_backend

        if dtype != "bfloat16":
            raise ValueError(
                "dtype='bfloat16' is the only resident dtype (the MFMA "
                "fit kernels are bf16 with fp32 accumulation); an fp8 "
                "opt-in is a roadmap item -- see ROADMAP.md and the "
                "round-2 ablation study for why it is not the default")
        if n_streams != 1:
            raise ValueError(
                "n_streams is reserved: one compute stream per GPU. "
                "Multi-stream overlap was MEASURED in round 2 (the "
                "pass-pipelined v17 epoch, profiles/"
                "r02_flagship_ablation.md) and lost 23% -- the big-LDS "
                "fwd/bwd blocks cannot co-reside on a CU, so concurrent "
                "kernels partition the chip. Honest single value until "
                "a co-residable design exists")
        if not (1 <= int(candidate_batch) <= 256):
            raise ValueError("candidate_batch must be in [1, 256]")
        if partition not in ("static", "dynamic"):
            raise ValueError("partition must be 'static' or 'dynamic'")
        self.partition = partition
        self.dynamic_block = int(dynamic_block)
        self.candidate_batch = candidate_batch
        self.n_streams = n_streams
        self.dtype = dtype
        self.n_threads = n_threads
        self.score_log = score_log
        self.resume_from_log = resume_from_log
        self.progress = progress
        self.device = device if device is not None else _default_device()
        self.backend = resolve_backend(backend)
        # push engine knobs down where backends look for them
        for attr in ("candidate_batch", "n_streams", "dtype", "score_log",
                     "resume_from_log", "progress", "partition",
                     "dynamic_block"):
            setattr(self.backend, attr, getattr(self, attr))

    @property
    def n_devices(self) -> int:
        try:
            import torch

            return torch.cuda.device_count() if torch.cuda.is_available() else 0
        except Exception:
            return 0

    def __repr__(self):
        return (f"HipContext(device={self.device}, "
                f"n_devices={self.n_devices}, "
                f"backend={type(self.backend).__name__}, "
                f"candidate_batch={self.candidate_batch})")


def _default_device():
    try:
        import torch

        if torch.cuda.is_available():
            local = int(os.environ.get("LOCAL_RANK", 0))
            return torch.device("cuda", local % torch.cuda.device_count())
    except Exception:
        pass
    return None


def createLocalContext(appName: str = "spark-sklearn-amd",
                       **kwargs) -> HipContext:
    """Local-mode context (name kept parallel to the reference's
    ``createLocalSparkSession``).  Uses the local GPU when present, else a
    CPU thread pool."""
    return HipContext(**kwargs)


def init_distributed(backend: Optional[str] = None,
                     timeout_seconds: int = 1800) -> HipContext:
    """Initialize torch.distributed from torchrun env vars and return a
    distributed context.

    On ROCm the ``"nccl"`` backend IS RCCL; collectives run over xGMI.
    Falls back to gloo when no GPU is visible (the CPU CI tier).
    """
    import datetime

    import torch
    import torch.distributed as dist

    if not dist.is_initialized():
        if backend is None:
            backend = "nccl" if torch.cuda.is_available() else "gloo"
        if torch.cuda.is_available():
            local = int(os.environ.get("LOCAL_RANK", 0))
            torch.cuda.set_device(local % torch.cuda.device_count())
        dist.init_process_group(
            backend=backend,
            timeout=datetime.timedelta(seconds=timeout_seconds))
    return HipContext(backend="dist")


# -- optional Prometheus metrics -------------------
_PROM: dict = {"enabled": False}


def enable_prometheus(port: int = 9109):
    """Expose search counters on a Prometheus endpoint (optional --
    requires prometheus_client, present in this image's wheelhouse).

    Metrics: ``ska_candidate_fits_total``, ``ska_searches_total``,
    ``ska_fits_per_sec`` (last search's rate -- the north-star metric).
    """
    if _PROM["enabled"]:
        return
    from prometheus_client import Counter, Gauge, start_http_server

    start_http_server(port)
    _PROM.update(
        enabled=True,
        fits=Counter("ska_candidate_fits_total",
                     "completed (candidate, fold) fits"),
        searches=Counter("ska_searches_total", "completed searches"),
        fps=Gauge("ska_fits_per_sec",
                  "candidate-fits/sec of the last search"))


def record_search_metrics(n_fits: int, fits_per_sec: float):
    """Called by BaseSearchCV.fit; no-op unless enable_prometheus ran."""
    if not _PROM["enabled"]:
        return
    _PROM["fits"].inc(n_fits)
    _PROM["searches"].inc()
    _PROM["fps"].set(fits_per_sec)
