#!/usr/bin/env python3
"""Flagship benchmark: BASELINE config 2.

LogisticRegression C x penalty grid on synthetic dense 1M x 256 bf16,
cv=5, candidate fits batched through the CDNA4 MFMA kernels -- the
(candidate x fold) task graph of the reference's GridSearchCV executed on
MI355X GPUs (metric of record in BASELINE.json: candidate-fits/sec).

One *step* = each rank fitting ``--cands-per-step`` candidates through all
``--cv`` folds (fit + test-fold scoring) -- i.e. a fixed slice of the 10k
grid; per-GPU work is constant as N grows (weak scaling).  A candidate-fit
is one (candidate, fold) model fit: ``value = cands * cv * N / seconds``.

Launch (the driver does this):
  1 GPU : python bench.py --gpus 1 --steps K --warmup W
  N GPUs: python -m torch.distributed.run --nnodes=1 --nproc-per-node N \
             --master-addr 127.0.0.1 bench.py --gpus N --steps K --warmup W

Each fit is a full-batch GD LogisticRegression (fixed --epochs data passes
over the 800k-row train fold, fp32 accumulation, bf16 storage/MFMA).
Convergence evidence vs sklearn on the same data: BASELINE.md.
"""

import argparse
import json
import os
import sys
import time

import numpy as np


def parse_args():
    p = argparse.ArgumentParser()
    p.add_argument("--gpus", type=int, default=1)
    p.add_argument("--steps", type=int, default=8)
    p.add_argument("--warmup", type=int, default=3)
    p.add_argument("--rows", type=int, default=1_000_000)
    p.add_argument("--features", type=int, default=256)
    p.add_argument("--cv", type=int, default=5)
    p.add_argument("--cands-per-step", type=int, default=256)
    p.add_argument("--epochs", type=int, default=25)
    p.add_argument("--grid-size", type=int, default=10_000)
    p.add_argument("--cpu", action="store_true",
                   help="debug: run the numpy tier instead of HIP")
    return p.parse_args()


def build_grid(n_points):
    """C x penalty grid, n_points candidates (BASELINE config 2)."""
    n_c = n_points // 2
    Cs = np.logspace(-3, 3, n_c)
    grid = []
    for pen in ("l2", "l1"):
        for c in Cs:
            grid.append({"C": float(c), "penalty": pen})
    return grid


def make_data(rows, features, seed=0):
    """Synthetic, rank-0 only; other ranks receive via RCCL broadcast."""
    rng = np.random.default_rng(seed)
    X = rng.standard_normal((rows, features), dtype=np.float32)
    w = rng.standard_normal(features, dtype=np.float32) / np.sqrt(features)
    logits = X @ w + 0.25 * rng.standard_normal(rows, dtype=np.float32)
    y = (logits > 0).astype(np.int64)
    return X, y


def main():
    args = parse_args()
    import torch

    dist = None
    rank, world = 0, 1
    if "RANK" in os.environ and int(os.environ.get("WORLD_SIZE", "1")) > 1:
        import torch.distributed as tdist

        dist = tdist
        backend = "nccl" if torch.cuda.is_available() else "gloo"
        local = int(os.environ.get("LOCAL_RANK", 0))
        if torch.cuda.is_available():
            torch.cuda.set_device(local % torch.cuda.device_count())
        dist.init_process_group(backend)
        rank, world = dist.get_rank(), dist.get_world_size()

    use_gpu = torch.cuda.is_available() and not args.cpu
    device = torch.device("cuda", torch.cuda.current_device()) if use_gpu \
        else torch.device("cpu")

    # ---- data: generate on rank 0, RCCL-broadcast into every GPU's HBM ----
    if rank == 0:
        X_np, y_np = make_data(args.rows, args.features)
    if dist is not None:
        if rank == 0:
            X_t = torch.as_tensor(X_np).to(device)
            y_t = torch.as_tensor(y_np.astype(np.float32)).to(device)
        else:
            X_t = torch.empty((args.rows, args.features),
                              dtype=torch.float32, device=device)
            y_t = torch.empty(args.rows, dtype=torch.float32, device=device)
        dist.broadcast(X_t, src=0)   # RC1: once, GB-scale, over xGMI
        dist.broadcast(y_t, src=0)
        X_host = X_t if use_gpu else X_t.numpy()
        y_host = y_t.cpu().numpy().astype(np.int64)
        del y_t
    else:
        X_host, y_host = X_np, y_np

    from sklearn.model_selection import KFold

    from spark_sklearn_amd.base_search import Task
    from spark_sklearn_amd.engine import run_batched_tasks
    from spark_sklearn_amd.models import LogisticRegression

    grid = build_grid(args.grid_size)
    est = LogisticRegression(max_iter=args.epochs, tol=0.0)

    n = args.rows
    splits = list(KFold(n_splits=args.cv).split(np.arange(n)))

    def make_step_tasks(step_idx):
        """This rank's candidate slice for one step (distinct per step)."""
        base = (step_idx * world + rank) * args.cands_per_step
        tasks = []
        t = 0
        for ci in range(args.cands_per_step):
            params = grid[(base + ci) % len(grid)]
            for fi, (tr, te) in enumerate(splits):
                tasks.append(Task(t, ci, fi, params, tr, te))
                t += 1
        return tasks

    def run_step(step_idx):
        tasks = make_step_tasks(step_idx)
        results = run_batched_tasks(
            est, X_host, y_host, tasks, scorer=None, fit_params=None,
            return_train_score=False, error_score="raise", verbose=0,
            device=device if use_gpu else None,
            column_batch=256)
        if dist is not None:
            # RC2: the GLOBAL fold-score table (world x local tasks rows);
            # each rank fills its own row block, one sum-all-reduce gives
            # every rank the full table -- the real search path's shape
            table = torch.zeros((world * len(tasks), 6),
                                dtype=torch.float64,
                                device=device if use_gpu else "cpu")
            base = rank * len(tasks)
            for r in results:
                table[base + r.index, 1] = r.test_score
            dist.all_reduce(table)
        return results

    def barrier_sync():
        if use_gpu:
            torch.cuda.synchronize()
        if dist is not None:
            dist.barrier()
            if use_gpu:
                torch.cuda.synchronize()

    for wstep in range(args.warmup):
        run_step(wstep)
    barrier_sync()

    t0 = time.perf_counter()
    for step in range(args.warmup, args.warmup + args.steps):
        run_step(step)
    barrier_sync()
    elapsed = time.perf_counter() - t0

    # MAX over ranks
    if dist is not None:
        e_t = torch.tensor([elapsed], dtype=torch.float64,
                           device=device if use_gpu else "cpu")
        dist.all_reduce(e_t, op=dist.ReduceOp.MAX)
        elapsed = float(e_t.item())

    fits_total = args.cands_per_step * args.cv * args.steps * world
    fits_per_sec = fits_total / elapsed
    if rank == 0:
        out = {
            "metric": "candidate_fits_per_sec",
            "value": round(fits_per_sec, 2),
            "unit": "candidate-fits/s",
            "n_gpus": world,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": round(1000.0 * elapsed / args.steps, 3),
            "higher_is_better": True,
            "scaling": "weak",
            "vs_baseline": None,
            "dtype": "bf16",
            "data": f"synthetic dense {args.rows}x{args.features}, "
                    f"random-init weights",
            "config": {
                "model": "LogisticRegression (full-batch GD, "
                         f"{args.epochs} epochs, OVR/binary)",
                "grid": "C x penalty, 10k points",
                "global_batch": args.cands_per_step * world,
                "rows": args.rows,
                "features": args.features,
                "cv": args.cv,
                "candidates_per_step_per_gpu": args.cands_per_step,
                "parallelism": f"task-dp{world}",
                "candidates_per_sec": round(fits_per_sec / args.cv, 2),
            },
        }
        print(json.dumps(out))
    if dist is not None:
        dist.destroy_process_group()


if __name__ == "__main__":
    sys.exit(main())
