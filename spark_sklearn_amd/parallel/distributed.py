"""torch.distributed backend: one process per GPU over RCCL/xGMI.

Replaces the reference's Spark runtime with the
MI355X-native communication plan of SURVEY.md section 5.8:

* RC1 -- the dataset is broadcast ONCE from rank 0 into every GPU's 288 GB
  HBM (``dist.broadcast`` -> rcclBroadcast over xGMI; tree/direct p2p fans
  out over the root's 7 links).  This replaces Spark's TorrentBroadcast.
* task dispatch -- needs no network at all on one node: tasks are
  partitioned statically, candidate-round-robin, so every rank derives its
  share from the task list deterministically (replaces the Spark
  scheduler).  ``partition="dynamic"`` switches to a bounded work queue:
  whole-candidate blocks pulled from the rendezvous store's atomic
  counter, balancing skewed per-candidate costs (round 2).
* RC2 -- fold scores are fixed-width records summed into a dense score
  table with one all-reduce (each rank fills only its rows; zeros
  elsewhere), so every rank assembles identical ``cv_results_``; an
  object all-gather carries the error TEXT of failed fits so
  FitFailedWarning keeps its diagnostic payload on every rank (round 2).
  This replaces ``collect()``.
* refit -- rank 0 refits on the full resident matrix (it already holds all
  of X after RC1; RC4 in SURVEY.md needs no collective) and broadcasts the
  fitted parameters so every rank's ``best_estimator_`` is usable.

On CPU (backend "gloo") the same code path runs with sklearn host fits --
this is the multi-process CI tier (world_size=2, no GPU).
"""

from __future__ import annotations

import pickle
from typing import List, Optional

import numpy as np

from ..base_search import Task, TaskResult
from .backend import Backend, execute_local_tasks

__all__ = ["DistributedBackend"]


class DistributedBackend(Backend):
    """Task-parallel search over an initialized torch.distributed group.

    Parameters
    ----------
    device : torch.device or None
        This rank's device.  ``None`` auto-selects ``cuda:LOCAL_RANK`` when
        a GPU is visible, else CPU (gloo tier).
    n_threads : int or None
        Thread-pool width for host fits on each rank.
    """

    def __init__(self, device=None, n_threads: Optional[int] = None,
                 partition: str = "static", dynamic_block: int = 8):
        import torch
        import torch.distributed as dist

        if not dist.is_initialized():
            raise RuntimeError(
                "DistributedBackend requires torch.distributed to be "
                "initialized (launch with torchrun / init_process_group)")
        if partition not in ("static", "dynamic"):
            raise ValueError("partition must be 'static' or 'dynamic'")
        self.rank = dist.get_rank()
        self.world_size = dist.get_world_size()
        self.n_threads = n_threads
        self.partition = partition
        self.dynamic_block = int(dynamic_block)
        self._fit_seq = 0
        if device is None and torch.cuda.is_available():
            import os

            local = int(os.environ.get("LOCAL_RANK", self.rank))
            device = torch.device("cuda", local % torch.cuda.device_count())
        if device is not None and getattr(device, "type", None) == "cuda":
            # pin the rank's device so nccl collectives (incl. the object
            # gathers) use the right GPU instead of everyone's cuda:0
            torch.cuda.set_device(device)
        self.device = device
        self._torch = torch
        self._dist = dist

    # -- RC1: dataset broadcast -------------------------------------------
    def broadcast_data(self, X, y):
        """Broadcast (X, y) from rank 0 to all ranks.

        Ranks other than 0 may pass ``X=None``; they receive rank 0's
        arrays.  GPU path: tensors land directly in HBM and stay resident
        for the whole search.  Numpy-in, numpy-out on the gloo tier.
        """
        torch, dist = self._torch, self._dist
        comm_dev = self.device if (self.device is not None and
                                   self.device.type == "cuda") else "cpu"

        def _bcast_array(arr):
            # shape/dtype header first (tiny object bcast), then payload
            if self.rank == 0:
                t = torch.as_tensor(np.ascontiguousarray(arr)).to(comm_dev)
                header = [tuple(t.shape), str(t.dtype)]
            else:
                header = [None, None]
            dist.broadcast_object_list(header, src=0)
            if self.rank != 0:
                dtype = getattr(torch, header[1].replace("torch.", ""))
                t = torch.empty(header[0], dtype=dtype, device=comm_dev)
            dist.broadcast(t, src=0)
            return t

        X_t = _bcast_array(X) if (self.rank == 0 and X is not None) or \
            self.rank != 0 else None
        if X_t is None:
            raise ValueError("rank 0 must provide X")
        has_y = [y is not None if self.rank == 0 else None]
        dist.broadcast_object_list(has_y, src=0)
        y_t = _bcast_array(y) if has_y[0] else None
        return X_t, y_t

    # -- partition + execute + RC2 gather ---------------------------------
    def local_task_filter(self, tasks: List[Task]) -> List[Task]:
        """Candidate-round-robin partition: rank r owns candidates with
        ``cand_index % world_size == r``, keeping all folds of a candidate
        on one rank so batched kernels see whole candidates."""
        return [t for t in tasks if t.cand_index % self.world_size == self.rank]

    def _run_dynamic(self, estimator, X, y, tasks, exec_kwargs):
        """Bounded dynamic partition: whole-candidate blocks are pulled
        from a shared atomic counter (the process group's rendezvous
        store -- works on gloo AND nccl) as each rank finishes its
        previous block, so skewed per-candidate costs balance.  Trades
        cross-candidate kernel batching width for balance; block size is
        the bound (HipContext(dynamic_block=...))."""
        dist = self._dist
        self._fit_seq += 1
        key = f"ska_dynq_{self._fit_seq}"
        store = dist.distributed_c10d._get_default_store()
        by_cand = {}
        order = []
        for t in tasks:
            if t.cand_index not in by_cand:
                by_cand[t.cand_index] = []
                order.append(t.cand_index)
            by_cand[t.cand_index].append(t)
        blk = max(1, self.dynamic_block)
        nblocks = (len(order) + blk - 1) // blk
        results = []
        pulled = 0
        while True:
            i = int(store.add(key, 1)) - 1
            if i >= nblocks:
                break
            pulled += 1
            block = []
            for c in order[i * blk:(i + 1) * blk]:
                block.extend(by_cand[c])
            results.extend(execute_local_tasks(
                estimator, X, y, block, **exec_kwargs))
        self.n_blocks_pulled_ = pulled
        return results

    def run_search(self, estimator, X, y, tasks, *, scorer, fit_params,
                   return_train_score, error_score, verbose):
        torch, dist = self._torch, self._dist
        exec_kwargs = dict(
            scorer=scorer, fit_params=fit_params,
            return_train_score=return_train_score, error_score=error_score,
            verbose=verbose, n_threads=self.n_threads, device=self.device,
            progress=getattr(self, "progress", None),
            column_batch=getattr(self, "candidate_batch", None))
        if self.partition == "dynamic":
            if verbose > 0 and self.rank == 0:
                print(f"[dist] {self.world_size} ranks, {len(tasks)} tasks "
                      f"(dynamic blocks of {self.dynamic_block} candidates)")
            results = self._run_dynamic(estimator, X, y, tasks, exec_kwargs)
        else:
            local = self.local_task_filter(tasks)
            if verbose > 0 and self.rank == 0:
                print(f"[dist] {self.world_size} ranks, "
                      f"{len(tasks)} tasks (~{len(local)} per rank)")
            results = execute_local_tasks(estimator, X, y, local,
                                          **exec_kwargs)

        # RC2: dense float64 score table, one sum-all-reduce.  Single
        # metric: (n_tasks, 7) fixed-width records.  Multimetric: the
        # test/train columns widen to M metrics each; the metric-name
        # order is agreed across ranks first (sorted keys of any rank's
        # dict-valued scores -- ranks whose fits all failed have none).
        comm_dev = self.device if (self.device is not None and
                                   self.device.type == "cuda" and
                                   dist.get_backend() == "nccl") else "cpu"
        local_keys = next(
            (sorted(r.test_score) for r in results
             if isinstance(r.test_score, dict)), None)
        gathered = [None] * self.world_size
        dist.all_gather_object(gathered, local_keys)
        keys = next((k for k in gathered if k), None)

        def _tvals(v):
            if keys is None:
                return [v]
            if isinstance(v, dict):
                return [v[k] for k in keys]
            return [v] * len(keys)     # failed fit: scalar error_score

        M = 1 if keys is None else len(keys)
        # rows are POSITIONS in the task list passed in (task indices are
        # global and may be sparse, e.g. under score-log resume)
        row_of = {t.index: i for i, t in enumerate(tasks)}
        table = torch.zeros((len(tasks), 5 + 2 * M), dtype=torch.float64,
                            device=comm_dev)
        for r in results:
            table[row_of[r.index]] = torch.tensor(
                [r.index] + _tvals(r.test_score) + _tvals(r.train_score) +
                [r.n_test, r.fit_time, r.score_time,
                 1.0 if r.error is not None else 0.0],
                dtype=torch.float64)
        dist.all_reduce(table, op=dist.ReduceOp.SUM)
        table = table.cpu().numpy()
        local_err = {r.index: str(r.error) for r in results
                     if r.error is not None}
        # gather real error text from every rank so the FitFailedWarning
        # keeps its diagnostic payload (round-1 gap: remote failures
        # surfaced as 'error text not gathered')
        err_shards = [None] * self.world_size
        dist.all_gather_object(err_shards, local_err)
        all_err = {}
        for shard in err_shards:
            if shard:
                all_err.update(shard)

        def _unpack(row, lo, failed):
            if keys is None or failed:
                return row[lo]         # scalar (error_score when failed)
            return dict(zip(keys, row[lo:lo + M]))

        out = []
        for row in table:
            failed = row[4 + 2 * M] > 0
            out.append(TaskResult(
                int(row[0]), _unpack(row, 1, failed),
                _unpack(row, 1 + M, failed), int(row[1 + 2 * M]),
                row[2 + 2 * M], row[3 + 2 * M],
                error=(all_err.get(int(row[0]),
                       "fit failed on a remote rank "
                       "(error text not gathered)") if failed else None)))
        return out

    # -- rank-0 refit + parameter broadcast --------------------------------
    def refit(self, estimator, X, y, fit_params):
        dist = self._dist
        if self.rank == 0:
            if y is not None:
                estimator.fit(X, y, **(fit_params or {}))
            else:
                estimator.fit(X, **(fit_params or {}))
            payload = [pickle.dumps(estimator)]
        else:
            payload = [None]
        dist.broadcast_object_list(payload, src=0)
        if self.rank != 0:
            estimator = pickle.loads(payload[0])
        return estimator
