"""Sparse-row containers: host scipy CSR <-> device CSR in HBM.

Behavioral counterpart of the reference's ``CSRVectorUDT`` (``udt.py``
~L1-80 [R]), which stores single-row ``scipy.sparse
.csr_matrix`` vectors inside Spark DataFrame columns via a UserDefinedType
with (indices, values, size) fields.  Without a Spark runtime the two jobs
that type performed are:

* a stable serialized record format for one sparse row ->
  :class:`CSRVectorUDT` (serialize/deserialize dicts, usable as pandas
  column values, e.g. as the ``xCol`` of :class:`KeyedEstimator`);
* sparse matrices resident on the GPU for the LDS-staged sparse kernels ->
  :class:`DeviceCSRMatrix` (indptr/indices/data triples in HBM).
"""

from __future__ import annotations

import numpy as np
import scipy.sparse as sp

__all__ = ["CSRVectorUDT", "DeviceCSRMatrix", "stack_csr_rows"]


class CSRVectorUDT:
    """Serialize/deserialize a single-row scipy CSR vector.

    Record layout mirrors the reference's sqlType struct: ``indices``
    (int32), ``values`` (float64), ``size`` (int) -- SURVEY.md C8.
    """

    @staticmethod
    def serialize(row) -> dict:
        row = sp.csr_matrix(row)
        if row.shape[0] != 1:
            raise ValueError(
                f"CSRVectorUDT stores single-row vectors, got shape "
                f"{row.shape}")
        return {
            "indices": np.asarray(row.indices, dtype=np.int32),
            "values": np.asarray(row.data, dtype=np.float64),
            "size": int(row.shape[1]),
        }

    @staticmethod
    def deserialize(record: dict) -> sp.csr_matrix:
        size = int(record["size"])
        indices = np.asarray(record["indices"], dtype=np.int32)
        values = np.asarray(record["values"], dtype=np.float64)
        indptr = np.array([0, len(indices)], dtype=np.int32)
        return sp.csr_matrix((values, indices, indptr), shape=(1, size))


def stack_csr_rows(rows) -> sp.csr_matrix:
    """Vertically stack serialized rows / 1-row CSR matrices / dense rows."""
    mats = []
    for r in rows:
        if isinstance(r, dict):
            mats.append(CSRVectorUDT.deserialize(r))
        elif sp.issparse(r):
            mats.append(sp.csr_matrix(r))
        else:
            mats.append(sp.csr_matrix(np.asarray(r).reshape(1, -1)))
    return sp.vstack(mats, format="csr")


class DeviceCSRMatrix:
    """CSR matrix resident in one GPU's HBM (indptr/indices/data triples).

    Feeds the LDS-staged sparse kernels (K5).
    ``data`` is stored bf16 (compute dtype) with fp32 accumulation inside
    the kernels; ``indices``/``indptr`` are int32.
    """

    def __init__(self, indptr, indices, data, shape, device=None):
        import torch

        dev = device if device is not None else (
            torch.device("cuda", 0) if torch.cuda.is_available() else "cpu")
        self.indptr = torch.as_tensor(indptr, dtype=torch.int32).to(dev)
        self.indices = torch.as_tensor(indices, dtype=torch.int32).to(dev)
        self.data = torch.as_tensor(
            np.asarray(data, dtype=np.float32)).to(dev).to(torch.bfloat16)
        self.shape = tuple(shape)
        self.device = self.data.device

    @classmethod
    def from_scipy(cls, mat: sp.spmatrix, device=None) -> "DeviceCSRMatrix":
        mat = sp.csr_matrix(mat)
        mat.sort_indices()
        return cls(mat.indptr, mat.indices, mat.data, mat.shape, device)

    def to_scipy(self) -> sp.csr_matrix:
        return sp.csr_matrix(
            (self.data.float().cpu().numpy(),
             self.indices.cpu().numpy(),
             self.indptr.cpu().numpy()),
            shape=self.shape)

    @property
    def nnz(self) -> int:
        return int(self.indices.numel())

    def __repr__(self):
        return (f"DeviceCSRMatrix(shape={self.shape}, nnz={self.nnz}, "
                f"device={self.device})")
