"""CSR / block-CSR matrix container on torch tensors.

Reimplements the capability of the reference ``Matrix``/``MatrixBase``
(include/matrix.h:64,988,1150): block-CSR storage with optional separate
diagonal, coloring attachment, and views tied to a DistributedManager.
MI355X-native differences:

* storage is plain torch tensors (int32 offsets/indices, fp64/fp32 values) so
  the same container serves host (CPU tests) and device (HIP kernels);
* blocks are stored row-major as a ``(nnz, b, b)`` values tensor;
* there is no COO dual representation — upload converts once to CSR;
* the "DIAG property" (external diagonal, reference include/matrix.h:24-26) is
  an optional ``diag`` tensor of ``(n, b, b)`` blocks; kernels take it when
  present.
"""

from __future__ import annotations

from typing import Optional

import numpy as np
import torch


class CSRMatrix:
    """Square (or rectangular) sparse matrix in CSR / block-CSR form."""

    def __init__(self, row_offsets: torch.Tensor, col_indices: torch.Tensor,
                 values: torch.Tensor, n_cols: Optional[int] = None,
                 block_dim: int = 1, diag: Optional[torch.Tensor] = None):
        assert row_offsets.dtype == torch.int32
        assert col_indices.dtype == torch.int32
        self.row_offsets = row_offsets
        self.col_indices = col_indices
        self.values = values
        self.block_dim = block_dim
        self.n_rows = int(row_offsets.numel() - 1)
        self.n_cols = int(n_cols) if n_cols is not None else self.n_rows
        self.diag = diag  # external diagonal blocks (n_rows, b, b) or None
        # lazily computed attachments
        self._diag_idx: Optional[torch.Tensor] = None   # position of diagonal in each row
        self.coloring = None                             # MatrixColoring
        self.manager = None                              # DistributedManager
        self._cache: dict = {}

    # ------------------------------------------------------------------ properties
    @property
    def nnz(self) -> int:
        return int(self.col_indices.numel())

    @property
    def device(self) -> torch.device:
        return self.values.device

    @property
    def dtype(self) -> torch.dtype:
        return self.values.dtype

    @property
    def is_cuda(self) -> bool:
        return self.values.is_cuda

    @property
    def shape(self):
        b = self.block_dim
        return (self.n_rows * b, self.n_cols * b)

    def clear_cache(self):
        self._cache.clear()
        self._diag_idx = None

    # ------------------------------------------------------------------ builders
    @classmethod
    def from_coo(cls, rows, cols, vals, n_rows: int, n_cols: Optional[int] = None,
                 device="cpu", dtype=torch.float64, block_dim: int = 1) -> "CSRMatrix":
        """Build (summing duplicates) from COO triples on host, then move."""
        import scipy.sparse as sp
        n_cols = n_cols if n_cols is not None else n_rows
        if block_dim == 1:
            m = sp.coo_matrix((np.asarray(vals, dtype=np.float64),
                               (np.asarray(rows), np.asarray(cols))),
                              shape=(n_rows, n_cols)).tocsr()
            m.sum_duplicates()
            return cls.from_scipy(m, device=device, dtype=dtype)
        raise NotImplementedError("block COO upload: use from_bsr")

    @classmethod
    def from_scipy(cls, m, device="cpu", dtype=torch.float64) -> "CSRMatrix":
        m = m.tocsr()
        m.sum_duplicates()     # canonicalize: kron/add chains leave duplicate
        m.sort_indices()       # (i,j) entries that break per-entry ops
        ro = torch.from_numpy(np.ascontiguousarray(m.indptr, dtype=np.int32))
        ci = torch.from_numpy(np.ascontiguousarray(m.indices, dtype=np.int32))
        v = torch.from_numpy(np.ascontiguousarray(m.data)).to(dtype)
        out = cls(ro.to(device), ci.to(device), v.to(device), n_cols=m.shape[1])
        return out

    @classmethod
    def from_bsr(cls, row_offsets, col_indices, block_values, n_cols=None,
                 device="cpu", dtype=torch.float64) -> "CSRMatrix":
        """block_values: (nnz, b, b) row-major blocks."""
        ro = torch.as_tensor(row_offsets, dtype=torch.int32)
        ci = torch.as_tensor(col_indices, dtype=torch.int32)
        bv = torch.as_tensor(block_values).to(dtype)
        assert bv.dim() == 3 and bv.shape[1] == bv.shape[2]
        out = cls(ro.to(device), ci.to(device), bv.to(device).contiguous(),
                  n_cols=n_cols, block_dim=int(bv.shape[1]))
        return out

    def to(self, device) -> "CSRMatrix":
        device = torch.device(device)
        if device == self.device:
            return self
        out = CSRMatrix(self.row_offsets.to(device), self.col_indices.to(device),
                        self.values.to(device), n_cols=self.n_cols,
                        block_dim=self.block_dim,
                        diag=self.diag.to(device) if self.diag is not None else None)
        out.manager = self.manager
        return out

    # ------------------------------------------------------------------ scipy view
    def to_scipy(self):
        """Host scipy.sparse.csr_matrix view (scalar) or bsr_matrix (block).

        Used by the CPU reference backend and by tests."""
        key = "scipy"
        if key in self._cache:
            return self._cache[key]
        import scipy.sparse as sp
        ro = self.row_offsets.cpu().numpy()
        ci = self.col_indices.cpu().numpy()
        v = self.values.detach().cpu().numpy()
        if self.block_dim == 1:
            m = sp.csr_matrix((v, ci, ro), shape=(self.n_rows, self.n_cols))
        else:
            m = sp.bsr_matrix((v, ci, ro),
                              shape=(self.n_rows * self.block_dim,
                                     self.n_cols * self.block_dim))
        if self.diag is not None:
            d = self.diag.detach().cpu().numpy()
            if self.block_dim == 1:
                m = (m + sp.diags(d.reshape(-1))).tocsr()
            else:
                b = self.block_dim
                dm = sp.block_diag([d[i] for i in range(self.n_rows)], format="bsr")
                m = (m + dm).tobsr(blocksize=(b, b))
        self._cache[key] = m
        return m

    # ------------------------------------------------------------------ diagonal ops
    def diag_index(self) -> torch.Tensor:
        """Index into values of each row's diagonal entry (reference:
        MatrixBase::computeDiagonal, include/matrix.h). -1 where missing."""
        if self._diag_idx is not None:
            return self._diag_idx
        from . import ops
        self._diag_idx = ops.compute_diag_index(self)
        return self._diag_idx

    def diagonal(self) -> torch.Tensor:
        """Dense diagonal: (n,) for scalar, (n, b, b) for block matrices."""
        key = "diagonal"
        if key in self._cache:
            return self._cache[key]
        if self.diag is not None:
            d = self.diag
        else:
            from . import ops
            d = ops.extract_diagonal(self)
        self._cache[key] = d
        return d

    def replace_coefficients(self, values: torch.Tensor,
                             diag: Optional[torch.Tensor] = None):
        """Swap numeric values keeping structure (reference:
        AMGX_matrix_replace_coefficients, include/amgx_c.h:603)."""
        assert values.shape == self.values.shape
        self.values = values.to(self.values.dtype).to(self.device)
        if diag is not None:
            self.diag = diag.to(self.values.dtype).to(self.device)
        self._cache.clear()

    def __repr__(self) -> str:  # pragma: no cover
        return (f"CSRMatrix(n={self.n_rows}x{self.n_cols}, nnz={self.nnz}, "
                f"b={self.block_dim}, dtype={self.dtype}, dev={self.device})")
