"""Multicolor ILU(0) smoother/preconditioner.

Reference: src/solvers/multicolor_ilu_solver.cu (2,222 LoC, 8 kernels):
ILU(0) factorization in COLOR order with color-parallel triangular solves.
"k < i" in the elimination means color(k) < color(i); rows of one color are
eliminated in parallel (they cannot couple through a valid distance-1
coloring for the L/U pattern... coupling happens only via earlier colors;
same-color coupling is dropped exactly as the reference does).
"""

from __future__ import annotations

from .. import ops
from .base import register_solver
from .smoothers import _SmootherBase


def ilu0_setup(A, coloring):
    return ops._backend(A).ilu0_setup(A, coloring)


def ilu0_solve(A, factors, coloring, r, x, relaxation=1.0):
    return ops._backend(A).ilu0_solve(A, factors, coloring, r, x, relaxation)


def extended_sparsity(A, level: int):
    """Level-k fill pattern: pattern(A) grown k times by one product with
    pattern(A) (reference csr_sparsity_ilu1, include/csr_multiply.h:88, for
    k=1). Returns a CSRMatrix with A's values embedded and explicit zeros in
    the fill slots; halo entries (distributed) are carried over unchanged."""
    import numpy as np
    import scipy.sparse as sp

    from ..matrix import CSRMatrix
    n = A.n_rows
    ro = A.row_offsets.cpu().numpy().astype(np.int64)
    ci = A.col_indices.cpu().numpy().astype(np.int64)
    va = A.values.cpu().numpy().reshape(A.nnz, -1)[:, 0]
    rows = np.repeat(np.arange(n), np.diff(ro))
    loc = ci < n
    mloc = sp.csr_matrix((va[loc], (rows[loc], ci[loc])), shape=(n, n))
    pat = (mloc != 0).astype(np.int8)
    pat.setdiag(1)
    ext = pat.copy()
    for _ in range(max(level, 0)):
        ext = (ext @ pat + ext).tocsr()
        ext.data[:] = 1
    # embed A's local values at the union pattern (fill slots = explicit 0)
    coo = ext.tocoo()
    v = np.asarray(mloc[coo.row, coo.col]).ravel()
    out_local = sp.csr_matrix((v, (coo.row, coo.col)), shape=(n, n))
    out_local.sort_indices()
    if A.n_cols > n:        # append halo entries back per row
        hr, hc, hv = rows[~loc], ci[~loc], va[~loc]
        halo = sp.csr_matrix((hv, (hr, hc - n)),
                             shape=(n, A.n_cols - n))
        full = sp.hstack([out_local, halo]).tocsr()
    else:
        full = out_local
    M = CSRMatrix.from_scipy(full, dtype=A.dtype)
    # from_scipy drops nothing but sums dups; explicit zeros preserved
    return M.to(A.row_offsets.device)


@register_solver("MULTICOLOR_ILU")
class MulticolorILUSolver(_SmootherBase):
    def __init__(self, scope, resources):
        super().__init__(scope, resources)
        self.sparsity_level = int(scope.get("ilu_sparsity_level") or 0)

    def solver_setup(self):
        A = self.A
        if self.sparsity_level > 0 and A.block_dim != 1:
            raise NotImplementedError(
                "ILU(k>0) fill patterns are scalar-only (block ILU(0) is "
                "supported on host modes)")
        if self.sparsity_level > 0:
            # ILU(k): factor on the extended pattern; coloring computed on
            # that pattern so same-color rows stay decoupled (reference
            # ilu1_coloringA semantics)
            self.A_f = extended_sparsity(A, self.sparsity_level)
            from ..amg.coloring import MatrixColoring
            self.A_f.coloring = MatrixColoring.create(self.A_f, self.scope)
        else:
            self.A_f = A
            if A.coloring is None:
                from ..amg.coloring import MatrixColoring
                A.coloring = MatrixColoring.create(A, self.scope)
        self.factors = ilu0_setup(self.A_f, self.A_f.coloring)

    def solve_iteration(self, b, x):
        r = ops.residual(self.A, x, b)
        ilu0_solve(self.A_f, self.factors, self.A_f.coloring, r, x,
                   self.relaxation_factor)
        return False
