"""Kaczmarz and CF-Jacobi smoothers.

Reference: src/solvers/kaczmarz_solver.cu (multicolor/naive row-projection
Kaczmarz), src/solvers/cf_jacobi_solver.cu (coarse/fine-split Jacobi for the
classical path; cf_smoothing_mode src/core.cu:416).
"""

from __future__ import annotations

import torch

from .. import ops
from ..config import register_parameter
from .base import register_solver
from .smoothers import _SmootherBase

register_parameter("cf_smoothing_mode", int, 0,
                   "0: F then C sweeps; 1: C then F")


@register_solver("KACZMARZ")
class KaczmarzSolver(_SmootherBase):
    """Row-projection sweeps x += omega*(b_i - a_i.x)/||a_i||^2 * a_i,
    scheduled by color (same-color rows may still share columns: the scatter
    is accumulated atomically on device, sequentially on host)."""

    def __init__(self, scope, resources):
        super().__init__(scope, resources)
        if not scope.has("relaxation_factor"):
            self.relaxation_factor = 1.0

    def solver_setup(self):
        A = self.A
        if A.coloring is None:
            from ..amg.coloring import MatrixColoring
            A.coloring = MatrixColoring.create(A, self.scope)
        # row squared norms
        if A.block_dim != 1:
            raise NotImplementedError("Kaczmarz: scalar matrices")
        v2 = A.values.double() ** 2
        ro = A.row_offsets.to(torch.int64)
        seg = torch.zeros(A.n_rows, dtype=torch.float64, device=A.device)
        rows = torch.repeat_interleave(
            torch.arange(A.n_rows, device=A.device), ro[1:] - ro[:-1])
        seg.index_add_(0, rows, v2)
        seg = torch.where(seg > 0, seg, torch.ones_like(seg))
        self.row_norm_inv = (1.0 / seg).to(A.dtype)

    def solve_iteration(self, b, x):
        A = self.A
        col = A.coloring
        mgr = getattr(A, "manager", None)
        if mgr is not None and mgr.neighbors:
            mgr.exchange_halo(x)
        backend = ops._backend(A)
        if hasattr(backend, "kaczmarz_rows"):
            for c in range(col.num_colors):
                backend.kaczmarz_rows(A, self.row_norm_inv, b, x,
                                      col.rows_of(c), self.relaxation_factor)
        else:
            self._host_sweep(b, x)
        return False

    def _host_sweep(self, b, x):
        import numpy as np
        A = self.A
        m = A.to_scipy().tocsr()
        xv = x.reshape(-1).numpy()
        bv = b.reshape(-1).numpy()
        rni = self.row_norm_inv.numpy()
        for c in range(A.coloring.num_colors):
            for i in A.coloring.rows_of(c).numpy():
                s, e = m.indptr[i], m.indptr[i + 1]
                resid = bv[i] - m.data[s:e] @ xv[m.indices[s:e]]
                xv[m.indices[s:e]] += (self.relaxation_factor * resid
                                       * rni[i]) * m.data[s:e]


@register_solver("CF_JACOBI")
class CFJacobiSolver(_SmootherBase):
    """Coarse/fine-split Jacobi: a full fused Jacobi candidate is computed,
    then only the F rows (then only the C rows) are accepted — proper Jacobi
    semantics inside each class (reference src/solvers/cf_jacobi_solver.cu)."""

    def __init__(self, scope, resources):
        super().__init__(scope, resources)
        self.mode = scope.get("cf_smoothing_mode")
        if not scope.has("relaxation_factor"):
            self.relaxation_factor = 0.9

    def solver_setup(self):
        A = self.A
        self.dinv = ops.jacobi_dinv(A)
        cf = A._cache.get("cf_map")
        if cf is None:
            # no classical split attached: behave as plain Jacobi
            self.f_rows = self.c_rows = None
            return
        cf = cf.to(A.device)
        self.f_rows = (cf < 0).nonzero(as_tuple=True)[0]
        self.c_rows = (cf >= 0).nonzero(as_tuple=True)[0]

    def solve_iteration(self, b, x):
        sc = self._get_scratch(x)
        if self.f_rows is None:
            ops.jacobi_smooth(self.A, self.dinv, b, x, sc,
                              self.relaxation_factor)
            x.reshape(-1).copy_(sc)
            return False
        phases = (self.f_rows, self.c_rows) if self.mode == 0 \
            else (self.c_rows, self.f_rows)
        for rows in phases:
            ops.jacobi_smooth(self.A, self.dinv, b, x, sc,
                              self.relaxation_factor)
            x.reshape(-1).index_copy_(0, rows, sc.index_select(0, rows))
        return False
