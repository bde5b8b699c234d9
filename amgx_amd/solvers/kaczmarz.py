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
    scheduled by a DISTANCE-2 coloring so same-color rows share no columns —
    each color sweep is one race-free gather/segmented-reduce/scatter-add
    batch on the device."""

    def __init__(self, scope, resources):
        super().__init__(scope, resources)
        if not scope.has("relaxation_factor"):
            self.relaxation_factor = 1.0
        # symmetric color schedule (ascending then descending), same knob
        # as symmetric_GS. Note: the sweep is l2-symmetric, not A-adjoint,
        # so a Kaczmarz-smoothed AMG preconditioner still needs a flexible
        # Krylov method (FGMRES/PCGF), not plain PCG
        self.symmetric = bool(scope.get("symmetric_GS"))

    def solver_setup(self):
        A = self.A
        # Kaczmarz scatters into the COLUMNS of each row: parallel same-color
        # updates need rows that share no column, i.e. a DISTANCE-2 coloring
        # (reference: kaczmarz pairs with LOCALLY_DOWNWIND / 2-ring
        # colorings). Cached separately from the matrix's distance-1 one.
        self._kz_coloring = A._cache.get("kz_coloring")
        if self._kz_coloring is None:
            from ..amg.coloring import MatrixColoring
            from ..config import ConfigScope
            scheme = self.scope.get("matrix_coloring_scheme") \
                if self.scope.has("matrix_coloring_scheme") \
                else "LOCALLY_DOWNWIND"
            self._kz_coloring = MatrixColoring.create(
                A, ConfigScope(None, {"matrix_coloring_scheme": scheme,
                                      "coloring_level": 2}))
            A._cache["kz_coloring"] = self._kz_coloring
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
        col = self._kz_coloring
        mgr = getattr(A, "manager", None)
        if mgr is not None and mgr.neighbors:
            mgr.exchange_halo(x)
        # per color: r_rows = b - (Ax)_rows; x[cols of row] += w*r*rni*a —
        # race-free under the distance-2 coloring, device-capable via
        # gather/scatter_add (torch index ops drive the HIP kernels)
        ro = A.row_offsets.to(torch.int64)
        ci = A.col_indices.to(torch.int64)
        va = A.values.reshape(-1)
        xv = x.reshape(-1)
        bv = b.reshape(-1)
        order = list(range(col.num_colors))
        if self.symmetric:
            order = order + order[::-1]
        for c in order:
            rows = col.rows_of(c).to(torch.int64)
            if rows.numel() == 0:
                continue
            starts = ro[rows]
            counts = ro[rows + 1] - starts
            total = int(counts.sum().item())
            if total == 0:
                continue
            pos = (torch.repeat_interleave(starts, counts)
                   + torch.arange(total, device=xv.device, dtype=torch.int64)
                   - torch.repeat_interleave(
                       torch.cumsum(counts, 0) - counts, counts))
            cols_c = ci[pos]
            vals_c = va[pos]
            prod = vals_c * xv[cols_c]
            seg = torch.repeat_interleave(
                torch.arange(rows.numel(), device=xv.device), counts)
            ax = torch.zeros(rows.numel(), dtype=prod.dtype,
                             device=xv.device)
            ax.index_add_(0, seg, prod)
            coef = (self.relaxation_factor * (bv[rows] - ax)
                    * self.row_norm_inv[rows])
            xv.index_add_(0, cols_c, coef[seg] * vals_c)
        return False


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
