"""Multicolor DILU smoother — the reference's flagship block smoother
(src/solvers/multicolor_dilu_solver.cu, 14 kernels).

DILU preconditioner M = (E + L) E^{-1} (E + U) with L/U the strict
lower/upper parts in COLOR order and E chosen so diag(M) = diag-modified:

    E_i = A_ii - sum_{color(j) < color(i)} A_ij Einv_j A_ji

Applying M^{-1} r: forward substitution by ascending color
(w_i = Einv_i (r_i - sum_{color(j)<color(i)} A_ij w_j)), then backward by
descending color (z_i = w_i - Einv_i sum_{color(j)>color(i)} A_ij z_j).
A valid distance-1 coloring guarantees same-color entries are only the
diagonal, which is what makes each color sweep embarrassingly parallel.
"""

from __future__ import annotations

from .. import ops
from .base import register_solver
from .smoothers import _SmootherBase


def dilu_setup(A, coloring):
    return ops._backend(A).dilu_setup(A, coloring)


def dilu_solve(A, Einv, coloring, r, relaxation, x):
    """x += relaxation * M^{-1} r (forward+backward sweeps)."""
    return ops._backend(A).dilu_solve(A, Einv, coloring, r, relaxation, x)


@register_solver("MULTICOLOR_DILU")
class MulticolorDILUSolver(_SmootherBase):
    # relaxation_factor comes from the registry default (0.9), matching the
    # reference (src/core.cu:398): undamped DILU can diverge on the dense
    # rows of D2/aggressive coarse operators

    def solver_setup(self):
        A = self.A
        if A.coloring is None:
            from ..amg.coloring import MatrixColoring
            A.coloring = MatrixColoring.create(A, self.scope)
        self.Einv = dilu_setup(A, A.coloring)

    def solve_iteration(self, b, x):
        B = ops._backend(self.A)
        fused = getattr(B, "dilu_smooth", None)
        if fused is not None and fused(self.A, self.Einv, self.A.coloring,
                                       b, x, self.relaxation_factor):
            return False
        r = ops.residual(self.A, x, b)
        dilu_solve(self.A, self.Einv, self.A.coloring, r,
                   self.relaxation_factor, x)
        return False
