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


@register_solver("MULTICOLOR_ILU")
class MulticolorILUSolver(_SmootherBase):
    def __init__(self, scope, resources):
        super().__init__(scope, resources)
        if not scope.has("relaxation_factor"):
            self.relaxation_factor = 1.0
        self.sparsity_level = scope.get("ilu_sparsity_level")

    def solver_setup(self):
        A = self.A
        if A.coloring is None:
            from ..amg.coloring import MatrixColoring
            A.coloring = MatrixColoring.create(A, self.scope)
        self.factors = ilu0_setup(A, A.coloring)

    def solve_iteration(self, b, x):
        r = ops.residual(self.A, x, b)
        ilu0_solve(self.A, self.factors, self.A.coloring, r, x,
                   self.relaxation_factor)
        return False
