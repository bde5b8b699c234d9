"""AMG as a Solver (reference src/solvers/algebraic_multigrid_solver.cu:49-58):
wraps an AMGHierarchy; one solve_iteration = one cycle. Used directly or as a
preconditioner/smoother of an outer Krylov solver."""

from __future__ import annotations

from .base import Solver, register_solver


@register_solver("AMG")
class AMGSolver(Solver):
    is_smoother = True

    def solver_setup(self):
        from ..amg.amg import AMGHierarchy
        self.hierarchy = AMGHierarchy(self.scope, self.res)
        self.hierarchy.setup(self.A)

    def solve_iteration(self, b, x):
        self.hierarchy.cycle(b, x, zero_initial_guess=self._first)
        self._first = False
        return False

    def solve_init(self, b, x, zero_initial_guess):
        self._first = zero_initial_guess

    def solve(self, b, x, zero_initial_guess=False):
        # cheap path: no residual monitoring => run max_iters cycles directly
        if not self.monitor_residual:
            self._first = zero_initial_guess
            for _ in range(self.max_iters):
                self.solve_iteration(b, x)
            st = self.status
            st.status = st.SUCCESS
            st.iterations = self.max_iters
            return st
        return super().solve(b, x, zero_initial_guess)
