"""AMG as a Solver (reference src/solvers/algebraic_multigrid_solver.cu:49-58):
wraps an AMGHierarchy; one solve_iteration = one cycle. Used directly or as a
preconditioner/smoother of an outer Krylov solver.

MI355X-native addition: when running on device, single-process, the whole
cycle (smoother sweeps, residual, transfers, coarse GEMV — O(100) kernel
launches) is captured into a hipGraph once and replayed per application,
eliminating per-launch overhead on the launch-bound coarse levels. The
capture is VALIDATED against one eager cycle on the same input and disabled
on any mismatch or capture failure (eager fallback), so correctness never
depends on graph support.
"""

from __future__ import annotations

import torch

from ..config import register_parameter
from .base import Solver, register_solver

register_parameter("use_hip_graph", int, 1,
                   "capture the AMG cycle into a hipGraph (device path)")


@register_solver("AMG")
class AMGSolver(Solver):
    is_smoother = True

    def __init__(self, scope, resources):
        super().__init__(scope, resources)
        self.use_graph = bool(scope.get("use_hip_graph"))
        self._graph = None
        self._graph_failed = False

    def solver_setup(self):
        from ..amg.amg import AMGHierarchy
        reuse = int(self.scope.get("structure_reuse_levels") or 0)
        old = getattr(self, "hierarchy", None)
        if (reuse != 0 and old is not None and old.levels
                and old.levels[0].A.n_rows == self.A.n_rows
                and old.levels[0].A.nnz == self.A.nnz):
            # values-only re-setup on the cached structure (reference
            # structure_reuse_levels, src/core.cu:437 + amg.cu reuse path)
            old.resetup(self.A, reuse)
        else:
            self.hierarchy = AMGHierarchy(self.scope, self.res)
            self.hierarchy.setup(self.A)
        self._graph = None
        self._graph_failed = False

    # ------------------------------------------------------------- graph path
    def _graph_eligible(self) -> bool:
        return (self.use_graph and not self._graph_failed
                and self.res.is_cuda
                and getattr(self.A, "manager", None) is None)

    def _capture_graph(self, b):
        g_in = torch.zeros_like(b.reshape(-1))
        g_out = torch.zeros_like(g_in)
        side = torch.cuda.Stream()
        side.wait_stream(torch.cuda.current_stream())
        with torch.cuda.stream(side):
            for _ in range(2):      # warm up allocations/caches
                self.hierarchy.cycle(g_in, g_out, zero_initial_guess=True)
        torch.cuda.current_stream().wait_stream(side)
        graph = torch.cuda.CUDAGraph()
        with torch.cuda.graph(graph):
            self.hierarchy.cycle(g_in, g_out, zero_initial_guess=True)
        # validate: replay must equal an eager cycle on the same input
        probe = torch.rand_like(g_in)
        g_in.copy_(probe)
        graph.replay()
        replay_out = g_out.clone()
        eager_out = torch.zeros_like(g_out)
        self.hierarchy.cycle(probe, eager_out, zero_initial_guess=True)
        torch.cuda.synchronize()
        scale = float(eager_out.abs().max().item()) or 1.0
        if float((replay_out - eager_out).abs().max().item()) > 1e-10 * scale:
            raise RuntimeError("hipGraph replay mismatch")
        return graph, g_in, g_out

    def _apply_graph(self, b, x) -> bool:
        """x = cycle(b) via graph replay. Returns False when unavailable."""
        if not self._graph_eligible():
            return False
        if self._graph is None:
            try:
                self._graph, self._g_in, self._g_out = self._capture_graph(b)
            except Exception:
                self._graph_failed = True
                self._graph = None
                return False
        self._g_in.copy_(b.reshape(-1))
        self._graph.replay()
        x.reshape(-1).copy_(self._g_out)
        return True

    # ----------------------------------------------------------------- solve
    def solve_iteration(self, b, x):
        if self._first and self._apply_graph(b, x):
            self._first = False
            return False
        self.hierarchy.cycle(b, x, zero_initial_guess=self._first)
        self._first = False
        return False

    def solve_init(self, b, x, zero_initial_guess):
        self._first = zero_initial_guess

    def solve(self, b, x, zero_initial_guess=False):
        # preconditioner fast path: one (or a few) cycles, no monitoring.
        # With a scaler configured, the base path must wrap the cycles so
        # the system is scaled/unscaled correctly.
        if not self.monitor_residual and getattr(self, "scaler", None) \
                is None:
            self._first = zero_initial_guess
            for _ in range(self.max_iters):
                self.solve_iteration(b, x)
            st = self.status
            st.status = st.SUCCESS
            st.iterations = self.max_iters
            return st
        return super().solve(b, x, zero_initial_guess)
