"""AMG hierarchy: setup loop + V/W/F/CG cycles.

Reference: src/amg.cu (AMG_Setup::setup :147-804, AMG_Solve :1086-1120),
src/cycles/fixed_cycle.cu:59-230 (the recursive cycle engine), src/amg_level.cu.
"""

from __future__ import annotations

import time
from typing import List

import torch

from .. import ops
from ..config import ConfigScope
from ..solvers.base import create_solver
from .level import AMGLevel, create_level
from ..output import amgx_output


class AMGHierarchy:
    """Owns the level chain, smoothers and the coarse solver."""

    def __init__(self, scope: ConfigScope, resources):
        self.scope = scope
        self.res = resources
        self.levels: List[AMGLevel] = []
        self.coarse_solver = None
        self.algorithm = scope.get("algorithm")
        self.cycle_type = scope.get("cycle")
        self.presweeps = scope.get("presweeps")
        self.postsweeps = scope.get("postsweeps")
        self.coarsest_sweeps = scope.get("coarsest_sweeps")
        self.max_levels = scope.get("max_levels")
        self.min_coarse_rows = scope.get("min_coarse_rows")
        self.coarsen_threshold = scope.get("coarsen_threshold")
        self.cycle_iters = scope.get("cycle_iters")
        self.print_grid_stats = bool(scope.get("print_grid_stats"))
        self.setup_time = 0.0
        from ..utils.profiler import PhaseProfiler
        self.profiler = PhaseProfiler(
            enabled=bool(scope.get("obtain_timings")))

    # ------------------------------------------------------------------ setup
    def setup(self, A):
        """Reference AMG_Setup::setup loop (src/amg.cu:201-415): coarsen until
        min_coarse_rows / max_levels / insufficient coarsening, then build the
        coarse solver and per-level smoothers."""
        t0 = time.perf_counter()
        self.levels = []
        dist = getattr(A, "manager", None)
        A = self._permute_fine(A)
        level = create_level(self.algorithm, A, self.scope, 0)
        self.levels.append(level)
        while True:
            A_l = level.A
            n = A_l.n_rows
            n_global = dist.global_rows(n) if dist is not None else n
            if (len(self.levels) >= self.max_levels
                    or n_global <= max(self.min_coarse_rows, 2)):
                break
            nc = level.create_coarse_vertices()
            nc_global = dist.global_rows(nc) if dist is not None else nc
            # insufficient coarsening -> stop (reference src/amg.cu:365-367)
            if nc_global >= n_global * self.coarsen_threshold or nc_global == 0 \
                    or nc_global == n_global:
                break
            Ac = level.create_coarse_matrix()
            level.alloc_coarse_vectors(Ac)
            nxt = create_level(self.algorithm, Ac, self.scope,
                               len(self.levels))
            level.next = nxt
            self.levels.append(nxt)
            level = nxt
        # smoothers on all but the coarsest level; coarse solver on the last
        for lvl in self.levels[:-1]:
            lvl.smoother = self._make_smoother()
            lvl.smoother.setup(lvl.A)
        self._setup_coarse_solver()
        if self.res.is_cuda:
            torch.cuda.synchronize()
        self.setup_time = time.perf_counter() - t0
        if self.print_grid_stats and self.res.rank == 0:
            amgx_output(self.grid_stats() + "\n")

    def resetup(self, A, reuse_levels: int):
        """Values-only re-setup: keep the coarsening structure (aggregates /
        C-F splits) of the first ``reuse_levels`` levels (<0 = all) and
        rebuild only Galerkin values, smoother factors and the coarse solver
        (reference structure_reuse_levels)."""
        t0 = time.perf_counter()
        if reuse_levels < 0:
            reuse_levels = len(self.levels)
        if getattr(self, "_fine_perm", None) is not None:
            # values-only refresh of the permuted fine copy (same structure)
            A0 = self.levels[0].A
            if A0.nnz == A.nnz:
                A0.values = A.values.reshape(-1)[self._fine_pos].contiguous()
                A0._cache.clear()
                A0._diag_idx = None
                A0.coloring = self._fine_coloring
            else:                     # structure changed: re-permute fully
                self.levels[0].A = self._permute_fine(A)
        else:
            self.levels[0].A = A
        A._cache.pop("cf_map", None)
        for i, lvl in enumerate(self.levels[:-1]):
            if i >= reuse_levels:
                # from here down, recoarsen from scratch
                self._truncate_and_recoarsen(i)
                break
            Ac = lvl.rebuild_coarse_values()
            if Ac is None:          # level cannot reuse -> full recoarsen
                self._truncate_and_recoarsen(i)
                break
            self.levels[i + 1].A = Ac
        for lvl in self.levels[:-1]:
            if lvl.smoother is None:
                lvl.smoother = self._make_smoother()
            lvl.smoother.setup(lvl.A)
        self._setup_coarse_solver()
        if self.res.is_cuda:
            torch.cuda.synchronize()
        self.setup_time = time.perf_counter() - t0

    def _truncate_and_recoarsen(self, i: int):
        """Drop levels below i and continue the setup loop from level i."""
        level = self.levels[i]
        self.levels = self.levels[:i + 1]
        level.next = None
        dist = getattr(level.A, "manager", None)
        while True:
            A_l = level.A
            n_global = dist.global_rows(A_l.n_rows) if dist is not None \
                else A_l.n_rows
            if (len(self.levels) >= self.max_levels
                    or n_global <= max(self.min_coarse_rows, 2)):
                break
            nc = level.create_coarse_vertices()
            nc_global = dist.global_rows(nc) if dist is not None else nc
            if nc_global >= n_global * self.coarsen_threshold \
                    or nc_global in (0, n_global):
                break
            Ac = level.create_coarse_matrix()
            level.alloc_coarse_vectors(Ac)
            nxt = create_level(self.algorithm, Ac, self.scope,
                               len(self.levels))
            level.next = nxt
            self.levels.append(nxt)
            level = nxt

    # smoother tuning knobs a string-declared smoother inherits from the
    # declaring AMG scope (reference: a sub-solver without its own scope
    # resolves parameters in the scope it was declared in)
    _SMOOTHER_INHERIT = (
        "relaxation_factor", "symmetric_GS", "GS_L1_variant",
        "jacobi_l1_variant", "matrix_coloring_scheme", "coloring_level",
        "max_uncolored_percentage", "num_colors", "max_num_hash",
        "ilu_sparsity_level", "chebyshev_polynomial_order",
        "chebyshev_lambda_estimate_mode", "cheby_max_lambda",
        "cheby_min_lambda", "kpz_order", "cf_smoothing_mode")

    def _make_smoother(self):
        name, sub = self.scope.sub_solver("smoother", "BLOCK_JACOBI")
        inh = {k: self.scope.get(k) for k in self._SMOOTHER_INHERIT
               if not sub.has(k) and self.scope.has(k)}
        if inh:
            sub = sub.child(dict(sub.node, **inh))
        return create_solver(name, sub, self.res)

    def _setup_coarse_solver(self):
        coarsest = self.levels[-1]
        mgr = getattr(coarsest.A, "manager", None)
        if mgr is not None:
            maxr = self.scope.get("dense_lu_max_rows")
            n_glob = mgr.global_rows(coarsest.A.n_rows)
            if not maxr or n_glob <= maxr:
                from ..solvers.dense_lu import GatheredDenseLU
                self.coarse_solver = GatheredDenseLU(self.res)
                self.coarse_solver.setup(coarsest.A)
                return
            # stalled distributed coarsening: never densify a huge gathered
            # matrix — smooth the coarsest level instead
            self.coarse_solver = None
            coarsest.smoother = self._make_smoother()
            coarsest.smoother.setup(coarsest.A)
            return
        name, sub = self.scope.sub_solver("coarse_solver", "DENSE_LU_SOLVER")
        if (sub is not None and not sub.has("max_iters")
                and self.scope.has("max_coarse_iters")):
            sub = sub.child(dict(sub.node,
                                 max_iters=self.scope.get("max_coarse_iters")))
        if name == "DENSE_LU_SOLVER":
            # a stalled coarsening must never densify a huge matrix (an
            # n^2 inverse would OOM the box): beyond the cap, smooth instead
            maxr = self.scope.get("dense_lu_max_rows")
            if maxr and coarsest.A.n_rows > maxr:
                name = None
        if name and name != "NOSOLVER":
            self.coarse_solver = create_solver(name, sub, self.res)
            self.coarse_solver.setup(coarsest.A)
        else:
            # coarsest_sweeps smoothing instead (reference fixed_cycle.cu:146)
            self.coarse_solver = None
            coarsest.smoother = self._make_smoother()
            coarsest.smoother.setup(coarsest.A)

    # ------------------------------------------------------------------ cycles
    def _permute_fine(self, A):
        """Color-renumber the FINE level (rows + columns) so its smoother
        sweeps access vectors color-contiguously (full reorder-by-color,
        reference include/matrix.h:766): the hierarchy then lives entirely
        in the permuted numbering and cycle() maps b/x at the boundary (two
        gathers per cycle vs ~num_colors x random-line vector traffic per
        sweep). Single-process scalar device matrices only."""
        self._fine_perm = None
        import os
        # Measured on MI355X (gpurun call 17): color-permuting the FINE
        # level costs ~8% on the 256^3 bench — the natural band ordering's
        # +-1-neighbor cache-line sharing in the x/w gathers beats
        # color-contiguous vector access. Opt-in for matrices without
        # banded structure.
        if not os.environ.get("AMGX_AMD_FINE_PERM"):
            return A
        if (not A.values.is_cuda or A.block_dim != 1
                or getattr(A, "manager", None) is not None
                or A.n_rows < 4096):
            return A
        from ..matrix import CSRMatrix
        from .coloring import MatrixColoring
        col = MatrixColoring.create(A, self.scope)
        dev = A.values.device
        perm = col.rows_sorted.to(torch.int64)
        iperm = torch.empty_like(perm)
        iperm[perm] = torch.arange(A.n_rows, dtype=torch.int64, device=dev)
        ro64 = A.row_offsets.to(torch.int64)
        deg = ro64[1:] - ro64[:-1]
        counts = deg[perm]
        csum = torch.cumsum(counts, 0)
        ro_p = torch.zeros(A.n_rows + 1, dtype=torch.int32, device=dev)
        ro_p[1:] = csum.to(torch.int32)
        pos = (torch.repeat_interleave(ro64[perm], counts)
               + torch.arange(int(A.nnz), device=dev, dtype=torch.int64)
               - torch.repeat_interleave(csum - counts, counts))
        rows_p = torch.repeat_interleave(
            torch.arange(A.n_rows, dtype=torch.int64, device=dev), counts)
        ci_p = iperm[A.col_indices.to(torch.int64)[pos]]
        # per-row column sort (one radix argsort of the combined key)
        key = rows_p * A.n_rows + ci_p
        order = torch.argsort(key)
        pos = pos[order]
        Ap = CSRMatrix(ro_p, ci_p[order].to(torch.int32).contiguous(),
                       A.values[pos].contiguous(), n_cols=A.n_cols)
        Ap.coloring = MatrixColoring(col.colors[perm].contiguous(),
                                     col.num_colors)
        self._fine_perm = perm
        self._fine_iperm = iperm
        self._fine_pos = pos          # values gather map for resetup
        self._fine_coloring = Ap.coloring
        self._pb = torch.empty(A.n_rows, dtype=A.values.dtype, device=dev)
        self._px = torch.empty_like(self._pb)
        return Ap

    def cycle(self, b: torch.Tensor, x: torch.Tensor,
              zero_initial_guess: bool = True):
        """One AMG cycle on the finest level."""
        if self._fine_perm is not None:
            bp = self._pb
            xp = self._px
            if bp.dtype != b.dtype:
                bp = self._pb = torch.empty_like(b.reshape(-1))
                xp = self._px = torch.empty_like(bp)
            torch.index_select(b.reshape(-1), 0, self._fine_perm, out=bp)
            if not zero_initial_guess:
                torch.index_select(x.reshape(-1), 0, self._fine_perm,
                                   out=xp)
            self._cycle(0, bp, xp, zero_initial_guess, self.cycle_type)
            x.reshape(-1)[self._fine_perm] = xp
            return
        self._cycle(0, b, x, zero_initial_guess, self.cycle_type)

    def _cycle(self, li: int, b, x, zero_guess: bool, ctype: str):
        """Reference FixedCycle::cycle (src/cycles/fixed_cycle.cu:59-230)."""
        level = self.levels[li]
        prof = self.profiler
        if level.r is not None and level.r.dtype != b.dtype:
            # mixed precision (dDFI): matrix stores fp32, but ALL cycle
            # vectors carry the vector precision — re-key the scratch
            # vectors lazily off the rhs dtype
            level.r = level.r.to(b.dtype)
            if level.bc is not None:
                level.bc = level.bc.to(b.dtype)
                level.xc = level.xc.to(b.dtype)
        if li == len(self.levels) - 1:
            prof.tic("coarseSolve")
            if self.coarse_solver is not None:
                self.coarse_solver.solve(b, x, zero_initial_guess=True)
            else:
                if zero_guess:
                    x.zero_()
                level.smoother.sweep(b, x, self.coarsest_sweeps)
            prof.toc("coarseSolve")
            return
        if zero_guess:
            x.zero_()
        pre = self.presweeps
        if li == 0:
            fs = int(self.scope.get("finest_sweeps") or -1)
            if fs >= 0:       # reference finest_sweeps override
                pre = fs
        if pre > 0:
            prof.tic("Smoother")
            level.smoother.sweep(b, x, pre)
            prof.toc("Smoother")
        prof.tic("computeResidual")
        ops.residual(level.A, x, b, level.r)
        prof.toc("computeResidual")
        prof.tic("restrictResidual")
        level.restrict_residual(level.r, level.bc)
        prof.toc("restrictResidual")
        if ctype in ("CG", "CGF"):
            # K-cycle (reference src/cycles/cg_cycle.cu + cg_flex_cycle.cu):
            # accelerate the coarse correction with cycle_iters FCG steps
            # preconditioned by the next-level cycle. CGF uses the flexible
            # (Polak-Ribiere) beta; at this granularity both use the same
            # 2-step truncated orthogonalization.
            self._kcycle_coarse(li + 1, level.bc, level.xc)
        else:
            # V recurses once; W twice; F = F then V
            # (reference src/cycles/{v,w,f}_cycle.cu)
            repeats = 2 if ctype in ("W", "F") and li + 2 < len(self.levels) \
                else 1
            level.xc.zero_()
            for rep in range(repeats):
                sub = "V" if (ctype == "F" and rep > 0) else ctype
                self._cycle(li + 1, level.bc, level.xc, rep == 0, sub)
        prof.tic("prolongate")
        level.prolongate_and_apply(level.xc, x)
        prof.toc("prolongate")
        if self.postsweeps > 0:
            prof.tic("Smoother")
            level.smoother.sweep(b, x, self.postsweeps)
            prof.toc("Smoother")

    def _kcycle_coarse(self, li: int, b, x):
        """Notay K-cycle coarse correction (reference src/cycles/cg_cycle.cu):
        truncated flexible-CG with up to cycle_iters (default 2) inner
        cycle-preconditioned steps."""
        A = self.levels[li].A
        mgr = getattr(A, "manager", None)

        def gdot(u, v):
            d = ops.dot(u.reshape(-1)[:mgr.owned_size] if mgr else u,
                        v.reshape(-1)[:mgr.owned_size] if mgr else v)
            return mgr.global_sum(d) if mgr else d

        x.zero_()
        c1 = torch.zeros_like(x)
        self._cycle(li, b, c1, True, "CG")
        v1 = ops.spmv(A, c1, torch.zeros_like(x))
        rho1 = gdot(c1, v1)
        a1 = gdot(c1, b)
        if rho1 == 0.0:
            return
        if self.cycle_iters < 2:
            ops.axpy(x, c1, a1 / rho1)
            return
        rt = b.clone()
        ops.axpy(rt, v1, -a1 / rho1)
        c2 = torch.zeros_like(x)
        self._cycle(li, rt, c2, True, "CG")
        v2 = ops.spmv(A, c2, torch.zeros_like(x))
        gamma = gdot(c2, v1)
        beta = gdot(c2, rt)
        rho2 = gdot(c2, v2) - gamma * gamma / rho1
        if rho2 == 0.0:
            ops.axpy(x, c1, a1 / rho1)
            return
        ops.axpy(x, c1, a1 / rho1 - gamma * beta / (rho1 * rho2))
        ops.axpy(x, c2, beta / rho2)

    # ------------------------------------------------------------------ stats
    def grid_stats(self) -> str:
        """Reference AMG::getGridStatisticsString (src/amg.cu:1230-1330)."""
        lines = ["AMG Grid:", "   Number of Levels: %d" % len(self.levels),
                 "      LVL        ROWS         NNZ    SPRSTY   MEM(MB)"]
        total_rows = total_nnz = 0
        fine_rows = self.levels[0].A.n_rows
        fine_nnz = self.levels[0].A.nnz
        for i, l in enumerate(self.levels):
            n, nnz = l.A.n_rows, l.A.nnz
            total_rows += n
            total_nnz += nnz
            sp = nnz / (n * n) if n else 0.0
            bd = l.A.block_dim
            mem = (nnz * (bd * bd * l.A.values.element_size() + 4)
                   + (n + 1) * 4) / 1e6
            lines.append(f"      {i:3d} {n:11d} {nnz:11d}  {sp:8.3g}"
                         f"  {mem:8.1f}")
        lines.append(f"      Grid Complexity: {total_rows / max(fine_rows, 1):.5g}")
        lines.append(f"      Operator Complexity: {total_nnz / max(fine_nnz, 1):.5g}")
        return "\n".join(lines)
