"""Smoothers: Block-Jacobi, L1-Jacobi, (multicolor) Gauss-Seidel, Chebyshev.

Reference: src/solvers/{block_jacobi,jacobi_l1,multicolor_gauss_seidel,
gauss_seidel,cheb,chebyshev_poly}_solver.cu. All are Solvers (usable
standalone or composed as AMG smoothers); one ``solve_iteration`` = one sweep.
"""

from __future__ import annotations

import math

import torch

from .. import ops
from .base import Solver, register_solver


class _SmootherBase(Solver):
    is_smoother = True

    def sweep(self, b, x, n: int = 1):
        for _ in range(n):
            self.solve_iteration(b, x)

    def _get_scratch(self, like):
        sc = getattr(self, "_scratch", None)
        if sc is None or sc.numel() != like.numel() or sc.device != like.device:
            sc = self._scratch = torch.zeros_like(like.reshape(-1))
        return sc

    def _apply_dinv(self, v):
        """v <- D^-1 v with self.dinv (scalar vector or inverted blocks);
        distributed: owned slice only, halo tail untouched."""
        if self.A.block_dim == 1:
            mgr = getattr(self.A, "manager", None)
            if mgr is not None:
                vo = v.reshape(-1)[:mgr.owned_size]
                vo.mul_(self.dinv.reshape(-1))
                return
            v.mul_(self.dinv.reshape(-1))
        else:
            bd = self.A.block_dim
            n_owned = self.A.n_rows * bd     # halo tail untouched (dist)
            vo = v.reshape(-1)[:n_owned]
            vo.copy_(torch.bmm(self.dinv.to(v.dtype),
                               vo.reshape(-1, bd, 1)).reshape(-1))

    def _power_iteration(self, iters: int) -> float:
        """lambda_max(D^-1 A) estimate (reference cheb_solver.cu mode 0)."""
        mgr = getattr(self.A, "manager", None)
        n = mgr.ext_size if mgr is not None \
            else self.A.n_rows * self.A.block_dim
        g = torch.Generator().manual_seed(7177)
        v = torch.rand(n, generator=g, dtype=torch.float64) \
            .to(self.A.dtype).to(self.A.device)
        Av = torch.zeros_like(v)
        lam = 1.0
        for _ in range(iters):
            nv = math.sqrt(max(self.dot(v, v), 0.0))
            if nv == 0:
                break
            ops.scal(v, 1.0 / nv)
            ops.spmv(self.A, v, Av)
            self._apply_dinv(Av)
            lam = self.dot(v, Av)
            v, Av = Av, v
        return abs(lam) * 1.05   # safety factor


@register_solver("BLOCK_JACOBI")
class BlockJacobiSolver(_SmootherBase):
    """Damped (block-)Jacobi; relaxation_factor default 0.9
    (reference src/solvers/block_jacobi_solver.cu, Dinv per-block inversion)."""

    l1 = False

    def solver_setup(self):
        self.dinv = ops.jacobi_dinv(self.A, l1=self.l1)
        # Spectral safeguard: Galerkin (D2) coarse operators can reach
        # lam_max(D^-1 A) ~ 4, where the fixed 0.9 damping DIVERGES as a
        # smoother (measured 3.96 on the level-3 operator of a 64^3 Poisson
        # classical hierarchy; a W-cycle visits that level 8x per cycle and
        # amplifies the instability while V masks it). Clamp the effective
        # relaxation so |1 - w*lam| < 1 holds across the whole spectrum.
        if not self.A.dtype.is_complex:
            lam = self._power_iteration(10)
            if lam * self.relaxation_factor > 1.8:
                self.relaxation_factor = 1.8 / lam

    def solve_iteration(self, b, x):
        # fused single-pass sweep with ping-pong scratch, copied back once
        sc = self._get_scratch(x)
        ops.jacobi_smooth(self.A, self.dinv, b, x, sc, self.relaxation_factor)
        x.reshape(-1).copy_(sc)
        return False

    def sweep(self, b, x, n: int = 1):
        # ping-pong across sweeps: one copy at the end only when n is odd
        sc = self._get_scratch(x)
        cur, other = x.reshape(-1), sc
        for _ in range(n):
            ops.jacobi_smooth(self.A, self.dinv, b, cur, other,
                              self.relaxation_factor)
            cur, other = other, cur
        if cur.data_ptr() != x.reshape(-1).data_ptr():
            x.reshape(-1).copy_(cur)


@register_solver("JACOBI_L1")
class JacobiL1Solver(_SmootherBase):
    """L1-Jacobi (reference src/solvers/jacobi_l1_solver.cu): diag gets the
    off-diagonal L1 row sum added; undamped by default in the reference."""

    def __init__(self, scope, resources):
        super().__init__(scope, resources)
        if not scope.has("relaxation_factor"):
            self.relaxation_factor = 1.0

    def solver_setup(self):
        self.dinv = ops.jacobi_dinv(self.A, l1=True)

    def solve_iteration(self, b, x):
        sc = self._get_scratch(x)
        ops.jacobi_smooth(self.A, self.dinv, b, x, sc, self.relaxation_factor)
        x.reshape(-1).copy_(sc)
        return False


@register_solver("MULTICOLOR_GS")
class MulticolorGSSolver(_SmootherBase):
    """Multicolor Gauss-Seidel (reference
    src/solvers/multicolor_gauss_seidel_solver.cu): rows of one color update
    in parallel; colors sweep ascending, then descending when symmetric_GS."""

    def __init__(self, scope, resources):
        super().__init__(scope, resources)
        self.symmetric = bool(scope.get("symmetric_GS"))
        if not scope.has("relaxation_factor"):
            self.relaxation_factor = 1.0

    def solver_setup(self):
        A = self.A
        if A.coloring is None:
            from ..amg.coloring import MatrixColoring
            A.coloring = MatrixColoring.create(A, self.scope)
        self.dinv = ops.jacobi_dinv(A, l1=bool(self.scope.get("GS_L1_variant")))

    def solve_iteration(self, b, x):
        ops.gs_sweep(self.A, self.dinv, b, x, self.A.coloring,
                     self.relaxation_factor, self.symmetric)
        return False


@register_solver("GS")
class GSSolver(MulticolorGSSolver):
    """Sequential-flavor GS; on device we use the multicolor schedule
    (reference src/solvers/gauss_seidel_solver.cu is serial-ish; a serial
    sweep has no MI355X-native expression, color-parallel is the idiom)."""


@register_solver("FIXCOLOR_GS")
class FixcolorGSSolver(MulticolorGSSolver):
    """Fixed 8-coloring GS for structured cubic grids (reference
    src/solvers/fixcolor_gauss_seidel_solver.cu: color = the 2x2x2 lattice
    parity of the row's (x,y,z) grid coordinates, num_colors=8, no coloring
    computation). Falls back to the computed multicolor schedule when the
    matrix is not a row-major cube (the fixed lattice would race)."""

    def solver_setup(self):
        A = self.A
        if A.coloring is None:
            A.coloring = self._lattice_coloring(A) or \
                __import__("amgx_amd.amg.coloring",
                           fromlist=["MatrixColoring"]) \
                .MatrixColoring.create(A, self.scope)
        self.dinv = ops.jacobi_dinv(A, l1=bool(self.scope.get("GS_L1_variant")))

    @staticmethod
    def _lattice_coloring(A):
        """color(g) = parity bits of (x, y, z) for a row-major n1d^3 grid;
        validated distance-1 against A's actual structure before use."""
        from ..amg.coloring import MatrixColoring
        n = A.n_rows
        n1d = round(n ** (1.0 / 3.0))
        if n1d ** 3 != n:
            return None
        g = torch.arange(n, device=A.row_offsets.device)
        x = g % n1d
        y = (g // n1d) % n1d
        z = g // (n1d * n1d)
        colors = ((z & 1) | ((y & 1) << 1) | ((x & 1) << 2)).to(torch.int32)
        col = MatrixColoring(colors, 8)
        return col if col.validate(A) else None


@register_solver("CHEBYSHEV")
class ChebyshevSolver(_SmootherBase):
    """Chebyshev iteration preconditioned by the (L1-)diagonal; lambda_max
    estimated by power iteration on D^-1 A when
    chebyshev_lambda_estimate_mode=0 (reference src/solvers/cheb_solver.cu,
    src/core.cu:409-412)."""

    def __init__(self, scope, resources):
        super().__init__(scope, resources)
        self.est_mode = scope.get("chebyshev_lambda_estimate_mode")

    def solver_setup(self):
        self.dinv = ops.jacobi_dinv(self.A, l1=True)
        if self.est_mode in (0, 1):
            self.lmax = self._power_iteration(16)
            self.lmin = self.lmax / 8.0
        elif self.est_mode == 2:
            # preconditioned spectrum assumed compressed to ~[0,1]
            # (reference cheb_solver.cu:196: lmax=0.9)
            self.lmax = 0.9
            self.lmin = self.lmax / 8.0
        else:   # mode 3: user-provided estimates (cheb_solver.cu:209-211)
            self.lmax = self.scope.get("cheby_max_lambda")
            self.lmin = self.scope.get("cheby_min_lambda")
        self._init_cheb()

    def _init_cheb(self):
        self.theta = 0.5 * (self.lmax + self.lmin)
        self.delta = 0.5 * (self.lmax - self.lmin)

    def solve_init(self, b, x, zero_initial_guess):
        self._sigma = self.theta / self.delta if self.delta else 1.0
        self._rho = 1.0 / self._sigma if self._sigma else 1.0
        self._d = None

    def sweep(self, b, x, n: int = 1):
        # smoother entry (AMG per-level sweeps bypass solve()): the
        # Chebyshev recurrence state must be re-seeded per sweep sequence
        self.solve_init(b, x, False)
        for _ in range(n):
            self.solve_iteration(b, x)

    def solve_iteration(self, b, x):
        r = ops.residual(self.A, x, b)
        self._apply_dinv(r)
        if self._d is None:
            self._d = r.mul_(1.0 / self.theta)
        else:
            rho_new = 1.0 / (2.0 * self._sigma - self._rho)
            a = rho_new * self._rho
            c = 2.0 * rho_new / self.delta
            ops.axpby(self._d, r, c, a)
            self._rho = rho_new
        ops.axpy(x, self._d, 1.0)
        return False


@register_solver("CHEBYSHEV_POLY")
class ChebyshevPolySolver(ChebyshevSolver):
    """Chebyshev polynomial smoother of fixed order (reference
    src/solvers/chebyshev_poly.cu, order 5 default): one solve_iteration runs
    the full order-k polynomial."""

    def __init__(self, scope, resources):
        super().__init__(scope, resources)
        self.order = scope.get("chebyshev_polynomial_order")

    def solve_iteration(self, b, x):
        self.solve_init(b, x, False)
        for _ in range(self.order):
            super().solve_iteration(b, x)
        return False


@register_solver("POLYNOMIAL")
@register_solver("KPZ_POLYNOMIAL")
class PolynomialSolver(ChebyshevPolySolver):
    """Polynomial smoother family; maps to the Chebyshev polynomial engine
    (reference src/solvers/polynomial_solver.cu, kpz_order param)."""

    def __init__(self, scope, resources):
        super().__init__(scope, resources)
        if scope.has("kpz_order"):
            self.order = int(scope.get("kpz_order"))
