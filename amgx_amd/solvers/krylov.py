"""Krylov solvers: CG/PCG/PCGF, BiCGStab/PBiCGStab, GMRES/FGMRES, IDR(s).

Reference: src/solvers/{cg,pcg,pcgf,bicgstab,pbicgstab,gmres,fgmres,idr}_solver.cu.
Each solve_iteration matches the reference's operation order so iteration
counts are comparable (iteration parity is the primary
correctness-of-algorithm signal).
"""

from __future__ import annotations

import math

import torch

from .. import ops
from .base import Solver, register_solver


@register_solver("PCG")
@register_solver("CG")
class PCGSolver(Solver):
    """Preconditioned CG (reference src/solvers/pcg_solver.cu:109-189).
    Plain CG is PCG with no preconditioner (NOSOLVER)."""

    def __init__(self, scope, resources):
        super().__init__(scope, resources)
        self.flexible = bool(scope.get("pcg_flexible"))

    def solver_setup(self):
        # keep the preconditioner object across resetup so its hierarchy can
        # honor structure_reuse_levels (reference AMGX_solver_resetup)
        if getattr(self, "precond", None) is None:
            self.precond = self.make_preconditioner()
        if self.precond is not None:
            self.precond.setup(self.A)

    def solve_init(self, b, x, zero_initial_guess):
        self.r = ops.residual(self.A, x, b)
        if self.precond is not None:
            self.z = self.new_vec(self.r)
            self.precond.solve(self.r, self.z, zero_initial_guess=True)
        else:
            self.z = self.r.clone()
        self.p = self.z.clone()
        self.Ap = self.new_vec(self.r)
        self.rz = self.dot(self.r, self.z)
        self._rnorm = None

    def solve_iteration(self, b, x):
        ops.spmv(self.A, self.p, self.Ap)
        pAp = self.dot(self.p, self.Ap)
        if pAp == 0.0:
            return True
        alpha = self.rz / pAp
        ops.axpy(x, self.p, alpha)
        ops.axpy(self.r, self.Ap, -alpha)
        self._rnorm = self.compute_norm(self.r)
        if self.precond is not None:
            if self.flexible:
                z_old = self.z.clone()
            self.precond.solve(self.r, self.z, zero_initial_guess=True)
            if self.flexible:
                # Polak-Ribiere beta (reference src/solvers/pcgf_solver.cu)
                rz_new = self.dot(self.r, self.z)
                beta = (rz_new - self.dot(self.r, z_old)) / self.rz if self.rz else 0.0
                self.rz = rz_new
            else:
                rz_new = self.dot(self.r, self.z)
                beta = rz_new / self.rz if self.rz else 0.0
                self.rz = rz_new
        else:
            self.z.copy_(self.r)
            rz_new = self.dot(self.r, self.z)
            beta = rz_new / self.rz if self.rz else 0.0
            self.rz = rz_new
        ops.axpby(self.p, self.z, 1.0, beta)
        return False

    def last_residual_norm(self, b, x):
        return self._rnorm if self._rnorm is not None else super().last_residual_norm(b, x)


@register_solver("PCGF")
class PCGFSolver(PCGSolver):
    def __init__(self, scope, resources):
        super().__init__(scope, resources)
        self.flexible = True


@register_solver("PBICGSTAB")
@register_solver("BICGSTAB")
class BiCGStabSolver(Solver):
    """(Preconditioned) BiCGStab (reference src/solvers/pbicgstab_solver.cu)."""

    def solver_setup(self):
        # keep the preconditioner object across resetup so its hierarchy can
        # honor structure_reuse_levels (reference AMGX_solver_resetup)
        if getattr(self, "precond", None) is None:
            self.precond = self.make_preconditioner()
        if self.precond is not None:
            self.precond.setup(self.A)

    def _apply_M(self, v, out):
        if self.precond is None:
            out.copy_(v)
        else:
            self.precond.solve(v, out, zero_initial_guess=True)

    def solve_init(self, b, x, zero_initial_guess):
        self.r = ops.residual(self.A, x, b)
        self.r_tld = self.r.clone()
        self.p = self.new_vec(self.r)
        self.v = self.new_vec(self.r)
        self.ph = self.new_vec(self.r)
        self.sh = self.new_vec(self.r)
        self.t = self.new_vec(self.r)
        self.rho_old = self.alpha = self.omega = 1.0
        self._rnorm = None

    def solve_iteration(self, b, x):
        rho = self.dot(self.r_tld, self.r)
        if rho == 0.0:
            return True
        if self.omega == 0.0:
            return True
        beta = (rho / self.rho_old) * (self.alpha / self.omega)
        # p = r + beta*(p - omega*v)
        ops.axpy(self.p, self.v, -self.omega)
        ops.axpby(self.p, self.r, 1.0, beta)
        self._apply_M(self.p, self.ph)
        ops.spmv(self.A, self.ph, self.v)
        denom = self.dot(self.r_tld, self.v)
        if denom == 0.0:
            return True
        self.alpha = rho / denom
        ops.axpy(self.r, self.v, -self.alpha)  # s = r - alpha v (in r)
        snorm = self.compute_norm(self.r)
        if self.convergence.converged(snorm):
            ops.axpy(x, self.ph, self.alpha)
            self._rnorm = snorm
            self.rho_old = rho
            return True
        self._apply_M(self.r, self.sh)
        ops.spmv(self.A, self.sh, self.t)
        tt = self.dot(self.t, self.t)
        self.omega = self.dot(self.t, self.r) / tt if tt != 0.0 else 0.0
        ops.axpy(x, self.ph, self.alpha)
        ops.axpy(x, self.sh, self.omega)
        ops.axpy(self.r, self.t, -self.omega)
        self._rnorm = self.compute_norm(self.r)
        self.rho_old = rho
        return False

    def last_residual_norm(self, b, x):
        return self._rnorm if self._rnorm is not None else super().last_residual_norm(b, x)


@register_solver("FGMRES")
@register_solver("GMRES")
class FGMRESSolver(Solver):
    """Right-preconditioned flexible GMRES with restart (reference
    src/solvers/fgmres_solver.cu: Arnoldi + host-side Givens on the small
    Hessenberg; gmres_n_restart default 20, src/core.cu:390)."""

    def __init__(self, scope, resources):
        super().__init__(scope, resources)
        self.restart = scope.get("gmres_n_restart")
        # truncated orthogonalization window (reference gmres_krylov_dim,
        # src/core.cu:391; 0 = full restart-length orthogonalization)
        self.krylov_dim = int(scope.get("gmres_krylov_dim") or 0)

    def solver_setup(self):
        # keep the preconditioner object across resetup so its hierarchy can
        # honor structure_reuse_levels (reference AMGX_solver_resetup)
        if getattr(self, "precond", None) is None:
            self.precond = self.make_preconditioner()
        if self.precond is not None:
            self.precond.setup(self.A)

    def _apply_M(self, v, out):
        if self.precond is None:
            out.copy_(v)
        else:
            self.precond.solve(v, out, zero_initial_guess=True)

    def solve_init(self, b, x, zero_initial_guess):
        m = self.restart
        self.V = []          # Krylov basis
        self.Z = []          # preconditioned vectors (flexible)
        # complex modes keep a complex Hessenberg; rotations have real c,
        # complex s (LAPACK zrotg convention)
        hd = torch.complex128 if b.is_complex() else torch.float64
        self._hd = hd
        self.H = torch.zeros(m + 1, m, dtype=hd)
        self.cs = torch.zeros(m, dtype=torch.float64)
        self.sn = torch.zeros(m, dtype=hd)
        self.g = torch.zeros(m + 1, dtype=hd)
        self._restart_init(b, x)

    def _sc(self, v):
        """Hessenberg scalar: complex in complex mode, float otherwise."""
        return complex(v) if self._hd is torch.complex128 else float(v)

    def _restart_init(self, b, x):
        r = ops.residual(self.A, x, b)
        beta = math.sqrt(abs(self.dot(r, r)))
        self.beta = beta
        self.j = 0
        self.V = [r / beta if beta > 0 else r]
        self.Z = []
        self.g.zero_()
        self.g[0] = beta
        self._rnorm = beta

    # -- device fast path helpers (<=1 host sync per iteration) -----------
    def _dot_dev(self, x, y):
        """Device-scalar (possibly all-reduced) dot; no host sync."""
        from ..ops import gpu as G
        mgr = getattr(self.A, "manager", None)
        if mgr is not None:
            x = x.reshape(-1)[:mgr.owned_size]
            y = y.reshape(-1)[:mgr.owned_size]
        h = G.dot_async(x, y)
        if mgr is not None:
            import torch.distributed as tdist
            tdist.all_reduce(h)
        return h

    def _iterate_device(self, j, w):
        """MGS projections with device-resident coefficients: all h_ij stay
        on the GPU (all-reduced on stream when distributed), w is normalized
        on device, and the iteration's ONE host sync reads the whole
        Hessenberg column afterwards (reference pipelined-dot intent,
        src/solvers/fgmres_solver.cu)."""
        from ..ops import gpu as G
        i0 = max(0, j + 1 - self.krylov_dim) if self.krylov_dim > 0 else 0
        if not hasattr(self, "_hbuf") or self._hbuf.numel() < self.restart + 2 \
                or self._hbuf.dtype != w.dtype:
            self._hbuf = torch.zeros(self.restart + 2, dtype=w.dtype,
                                     device=w.device)
        hbuf = self._hbuf
        for i in range(i0, j + 1):
            hi = self._dot_dev(self.V[i], w)
            hbuf[i] = hi
            G.axpy_dalpha(w, self.V[i], hi, -1.0)
        ww = self._dot_dev(w, w)
        hbuf[j + 1] = ww
        G.scal_drsqrt(w, ww)              # w /= ||w|| (no-op on breakdown)
        hcol = hbuf[i0:j + 2].double().cpu()   # the ONE sync
        for i in range(i0, j + 1):
            self.H[i, j] = float(hcol[i - i0])
        return math.sqrt(max(float(hcol[j + 1 - i0]), 0.0)), True

    def solve_iteration(self, b, x):
        j = self.j
        m = self.restart
        vj = self.V[j]
        z = self.new_vec(vj)
        self._apply_M(vj, z)
        self.Z.append(z)
        w = self.new_vec(vj)
        ops.spmv(self.A, z, w)
        fast = (torch.is_tensor(w) and w.is_cuda and not w.is_complex())
        if fast:
            hnext, w_normalized = self._iterate_device(j, w)
            self.H[j + 1, j] = hnext
        else:
            # modified Gram-Schmidt: h_ij = <v_i, w> (conjugated in the
            # first arg); truncated to the last krylov_dim vectors when set
            w_normalized = False
            i0 = max(0, j + 1 - self.krylov_dim) if self.krylov_dim > 0 else 0
            for i in range(i0, j + 1):
                hij = self.dot(self.V[i], w)
                self.H[i, j] = hij
                ops.axpy(w, self.V[i], -hij)
            hnext = math.sqrt(abs(self.dot(w, w)))
            self.H[j + 1, j] = hnext
        # apply stored Givens rotations to column j
        for i in range(j):
            ci, si = float(self.cs[i]), self._sc(self.sn[i])
            si_c = si.conjugate() if isinstance(si, complex) else si
            hi, hi1 = self._sc(self.H[i, j]), self._sc(self.H[i + 1, j])
            self.H[i, j] = ci * hi + si * hi1
            self.H[i + 1, j] = -si_c * hi + ci * hi1
        # new rotation zeroing H[j+1, j] (hnext real >= 0)
        a = self._sc(self.H[j, j])
        denom = math.sqrt(abs(a) ** 2 + hnext * hnext)
        if denom == 0.0:
            self._update_x(x, j)
            return True
        if abs(a) == 0.0:
            c, s, rr = 0.0, 1.0, hnext
        else:
            phase = a / abs(a)
            c = abs(a) / denom
            s = phase * (hnext / denom)
            rr = phase * denom
        self.cs[j] = c
        self.sn[j] = s
        self.H[j, j] = rr
        self.H[j + 1, j] = 0.0
        gj = self._sc(self.g[j])
        sconj = s.conjugate() if isinstance(s, complex) else s
        self.g[j + 1] = -sconj * gj
        self.g[j] = c * gj
        self._rnorm = abs(self._sc(self.g[j + 1]))
        lucky = hnext == 0.0
        converged = self.convergence.converged(self._rnorm)
        if converged or lucky or j + 1 == m:
            self._update_x(x, j)
            if not (converged or lucky):
                self._restart_init(b, x)   # restart
            return converged or lucky
        self.V.append(w if w_normalized else w / hnext)
        self.j += 1
        return False

    def _update_x(self, x, j):
        # back-substitute y from the j+1 x j+1 triangular system
        y = torch.zeros(j + 1, dtype=self._hd)
        for i in range(j, -1, -1):
            s = self._sc(self.g[i]) - self._sc(
                self.H[i, i + 1:j + 1] @ y[i + 1:j + 1])
            hii = self._sc(self.H[i, i])
            y[i] = s / hii if hii != 0.0 else 0.0
        for i in range(j + 1):
            ops.axpy(x, self.Z[i], self._sc(y[i]))

    def last_residual_norm(self, b, x):
        return self._rnorm


@register_solver("IDR")
@register_solver("IDRMSYNC")
class IDRSolver(Solver):
    """IDR(s) (reference src/solvers/idr_solver.cu; subspace_dim_s=8 default,
    src/core.cu:393). Implemented as the standard IDR(s) biorthogonal variant."""

    def __init__(self, scope, resources):
        super().__init__(scope, resources)
        self.s = scope.get("subspace_dim_s")

    def solver_setup(self):
        # keep the preconditioner object across resetup so its hierarchy can
        # honor structure_reuse_levels (reference AMGX_solver_resetup)
        if getattr(self, "precond", None) is None:
            self.precond = self.make_preconditioner()
        if self.precond is not None:
            self.precond.setup(self.A)

    def _apply_M(self, v, out):
        if self.precond is None:
            out.copy_(v)
        else:
            self.precond.solve(v, out, zero_initial_guess=True)

    def solve_init(self, b, x, zero_initial_guess):
        s = self.s
        n = b.numel()
        self.r = ops.residual(self.A, x, b)
        g = torch.Generator().manual_seed(1234)
        P = torch.randn(s, n, generator=g, dtype=torch.float64)
        # orthonormalize rows (host QR on small s x n is overkill; MGS)
        for i in range(s):
            for k in range(i):
                P[i] -= (P[i] @ P[k]) * P[k]
            P[i] /= torch.linalg.vector_norm(P[i])
        self.P = P.to(self.r.dtype).to(self.r.device)
        self.G = [self.new_vec(self.r) for _ in range(s)]
        self.U = [self.new_vec(self.r) for _ in range(s)]
        self.M = torch.eye(s, dtype=torch.float64)
        self.omega = 1.0
        self._rnorm = None

    def solve_iteration(self, b, x):
        s = self.s
        f = torch.tensor([self.dot(self.P[i], self.r) for i in range(s)],
                         dtype=torch.float64)
        for k in range(s):
            # solve lower-triangular M[k:,k:] c = f[k:]
            c = torch.linalg.solve_triangular(self.M[k:, k:], f[k:].unsqueeze(1),
                                              upper=False).squeeze(1)
            v = self.r.clone()
            for i in range(k, s):
                ops.axpy(v, self.G[i], -float(c[i - k]))
            vh = self.new_vec(v)
            self._apply_M(v, vh)
            # U[k] = omega*vh + sum c_i U[i]
            uk = vh.mul_(self.omega)
            for i in range(k, s):
                ops.axpy(uk, self.U[i], float(c[i - k]))
            self.U[k] = uk.clone()
            ops.spmv(self.A, self.U[k], self.G[k])
            # biorthogonalize G[k] against P[0..k-1]
            for i in range(k):
                alpha = self.dot(self.P[i], self.G[k]) / self.M[i, i]
                ops.axpy(self.G[k], self.G[i], -float(alpha))
                ops.axpy(self.U[k], self.U[i], -float(alpha))
            for i in range(k, s):
                self.M[i, k] = self.dot(self.P[i], self.G[k])
            if self.M[k, k] == 0.0:
                return True
            beta = float(f[k] / self.M[k, k])
            ops.axpy(self.r, self.G[k], -beta)
            ops.axpy(x, self.U[k], beta)
            if k + 1 < s:
                for i in range(k + 1, s):
                    f[i] -= beta * self.M[i, k]
        # dimension-reduction step
        v = self.r.clone()
        vh = self.new_vec(v)
        self._apply_M(v, vh)
        t = self.new_vec(v)
        ops.spmv(self.A, vh, t)
        tt = self.dot(t, t)
        self.omega = self.dot(t, self.r) / tt if tt != 0.0 else 0.0
        ops.axpy(self.r, t, -self.omega)
        ops.axpy(x, vh, self.omega)
        self._rnorm = self.compute_norm(self.r)
        return False

    def last_residual_norm(self, b, x):
        return self._rnorm if self._rnorm is not None else super().last_residual_norm(b, x)
