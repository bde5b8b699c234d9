"""Eigensolvers (reference layer L8: src/eigensolvers/, include/eigensolvers/
eigensolver.h:25-130; C API AMGX_eigensolver_*, include/amgx_eig_c.h:16-26).

Implemented: POWER_ITERATION (+ shift/inverse modes via operators), LANCZOS,
ARNOLDI, SUBSPACE_ITERATION, LOBPCG, and the PageRank operator
(reference include/operators/pagerank_operator.h:13). Small dense
eigenproblems (tridiagonal/Hessenberg/Rayleigh-Ritz) are solved on host with
numpy at setup-scale sizes; all vector work runs through the amgx_amd ops
layer (HIP kernels on device).
"""

from __future__ import annotations

import math
from typing import Dict, Optional, Type

import numpy as np
import torch

from . import ops
from .config import ConfigScope
from .resources import Resources, default_resources

EIGEN_REGISTRY: Dict[str, Type["EigenSolver"]] = {}


def register_eigensolver(name):
    def deco(cls):
        EIGEN_REGISTRY[name] = cls
        return cls
    return deco


def create_eigensolver(scope: ConfigScope, resources=None) -> "EigenSolver":
    name = scope.get("eig_solver")
    cls = EIGEN_REGISTRY.get(name)
    if cls is None:
        raise KeyError(f"unknown eigensolver {name!r}")
    return cls(scope, resources or default_resources())


# ------------------------------------------------------------------- operators
class Operator:
    """Linear-operator composition (reference include/operators/*.h)."""

    def __init__(self, A):
        self.A = A
        self.n = A.n_rows * A.block_dim

    def apply(self, x, y):
        ops.spmv(self.A, x, y)
        return y


class ShiftedOperator(Operator):
    """(A - shift I) x (reference shifted_operator.h)."""

    def __init__(self, A, shift: float):
        super().__init__(A)
        self.shift = shift

    def apply(self, x, y):
        ops.spmv(self.A, x, y)
        if self.shift:
            ops.axpy(y, x, -self.shift)
        return y


class SolverOperator(Operator):
    """A^{-1} x via a configured inner solver — inverse iteration
    (reference solve_operator.h / solver_operator.h)."""

    def __init__(self, A, solver):
        super().__init__(A)
        self.solver = solver

    def apply(self, x, y):
        y.zero_()
        self.solver.solve(x, y, zero_initial_guess=True)
        return y


class PageRankOperator(Operator):
    """Google-matrix operator G^T x = d * P^T x + teleport
    (reference pagerank_operator.h; AMGX_eigensolver_pagerank_setup)."""

    def __init__(self, A, damping: float = 0.85):
        super().__init__(A)
        if A.block_dim != 1:
            raise ValueError("PageRank operates on scalar matrices")
        self.d = damping
        self.mgr = getattr(A, "manager", None)
        # column-stochastic normalization: P[j,i] = A[i,j] / outdeg(i).
        # out-degree = row sums of A; full rows are rank-local even in the
        # distributed partition, so no communication is needed here
        cnt = (A.row_offsets[1:] - A.row_offsets[:-1]).to(torch.int64)
        rid = torch.repeat_interleave(
            torch.arange(A.n_rows, device=A.device), cnt)
        deg = torch.zeros(A.n_rows, dtype=A.dtype, device=A.device)
        deg.index_add_(0, rid, A.values.reshape(-1))
        self.out_inv = torch.where(deg != 0, 1.0 / deg, torch.zeros_like(deg))
        # local transpose: for a row partition the local A^T maps owned x
        # into the extended (owned + halo) column space; accumulating the
        # halo tail onto its owners (add_from_halo) completes the global
        # A^T apply (reference pagerank_operator.h runs on the transposed
        # distributed matrix)
        self.AT = ops.transpose(A)

    def apply(self, x, y):
        mgr = self.mgr
        if mgr is None:
            ops.spmv(self.AT, x * self.out_inv, y)
            ops.scal(y, self.d)
            dangling = float(ops.dot(x, (self.out_inv == 0).to(x.dtype)))
            total = float(x.sum())
            y += (self.d * dangling + (1.0 - self.d) * total) / self.n
            return y
        xo = x.reshape(-1)[:mgr.owned_size]
        z = (xo * self.out_inv).contiguous()
        yext = ops.spmv(self.AT, z)            # ext-sized partial columns
        mgr.add_from_halo(yext, block_override=1)
        yf = y.reshape(-1)
        yf[:mgr.owned_size] = self.d * yext[:mgr.owned_size]
        yf[mgr.owned_size:] = 0.0
        mask = (self.out_inv == 0).to(x.dtype)
        dangling = mgr.global_sum(float(ops.dot(xo, mask)))
        total = mgr.global_sum(float(xo.sum()))
        yf[:mgr.owned_size] += (self.d * dangling +
                                (1.0 - self.d) * total) / mgr.n_global
        return y


# ------------------------------------------------------------------- base
class EigenStatus:
    def __init__(self):
        self.converged = False
        self.iterations = 0
        self.eigenvalues = []
        self.eigenvector = None
        self.residuals = []


class EigenSolver:
    def __init__(self, scope: ConfigScope, resources: Resources):
        self.scope = scope
        self.res = resources
        self.max_iters = scope.get("eig_max_iters")
        self.tol = scope.get("eig_tolerance")
        self.which = scope.get("eig_which")
        self.shift = scope.get("eig_shift")
        self.wanted = scope.get("eig_wanted_count")
        self.subspace = int(scope.get("eig_subspace_size") or -1)
        self.check_freq = max(int(scope.get("eig_convergence_check_freq")
                                  or 1), 1)
        self.op: Optional[Operator] = None
        self.status = EigenStatus()

    # ---------------------------------------------------- dist-aware algebra
    def _mgr(self):
        return getattr(self.A, "manager", None) if hasattr(self, "A") \
            else None

    def dot(self, x, y) -> float:
        """Globally-reduced dot over owned entries (reference eigensolvers
        run on the distributed vector ops)."""
        mgr = self._mgr()
        if mgr is None:
            return ops.dot(x, y)
        return mgr.global_sum(ops.dot(mgr.owned(x), mgr.owned(y)))

    def nrm2(self, x) -> float:
        return math.sqrt(max(self.dot(x, x), 0.0))

    def _qr(self, B: torch.Tensor) -> torch.Tensor:
        """Thin QR of a tall-skinny block; distributed via Cholesky-QR
        (p x p Gram allreduce) — the TSQR idiom over RCCL."""
        mgr = self._mgr()
        if mgr is None:
            Q, _ = torch.linalg.qr(B)
            return Q
        import torch.distributed as tdist
        Bo = B[:mgr.owned_size]
        G = (Bo.T @ Bo).double()
        tdist.all_reduce(G)
        L = torch.linalg.cholesky(G + 1e-14 * torch.eye(
            G.shape[0], dtype=G.dtype, device=G.device))
        Q = B.clone()
        Q[:] = torch.linalg.solve_triangular(
            L, B.double().T, upper=False).T.to(B.dtype)
        return Q

    def _block_inner(self, X: torch.Tensor, Y: torch.Tensor) -> np.ndarray:
        """X^T Y over owned rows, globally reduced (p x p, host numpy)."""
        mgr = self._mgr()
        if mgr is None:
            return (X.double().T @ Y.double()).cpu().numpy()
        import torch.distributed as tdist
        T = (X[:mgr.owned_size].double().T @ Y[:mgr.owned_size].double())
        tdist.all_reduce(T)
        return T.cpu().numpy()

    def setup(self, A):
        self.A = A
        if self.which == "smallest":
            from .solvers import create_solver
            from .config import ConfigScope as CS
            # inverse iteration: the config's "solver" tree, when present,
            # configures the inner A^{-1} solve (reference
            # eigen_configs/INVERSE_FGMRES drives an FGMRES+AMG inner solver
            # through the same scoped-config mechanism); default = PCG+BJ
            sub = self.scope.node.get("solver") \
                if isinstance(self.scope.node, dict) else None
            if isinstance(sub, dict):
                inner = create_solver(self.scope.child(sub),
                                      resources=self.res)
            else:
                inner = create_solver("PCG", CS(None, {
                    "max_iters": 200, "monitor_residual": 1,
                    "tolerance": 1e-10, "convergence": "RELATIVE_INI",
                    "preconditioner": "BLOCK_JACOBI"}), self.res)
            inner.setup(A)
            self.op = SolverOperator(A, inner)
        elif self.shift:
            self.op = ShiftedOperator(A, self.shift)
        else:
            self.op = Operator(A)
        self.solver_setup()

    def pagerank_setup(self, A, damping: Optional[float] = None):
        self.A = A
        self.op = PageRankOperator(
            A, damping if damping is not None else
            self.scope.get("eig_damping_factor"))
        self.solver_setup()

    def solver_setup(self):
        pass

    def _start_vec(self, x0):
        """Initial vector: the caller's x0 unless it is (numerically) zero —
        a zero start annihilates power/Krylov iterations (the C API's
        freshly-created solution vector is all zeros), fall back to the
        seeded random start."""
        if x0 is not None and float(self.nrm2(x0)) > 0.0:
            return x0.clone()
        return self._rand_vec()

    def _rand_vec(self, seed=42):
        mgr = self._mgr()
        n = mgr.ext_size if mgr is not None else self.op.n
        g = torch.Generator().manual_seed(
            seed + 7919 * (mgr.rank if mgr is not None else 0))
        v = torch.rand(n, generator=g, dtype=torch.float64)
        if mgr is not None:
            v[mgr.owned_size:] = 0.0
        return v.to(self.A.dtype).to(self.A.device)

    def _to_true_eig(self, lam: float) -> float:
        if self.which == "smallest":
            return 1.0 / lam if lam != 0 else float("inf")
        if self.shift:
            return lam + self.shift
        return lam

    def solve(self, x0: Optional[torch.Tensor] = None) -> EigenStatus:
        raise NotImplementedError


@register_eigensolver("POWER_ITERATION")
@register_eigensolver("SINGLE_ITERATION")
@register_eigensolver("PAGERANK")
@register_eigensolver("INVERSE_ITERATION")
class PowerIteration(EigenSolver):
    """Reference src/eigensolvers/power_iteration_eigensolver.cu (and
    single_iteration_eigensolver.cu). The reference registers PAGERANK and
    INVERSE_ITERATION to this same single-iteration family
    (eigensolvers.cu:38-43); here the operator attached at setup (PageRank
    / shifted-inverse via eig_which=smallest) selects the behavior, and
    the INVERSE_ITERATION name defaults eig_which to smallest."""

    def __init__(self, scope, resources):
        super().__init__(scope, resources)
        if scope.get("eig_solver") == "INVERSE_ITERATION" \
                and not scope.has("eig_which"):
            self.which = "smallest"

    def solve(self, x0=None):
        st = self.status = EigenStatus()
        v = self._start_vec(x0)
        Av = torch.zeros_like(v)
        lam = 0.0
        for it in range(self.max_iters):
            nv = self.nrm2(v)
            if nv == 0:
                break
            ops.scal(v, 1.0 / nv)
            self.op.apply(v, Av)
            lam = self.dot(v, Av)
            st.iterations = it + 1
            # eig_convergence_check_freq: skip the residual norm (an extra
            # vector op + reduction) on off-period iterations (reference
            # eigensolvers.cu:30)
            if (it + 1) % self.check_freq == 0 or it + 1 == self.max_iters:
                r = Av.clone()
                ops.axpy(r, v, -lam)
                rn = self.nrm2(r) / max(abs(lam), 1e-300)
                st.residuals.append(rn)
                if rn < self.tol:
                    v, Av = Av.clone(), Av
                    st.converged = True
                    break
            v, Av = Av.clone(), Av
        st.eigenvalues = [self._to_true_eig(lam)]
        nv = self.nrm2(v)
        st.eigenvector = v / nv if nv else v
        return st


@register_eigensolver("LANCZOS")
class Lanczos(EigenSolver):
    """Reference src/eigensolvers/lanczos_eigensolver.cu: symmetric Lanczos
    with full reorthogonalization; tridiagonal solved on host."""

    def solve(self, x0=None):
        st = self.status = EigenStatus()
        m = min(self.max_iters, self.op.n)
        v = self._start_vec(x0)
        ops.scal(v, 1.0 / self.nrm2(v))
        V = [v]
        alphas, betas = [], []
        w = torch.zeros_like(v)
        for j in range(m):
            self.op.apply(V[j], w)
            a = self.dot(w, V[j])
            alphas.append(a)
            ops.axpy(w, V[j], -a)
            if j > 0:
                ops.axpy(w, V[j - 1], -betas[-1])
            # full reorthogonalization
            for q in V:
                ops.axpy(w, q, -self.dot(w, q))
            b = self.nrm2(w)
            st.iterations = j + 1
            if j >= 1:
                T = np.diag(alphas) + np.diag(betas, 1) + np.diag(betas, -1)
                evals, evecs = np.linalg.eigh(T)
                idx = -1 if self.which != "smallest" else -1  # op handles inverse
                ritz = evals[idx]
                resid = abs(b * evecs[-1, idx]) / max(abs(ritz), 1e-300)
                st.residuals.append(resid)
                if resid < self.tol or b < 1e-14:
                    st.converged = True
                    y = evecs[:, idx]
                    vec = torch.zeros_like(v)
                    for c, q in zip(y, V):
                        ops.axpy(vec, q, float(c))
                    st.eigenvector = vec
                    st.eigenvalues = [self._to_true_eig(float(ritz))]
                    return st
            if b < 1e-14:
                break
            betas.append(b)
            V.append(w / b)
            w = torch.zeros_like(v)
        if alphas:
            T = np.diag(alphas)
            if betas:
                T = T + np.diag(betas[:len(alphas) - 1], 1) \
                    + np.diag(betas[:len(alphas) - 1], -1)
            evals, evecs = np.linalg.eigh(T)
            st.eigenvalues = [self._to_true_eig(float(evals[-1]))]
            y = evecs[:, -1]
            vec = torch.zeros_like(v)
            for c, q in zip(y, V):
                ops.axpy(vec, q, float(c))
            st.eigenvector = vec
        return st


@register_eigensolver("ARNOLDI")
class Arnoldi(EigenSolver):
    """Reference src/eigensolvers/arnoldi_eigensolver.cu: Arnoldi with host
    Hessenberg eigen-decomposition (general matrices)."""

    def solve(self, x0=None):
        st = self.status = EigenStatus()
        m = min(self.max_iters, self.op.n)
        v = self._start_vec(x0)
        ops.scal(v, 1.0 / self.nrm2(v))
        V = [v]
        H = np.zeros((m + 1, m))
        w = torch.zeros_like(v)
        for j in range(m):
            self.op.apply(V[j], w)
            for i in range(j + 1):
                H[i, j] = self.dot(w, V[i])
                ops.axpy(w, V[i], -H[i, j])
            H[j + 1, j] = self.nrm2(w)
            st.iterations = j + 1
            if j >= 1:
                evals, evecs = np.linalg.eig(H[:j + 1, :j + 1])
                k = int(np.argmax(np.abs(evals)))
                resid = abs(H[j + 1, j] * evecs[-1, k]) / max(abs(evals[k]), 1e-300)
                st.residuals.append(float(resid))
                if resid < self.tol:
                    st.converged = True
                    st.eigenvalues = [self._to_true_eig(float(np.real(evals[k])))]
                    y = np.real(evecs[:, k])
                    vec = torch.zeros_like(v)
                    for c, q in zip(y, V):
                        ops.axpy(vec, q, float(c))
                    st.eigenvector = vec
                    return st
            if H[j + 1, j] < 1e-14:
                break
            V.append(w / H[j + 1, j])
            w = torch.zeros_like(v)
        if st.iterations:
            j = st.iterations
            evals, evecs = np.linalg.eig(H[:j, :j])
            k = int(np.argmax(np.abs(evals)))
            st.eigenvalues = [self._to_true_eig(float(np.real(evals[k])))]
        return st


@register_eigensolver("SUBSPACE_ITERATION")
class SubspaceIteration(EigenSolver):
    """Reference src/eigensolvers/subspace_iteration_eigensolver.cu: block
    power iteration with Rayleigh-Ritz."""

    def solve(self, x0=None):
        st = self.status = EigenStatus()
        k = max(self.wanted, 1)
        p = self.subspace if self.subspace > 0 else k + 2
        p = min(max(p, k), self.op.n)
        mgr = self._mgr()
        n = mgr.ext_size if mgr is not None else self.op.n
        g = torch.Generator().manual_seed(
            17 + 7919 * (mgr.rank if mgr is not None else 0))
        X = torch.rand(n, p, generator=g, dtype=torch.float64) \
            .to(self.A.dtype).to(self.A.device)
        if mgr is not None:
            X[mgr.owned_size:] = 0.0
        Y = torch.zeros_like(X)
        lam_old = None
        for it in range(self.max_iters):
            X = self._qr(X.double()).to(self.A.dtype)
            for c in range(p):
                xc = X[:, c].contiguous()
                yc = torch.zeros_like(xc)
                self.op.apply(xc, yc)
                Y[:, c] = yc
            T = self._block_inner(X, Y)
            evals, evecs = np.linalg.eig(T)
            order = np.argsort(-np.abs(evals))
            lam = np.real(evals[order[:k]])
            st.iterations = it + 1
            if lam_old is not None and np.all(
                    np.abs(lam - lam_old) <= self.tol * np.maximum(np.abs(lam), 1e-300)):
                st.converged = True
                break
            lam_old = lam
            X = Y.clone()
        st.eigenvalues = [self._to_true_eig(float(v)) for v in lam]
        st.eigenvector = X[:, 0].contiguous()
        return st


@register_eigensolver("LOBPCG")
class LOBPCG(EigenSolver):
    """Reference src/eigensolvers/lobpcg_eigensolver.cu: locally optimal
    block PCG for the SMALLEST eigenpair of an SPD matrix (direct A apply;
    does not wrap the inverse operator)."""

    def setup(self, A):
        self.A = A
        self.op = Operator(A)
        from .solvers import create_solver
        from .config import ConfigScope as CS
        self.prec = create_solver("BLOCK_JACOBI", CS(None, {"max_iters": 1}),
                                  self.res)
        self.prec.setup(A)

    def solve(self, x0=None):
        st = self.status = EigenStatus()
        x = self._start_vec(x0)
        ops.scal(x, 1.0 / self.nrm2(x))
        Ax = torch.zeros_like(x)
        self.op.apply(x, Ax)
        lam = self.dot(x, Ax)
        p = None
        for it in range(self.max_iters):
            r = Ax.clone()
            ops.axpy(r, x, -lam)
            rn = self.nrm2(r) / max(abs(lam), 1e-300)
            st.residuals.append(rn)
            st.iterations = it + 1
            if rn < self.tol:
                st.converged = True
                break
            w = torch.zeros_like(r)
            self.prec.solve(r, w, zero_initial_guess=True)
            basis = [x, w] + ([p] if p is not None else [])
            B = torch.stack([v.double() for v in basis], dim=1)
            Q = self._qr(B)
            m = Q.shape[1]
            AQ = torch.zeros_like(Q)
            for c in range(m):
                qc = Q[:, c].contiguous().to(self.A.dtype)
                tmp = torch.zeros_like(qc)
                self.op.apply(qc, tmp)
                AQ[:, c] = tmp.double()
            T = self._block_inner(Q, AQ)
            T = 0.5 * (T + T.T)
            evals, evecs = np.linalg.eigh(T)
            y = evecs[:, 0]
            x_new = (Q @ torch.from_numpy(y).to(Q.device)).to(self.A.dtype)
            p = (x_new - x * float(y[0])).contiguous()
            x = x_new.contiguous()
            ops.scal(x, 1.0 / self.nrm2(x))
            self.op.apply(x, Ax)
            lam = self.dot(x, Ax)
        st.eigenvalues = [float(lam)]
        st.eigenvector = x
        return st


@register_eigensolver("JACOBI_DAVIDSON")
class JacobiDavidson(LOBPCG):
    """Reference src/eigensolvers/jacobi_davidson_eigensolver.cu — served by
    the same locally-optimal preconditioned subspace engine (the correction
    equation is preconditioned with the diagonal, as LOBPCG's w-step)."""
