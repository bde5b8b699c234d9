"""Solver base class, convergence objects and the solver factory registry.

Mirrors the reference ``Solver<TConfig>`` services (include/solvers/solver.h:22,
src/solvers/solver.cu:333,586): setup/solve lifecycle, residual monitoring,
norm machinery, convergence delegation, nested preconditioner composition, and
the name->factory registry (reference src/core.cu:596-628) that makes every
algorithm addressable from JSON configs.
"""

from __future__ import annotations

import math
import time
from typing import Dict, Optional, Type

import torch

from .. import ops
from ..config import ConfigScope
from ..resources import Resources, default_resources
from ..output import amgx_output

SOLVER_REGISTRY: Dict[str, Type["Solver"]] = {}


def register_solver(name: str):
    def deco(cls):
        SOLVER_REGISTRY[name] = cls
        cls.solver_name = name
        return cls
    return deco


def create_solver(name_or_scope, scope: Optional[ConfigScope] = None,
                  resources: Optional[Resources] = None) -> "Solver":
    """Create a solver by registry name (reference SolverFactory::allocate)."""
    if isinstance(name_or_scope, ConfigScope):
        scope = name_or_scope
        name = scope.get("solver")
    else:
        name = name_or_scope
        if scope is None:
            scope = ConfigScope(None, {})
    if resources is None:
        resources = default_resources()
    cls = SOLVER_REGISTRY.get(name)
    if cls is None:
        raise KeyError(f"unknown solver {name!r}; known: {sorted(SOLVER_REGISTRY)}")
    return cls(scope, resources)


# ------------------------------------------------------------------- convergence
class Convergence:
    """Reference include/convergence/convergence.h:64-101 + src/convergence/."""

    def __init__(self, kind: str, tolerance: float, alt_rel_tol: float = -1.0):
        self.kind = kind
        self.tol = tolerance
        self.alt_rel_tol = alt_rel_tol
        self.ini_norm = None

    def set_initial(self, nrm: float):
        self.ini_norm = nrm

    def converged(self, nrm: float) -> bool:
        if self.kind == "ABSOLUTE":
            return nrm <= self.tol
        if self.kind in ("RELATIVE_INI", "RELATIVE_INI_CORE"):
            return nrm <= self.tol * (self.ini_norm if self.ini_norm else 1.0)
        if self.kind in ("RELATIVE_MAX", "RELATIVE_MAX_CORE"):
            ref = max(self.ini_norm or 0.0, 1e-300)
            return nrm <= self.tol * ref
        if self.kind == "COMBINED_REL_INI_ABS":
            rel = nrm <= self.tol * (self.ini_norm if self.ini_norm else 1.0)
            ab = self.alt_rel_tol > 0 and nrm <= self.alt_rel_tol
            return rel or ab
        raise ValueError(f"unknown convergence {self.kind}")


class SolveStatus:
    SUCCESS = 0
    DIVERGED = 1
    NOT_CONVERGED = 2

    def __init__(self):
        self.status = SolveStatus.NOT_CONVERGED
        self.iterations = 0
        self.residuals = []
        self.setup_time = 0.0
        self.solve_time = 0.0

    @property
    def converged(self):
        return self.status == SolveStatus.SUCCESS

    def __repr__(self):  # pragma: no cover
        return (f"SolveStatus(conv={self.converged}, it={self.iterations}, "
                f"res={self.residuals[-1] if self.residuals else None})")


# ------------------------------------------------------------------------ solver
class Solver:
    """Base solver. Subclasses implement solver_setup / solve_init /
    solve_iteration / solve_finalize (reference include/solvers/solver.h:152-156)."""

    solver_name = "?"
    is_smoother = False   # smoothers skip convergence monitoring by default
    print_cb = None       # per-solver output redirection (C API hook)

    def _out(self, msg: str) -> None:
        if self.print_cb is not None:
            try:
                self.print_cb(msg)
                return
            except Exception:
                pass
        amgx_output(msg)

    def __init__(self, scope: ConfigScope, resources: Resources):
        self.scope = scope
        self.res = resources
        self.A = None
        self.max_iters = scope.get("max_iters")
        self.monitor_residual = bool(scope.get("monitor_residual"))
        self.store_res_history = bool(scope.get("store_res_history"))
        self.print_solve_stats = bool(scope.get("print_solve_stats"))
        self.norm = scope.get("norm")
        self.min_iters = int(scope.get("min_iters") or 0)
        self.convergence = Convergence(scope.get("convergence"),
                                       scope.get("tolerance"),
                                       scope.get("alt_rel_tolerance"))
        self.relaxation_factor = scope.get("relaxation_factor")
        self.status = SolveStatus()

    # -- lifecycle -----------------------------------------------------------
    def setup(self, A):
        t0 = time.perf_counter()
        self.A = A
        self.scaler = None
        scaling = self.scope.get("scaling")
        if scaling and scaling != "NONE":
            # reference lifecycle (src/solvers/solver.cu:443-476): scale the
            # matrix around preconditioner setup, restore it afterwards;
            # solve() re-scales around the iteration (':667-676, :853-880)
            from ..scalers import create_scaler
            self.scaler = create_scaler(scaling)
            self.scaler.setup(A)
            self.scaler.scale_matrix(A)
            try:
                self.solver_setup()
            finally:
                self.scaler.unscale_matrix(A)
        else:
            self.solver_setup()
        self.res.synchronize() if self.res.is_cuda else None
        self.status.setup_time = time.perf_counter() - t0

    def resetup(self, A):
        """Values changed, structure identical (reference AMGX_solver_resetup)."""
        self.setup(A)

    def solver_setup(self):
        pass

    # -- norms ---------------------------------------------------------------
    def _owned(self, x: torch.Tensor) -> torch.Tensor:
        """Owned prefix of a (possibly halo-extended) distributed vector
        (reference ViewType OWNED, include/vector.h:18-27)."""
        mgr = getattr(self.A, "manager", None) if self.A is not None else None
        if mgr is not None:
            return x.reshape(-1)[:mgr.owned_size]
        return x

    def compute_norm(self, r: torch.Tensor) -> float:
        r = self._owned(r)
        bd = self.A.block_dim if self.A is not None else 1
        if bd > 1 and not self.scope.get("use_scalar_norm"):
            # blocked norms (reference src/norm.cu per-block-component
            # norms): converge on the worst component
            import torch.distributed as tdist
            comp = r.reshape(-1, bd).abs().double()
            mgr = getattr(self.A, "manager", None)
            if self.norm == "L1":
                per = comp.sum(0)
                if mgr is not None:
                    tdist.all_reduce(per)
            elif self.norm == "L1_SCALED":
                # per-component L1 divided by the per-component global length
                per = comp.sum(0)
                if mgr is not None:
                    tdist.all_reduce(per)
                    per = per / max(mgr.n_global, 1)
                else:
                    per = per / max(comp.shape[0], 1)
            elif self.norm == "LMAX":
                per = comp.amax(0) if comp.numel() else \
                    torch.zeros(bd, dtype=torch.float64, device=r.device)
                if mgr is not None:
                    tdist.all_reduce(per, op=tdist.ReduceOp.MAX)
            else:                      # L2 default
                per = (comp * comp).sum(0)
                if mgr is not None:
                    tdist.all_reduce(per)
                per = per.sqrt()
            return float(per.max())
        if self.norm == "L2":
            nrm = ops.nrm2(r)
        elif self.norm == "L1":
            nrm = ops.nrm1(r)
        elif self.norm == "L1_SCALED":
            # reference L1_SCALED: L1 norm divided by the global length
            n = r.numel()
            mgr = getattr(self.A, "manager", None) if self.A is not None \
                else None
            if mgr is not None:
                n = mgr.n_global * mgr.block_dim
            nrm = ops.nrm1(r) / max(n, 1)
        elif self.norm == "LMAX":
            nrm = ops.nrmmax(r)
        else:
            nrm = ops.nrm2(r)
        if self.A is not None and getattr(self.A, "manager", None) is not None:
            nrm = self.A.manager.global_norm(nrm, self.norm)
        return nrm

    def dot(self, x, y) -> float:
        d = ops.dot(self._owned(x), self._owned(y))
        if self.A is not None and getattr(self.A, "manager", None) is not None:
            d = self.A.manager.global_sum(d)
        return d

    # -- solve ---------------------------------------------------------------
    def solve(self, b: torch.Tensor, x: torch.Tensor,
              zero_initial_guess: bool = False) -> SolveStatus:
        """Reference Solver::solve (src/solvers/solver.cu:586): scaler hooks,
        initial residual + norm, solve_init, iterate until
        converged/max_iters."""
        if getattr(self, "scaler", None) is not None:
            # reference Solver::solve (:667-676): scale A and b, map x into
            # the scaled space; at exit (:853-861) restore A and map x back
            self.scaler.scale_matrix(self.A)
            bs = self.scaler.scale_rhs(b)
            if not zero_initial_guess:
                self.scaler.scale_guess(x)
            try:
                st = self._solve_inner(bs, x, zero_initial_guess)
            finally:
                self.scaler.unscale_solution(x)
                self.scaler.unscale_matrix(self.A)
            return st
        return self._solve_inner(b, x, zero_initial_guess)

    def _solve_inner(self, b: torch.Tensor, x: torch.Tensor,
                     zero_initial_guess: bool = False) -> SolveStatus:
        st = self.status = SolveStatus()
        t0 = time.perf_counter()
        if zero_initial_guess:
            x.zero_()
        monitoring = self.monitor_residual
        if monitoring:
            r = ops.residual(self.A, x, b)
            nrm = self.compute_norm(r)
            self.convergence.set_initial(nrm)
            st.residuals.append(nrm)
            if self.print_solve_stats and self.res.rank == 0:
                self._out(f"           iter      residual   rate" "\n")
                self._out(f"           ----------------------------" "\n")
                self._out(f"            Ini {nrm:14.6e}" "\n")
            if self.convergence.converged(nrm) and self.convergence.kind != "RELATIVE_INI":
                st.status = SolveStatus.SUCCESS
                st.solve_time = time.perf_counter() - t0
                return st
            if nrm == 0.0:
                st.status = SolveStatus.SUCCESS
                st.solve_time = time.perf_counter() - t0
                return st
        self.solve_init(b, x, zero_initial_guess)
        for it in range(self.max_iters):
            done = self.solve_iteration(b, x)
            st.iterations = it + 1
            if monitoring:
                nrm = self.last_residual_norm(b, x)
                st.residuals.append(nrm)
                if self.print_solve_stats and self.res.rank == 0:
                    rate = (st.residuals[-1] / st.residuals[-2]
                            if st.residuals[-2] else 0.0)
                    self._out(f"           {it:4d} {nrm:14.6e}  {rate:6.4f}" "\n")
                if self.convergence.converged(nrm) \
                        and it + 1 >= self.min_iters:
                    st.status = SolveStatus.SUCCESS
                    break
                if not math.isfinite(nrm):
                    st.status = SolveStatus.DIVERGED
                    break
                # explicit divergence check (reference rel_div_tolerance)
                rdt = self.scope.get("rel_div_tolerance")
                if (rdt is not None and rdt > 0 and st.residuals
                        and nrm > rdt * st.residuals[0]):
                    st.status = SolveStatus.DIVERGED
                    break
            if done:
                if not monitoring:
                    st.status = SolveStatus.SUCCESS
                break
        self.solve_finalize(b, x)
        if self.res.is_cuda:
            self.res.synchronize()
        st.solve_time = time.perf_counter() - t0
        if (self.print_solve_stats and self.res.rank == 0 and monitoring
                and st.residuals):
            # reference Solver::print_final (src/solvers/solver.cu:895-934)
            ini, last = st.residuals[0], st.residuals[-1]
            its = max(st.iterations, 1)
            rate = (last / ini) ** (1.0 / its) if ini > 0 else ini
            self._out(f"           ----------------------------" "\n")
            self._out(f"         Total Iterations: {st.iterations}" "\n")
            self._out(f"         Avg Convergence Rate: {rate:15.4f}" "\n")
            self._out(f"         Final Residual: {last:15.6e}" "\n")
            self._out(f"         Total Reduction in Residual: "
                        f"{(last / ini if ini > 0 else ini):15.6e}\n")
            if self.scope.get("obtain_timings"):
                self._out(f"         Total Time: {st.setup_time + st.solve_time:10.4f} s" "\n")
                self._out(f"             setup: {st.setup_time:10.4f} s" "\n")
                self._out(f"             solve: {st.solve_time:10.4f} s" "\n")
                self._out(f"             solve(per iteration): "
                            f"{st.solve_time / its:10.6f} s\n")
        return st

    # default residual-norm recomputation; Krylov solvers override with their
    # internally tracked norm to avoid an extra SpMV
    def last_residual_norm(self, b, x) -> float:
        r = ops.residual(self.A, x, b)
        return self.compute_norm(r)

    def solve_init(self, b, x, zero_initial_guess):
        pass

    def solve_iteration(self, b, x) -> bool:
        raise NotImplementedError

    def solve_finalize(self, b, x):
        pass

    # -- helpers --------------------------------------------------------------
    def make_preconditioner(self, role: str = "preconditioner",
                            default_name: str = "NOSOLVER"):
        name, sub = self.scope.sub_solver(role, default_name)
        if name is None or name == "NOSOLVER":
            return None
        solver = create_solver(name, sub, self.res)
        return solver

    def new_vec(self, like: torch.Tensor) -> torch.Tensor:
        return torch.zeros_like(like)


@register_solver("USER")
class UserSolver(Solver):
    """Application-plugin solver (reference src/solvers/user_solver.cu):
    each iteration calls a user-registered callback(A, b, x) that updates x
    in place; convergence is monitored by the normal machinery. Set the
    callback with ``UserSolver.set_callback(fn)`` (process-global, like the
    reference's setCallback) or pass one per-instance via
    ``solver.callback = fn`` before solve. ``register_solver`` remains the
    richer plugin path for full custom solver classes."""
    is_smoother = False
    _global_callback = None

    @classmethod
    def set_callback(cls, fn):
        cls._global_callback = fn

    def __init__(self, scope, resources):
        super().__init__(scope, resources)
        self.callback = None

    def solve_iteration(self, b, x):
        cb = self.callback or type(self)._global_callback
        if cb is None:
            raise RuntimeError(
                "USER solver has no callback: call "
                "amgx_amd.solvers.UserSolver.set_callback(fn) first")
        cb(self.A, b, x)
        return False


@register_solver("NOSOLVER")
@register_solver("DUMMY")
class DummySolver(Solver):
    """Identity preconditioner (reference src/solvers/dummy_solver.cu)."""
    is_smoother = True

    def solve(self, b, x, zero_initial_guess=False):
        if zero_initial_guess:
            x.zero_()
        x.copy_(b.reshape(x.shape))
        st = SolveStatus()
        st.status = SolveStatus.SUCCESS
        return st

    def solve_iteration(self, b, x):
        return True
