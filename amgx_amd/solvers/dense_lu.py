"""Dense coarse solver (reference src/solvers/dense_lu_solver.cu:514-580).

The reference densifies the (consolidated) coarse matrix and runs cuSolverDn
getrf/getrs. MI355X-native: the coarse problem is tiny (<= dense_lu_num_rows,
default 128), so we densify on host at setup, invert once (LAPACK via numpy on
the HOST — setup-time only), keep the inverse resident on device, and apply it
as one small GEMV kernel per coarse solve. No rocSOLVER/rocBLAS in the solve
path, and the solve is graph-capturable (no host sync).
"""

from __future__ import annotations

import numpy as np
import torch

from .. import ops
from .base import Solver, register_solver


@register_solver("DENSE_LU_SOLVER")
class DenseLUSolver(Solver):
    is_smoother = True   # no residual monitoring; direct solve

    def solver_setup(self):
        A = self.A
        m = A.to_scipy()
        dense = np.asarray(m.todense(), dtype=np.float64)
        n = dense.shape[0]
        if n == 0:
            self.Ainv = torch.zeros((0, 0), dtype=torch.float64, device=A.device)
            return
        # regularize exactly singular coarse systems (all-Neumann problems)
        try:
            inv = np.linalg.inv(dense)
        except np.linalg.LinAlgError:
            inv = np.linalg.pinv(dense)
        if not np.isfinite(inv).all():
            inv = np.linalg.pinv(dense)
        self.Ainv = torch.from_numpy(np.ascontiguousarray(inv)) \
            .to(A.values.dtype).to(A.device)

    def solve(self, b, x, zero_initial_guess=False):
        ops.dense_solve(self.Ainv, b, x)
        st = self.status
        st.status = st.SUCCESS
        return st

    def solve_iteration(self, b, x):
        ops.dense_solve(self.Ainv, b, x)
        return True


class GatheredDenseLU:
    """Distributed coarse solver: gather the GLOBAL coarse problem on every
    rank once at setup (reference exact_coarse_solve, src/core.cu:347 +
    consolidation, SURVEY.md §2.3 item 4 — with 288 GB HBM3E the coarse grid
    is replicated instead of consolidated), invert on host, then each coarse
    solve is: one small allreduce of b, one on-device GEMV, slice owned."""

    def __init__(self, resources):
        self.res = resources

    def setup(self, A):
        import torch.distributed as tdist
        mgr = A.manager
        self.mgr = mgr
        n_g = mgr.global_rows(mgr.n_local)
        self.n_global = int(n_g)
        # my rows' global ids (rows were renumbered interior-first)
        perm = mgr.row_perm.cpu().numpy()
        my_rows_global = mgr.row_start + perm            # new local i -> global
        ro = A.row_offsets.cpu().numpy()
        ci = A.col_indices.cpu().numpy()
        va = A.values.cpu().numpy()
        # columns: local -> global
        col_global = np.empty(A.n_cols, dtype=np.int64)
        col_global[:mgr.n_local] = my_rows_global
        if mgr.n_halo:
            col_global[mgr.n_local:] = mgr.halo_global
        b = A.block_dim
        triples = (my_rows_global, ro, ci, va, col_global)
        gathered = [None] * mgr.world
        tdist.all_gather_object(gathered, triples)
        nb = self.n_global * b
        dense = np.zeros((nb, nb), dtype=np.float64)
        for rows_g, ro_r, ci_r, va_r, colg in gathered:
            for i in range(len(ro_r) - 1):
                gi = rows_g[i]
                for k in range(ro_r[i], ro_r[i + 1]):
                    gj = colg[ci_r[k]]
                    if b == 1:
                        dense[gi, gj] += va_r[k]
                    else:
                        dense[gi * b:(gi + 1) * b, gj * b:(gj + 1) * b] += va_r[k]
        try:
            inv = np.linalg.inv(dense)
        except np.linalg.LinAlgError:
            inv = np.linalg.pinv(dense)
        if not np.isfinite(inv).all():
            inv = np.linalg.pinv(dense)
        self.Ainv = torch.from_numpy(np.ascontiguousarray(inv)) \
            .to(A.values.dtype).to(A.device)
        self.my_rows_global = torch.from_numpy(my_rows_global).to(A.device)
        self._bg = torch.zeros(nb, dtype=A.values.dtype, device=A.device
                               if A.device.type == "cuda" else "cpu")
        self._xg = torch.zeros_like(self._bg)
        self.block_dim = b

    def solve(self, bvec, x, zero_initial_guess=True):
        import torch.distributed as tdist
        mgr = self.mgr
        b = self.block_dim
        self._bg.zero_()
        if b == 1:
            self._bg[self.my_rows_global.to(self._bg.device)] = \
                bvec.reshape(-1)[:mgr.owned_size].to(self._bg.device)
        else:
            idx = (self.my_rows_global.to(torch.int64)[:, None] * b
                   + torch.arange(b, device=self._bg.device)[None, :]).reshape(-1)
            self._bg[idx] = bvec.reshape(-1)[:mgr.owned_size].to(self._bg.device)
        tdist.all_reduce(self._bg, op=tdist.ReduceOp.SUM)
        ops.dense_solve(self.Ainv.to(self._bg.device), self._bg, self._xg)
        if b == 1:
            x.reshape(-1)[:mgr.owned_size] = \
                self._xg[self.my_rows_global.to(self._xg.device)].to(x.device)
        else:
            idx = (self.my_rows_global.to(torch.int64)[:, None] * b
                   + torch.arange(b, device=self._xg.device)[None, :]).reshape(-1)
            x.reshape(-1)[:mgr.owned_size] = self._xg[idx].to(x.device)
        return None
