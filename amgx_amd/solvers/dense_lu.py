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
