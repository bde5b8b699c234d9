"""System pre-scalers (reference src/scalers/: BINORMALIZATION (446 LoC),
NBINORMALIZATION, DIAGONAL_SYMMETRIC; hooked into Solver::solve,
src/solvers/solver.cu:667-676).

Symmetric scaling A' = D A D with  A' x' = b',  x' = D^{-1} x,  b' = D b.
The solver scales the matrix once at setup and maps b/x per solve.
"""

from __future__ import annotations

import numpy as np
import torch



class Scaler:
    name = "NONE"

    def __init__(self):
        self.d = None

    def setup(self, A):
        raise NotImplementedError

    def scale_matrix(self, A):
        """A <- D A D in place (values only). Distributed: halo columns use
        the OWNER's d (one halo exchange) so every rank scales the same
        global operator."""
        d = self.d
        rows = torch.repeat_interleave(
            torch.arange(A.n_rows, device=A.device),
            (A.row_offsets[1:] - A.row_offsets[:-1]).to(torch.int64))
        dr = d[rows]
        mgr = getattr(A, "manager", None)
        if mgr is not None and mgr.n_halo:
            ext = mgr.new_ext_vec(torch.float64)
            ext[:mgr.owned_size] = d.reshape(-1).to(torch.float64)
            mgr.exchange_halo(ext, block_override=1)
            dcol_src = ext.to(d.dtype)
        else:
            dcol_src = torch.ones(A.n_cols, dtype=d.dtype, device=d.device)
            dcol_src[:d.numel()] = d
        dc = dcol_src[A.col_indices.to(torch.int64)]
        if A.block_dim == 1:
            A.values.mul_(dr * dc)
        else:
            A.values.mul_((dr * dc).view(-1, 1, 1))
        A.clear_cache()

    def unscale_matrix(self, A):
        """Restore A (divide the scaling back out; reference
        scaleMatrix(..., amgx::UNSCALE))."""
        d = self.d
        self.d = 1.0 / d
        try:
            self.scale_matrix(A)
        finally:
            self.d = d

    def scale_rhs(self, b):
        out = b.clone()
        out.reshape(-1)[:self.d.numel() * 1].mul_(self.d)
        return out

    def unscale_solution(self, y):
        y.reshape(-1)[:self.d.numel()].mul_(self.d)
        return y

    def scale_guess(self, x):
        x.reshape(-1)[:self.d.numel()].div_(self.d)
        return x


class DiagonalSymmetricScaler(Scaler):
    """D = |diag(A)|^{-1/2} (reference src/scalers/diagonal_symmetric_scaler.cu)."""
    name = "DIAGONAL_SYMMETRIC"

    def setup(self, A):
        if A.block_dim != 1:
            raise NotImplementedError(
                "scalers operate on scalar matrices (reference parity: "
                "src/scalers/* are 1x1-only)")
        d = A.diagonal().abs()
        d = torch.where(d > 0, d, torch.ones_like(d))
        self.d = (1.0 / torch.sqrt(d)).to(A.dtype)


class BinormalizationScaler(Scaler):
    """Livne-Golub style symmetric binormalization: find d with
    sum_j (a_ij d_i d_j)^2 ~= const for every row (reference
    src/scalers/binormalization_scaler.cu). Fixed-point sweeps on the
    squared matrix."""
    name = "BINORMALIZATION"

    def __init__(self, sweeps: int = 10):
        super().__init__()
        self.sweeps = sweeps

    def setup(self, A):
        if A.block_dim != 1:
            raise NotImplementedError(
                "scalers operate on scalar matrices (reference parity)")
        import scipy.sparse as sp
        m = A.to_scipy().tocsr()
        B = m.multiply(m)       # a_ij^2
        n = A.n_rows
        w = np.ones(n)
        for _ in range(self.sweeps):
            s = B @ w
            s = np.where(s > 0, s, 1.0)
            w = w / np.sqrt(np.sqrt(s))
            w *= n / w.sum()
        d = np.sqrt(w)
        self.d = torch.from_numpy(d).to(A.dtype).to(A.device)


class NBinormalizationScaler(BinormalizationScaler):
    name = "NBINORMALIZATION"


def create_scaler(name: str):
    return {"DIAGONAL_SYMMETRIC": DiagonalSymmetricScaler,
            "BINORMALIZATION": BinormalizationScaler,
            "NBINORMALIZATION": NBinormalizationScaler}[name]()
