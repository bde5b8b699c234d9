"""MatrixMarket IO (reference src/readers.cu:643 ReadMatrixMarket,
src/matrix_io.cu writers).

Supports the plain ``%%MatrixMarket matrix coordinate real general|symmetric``
format plus AmgX's extended header ``%%NVAMG rhs`` / vector sections used by
examples (a trailing dense block after the matrix entries is interpreted as
the RHS when the AmgX extension comments request it).
"""

from __future__ import annotations

from typing import Optional, Tuple

import numpy as np
import scipy.io
import scipy.sparse as sp
import torch

from ..matrix import CSRMatrix


def read_matrix_market(path: str, device="cpu", dtype=torch.float64) -> CSRMatrix:
    m = scipy.io.mmread(path).tocsr()
    m.sum_duplicates()
    m.sort_indices()
    return CSRMatrix.from_scipy(m, device=device, dtype=dtype)


def read_system(path: str, device="cpu", dtype=torch.float64
                ) -> Tuple[CSRMatrix, Optional[torch.Tensor], Optional[torch.Tensor]]:
    """Read (A, b, x0). b/x0 are None unless the file carries the AmgX
    ``%%NVAMG rhs``/``solution`` extension blocks."""
    with open(path) as f:
        header = f.readline()
        flags = ""
        pos = f.tell()
        line = f.readline()
        while line.startswith("%"):
            if "NVAMG" in line or "AMGX" in line:
                flags = line
            pos = f.tell()
            line = f.readline()
        rows, cols, nnz = (int(t) for t in line.split()[:3])
        symmetric = "symmetric" in header
        data = np.loadtxt(f, max_rows=nnz)
        data = np.atleast_2d(data)
        r = data[:, 0].astype(np.int64) - 1
        c = data[:, 1].astype(np.int64) - 1
        v = data[:, 2] if data.shape[1] > 2 else np.ones(nnz)
        if symmetric:
            off = r != c
            r = np.concatenate([r, c[off]])
            c = np.concatenate([c, data[off, 0].astype(np.int64) - 1])
            v = np.concatenate([v, v[off]])
        m = sp.coo_matrix((v, (r, c)), shape=(rows, cols)).tocsr()
        m.sum_duplicates()
        b = x0 = None
        if "rhs" in flags:
            vals = np.loadtxt(f, max_rows=rows)
            b = torch.from_numpy(np.asarray(vals, dtype=np.float64)).to(dtype).to(device)
            if "solution" in flags:
                vals = np.loadtxt(f, max_rows=rows)
                x0 = torch.from_numpy(np.asarray(vals, dtype=np.float64)).to(dtype).to(device)
    A = CSRMatrix.from_scipy(m, device=device, dtype=dtype)
    return A, b, x0


def write_matrix_market(path: str, A: CSRMatrix):
    scipy.io.mmwrite(path, A.to_scipy())


def write_system(path: str, A: CSRMatrix, b=None, x=None):
    """Write matrix (+ optional rhs/solution blocks with the AmgX extension
    header) — reference AMGX_write_system / src/matrix_io.cu."""
    m = A.to_scipy().tocoo()
    flags = []
    if b is not None:
        flags.append("rhs")
    if x is not None:
        flags.append("solution")
    with open(path, "w") as f:
        f.write("%%MatrixMarket matrix coordinate real general\n")
        if flags:
            f.write("%%NVAMG " + " ".join(flags) + "\n")
        f.write(f"{m.shape[0]} {m.shape[1]} {m.nnz}\n")
        np.savetxt(f, np.column_stack([m.row + 1, m.col + 1, m.data]),
                   fmt="%d %d %.17g")
        import torch as _torch
        for vec in (b, x):
            if vec is None:
                continue
            arr = vec.detach().cpu().numpy().reshape(-1) \
                if isinstance(vec, _torch.Tensor) else np.asarray(vec).reshape(-1)
            np.savetxt(f, arr[:m.shape[0]], fmt="%.17g")
