"""Synthetic problem generators (reference: embedded CUSP
``gallery::poisson5/7/9/27pt`` used throughout the tests, and
``AMGX_generate_distributed_poisson_7pt``, src/amgx_c.cu:4566-4731)."""

from __future__ import annotations

import numpy as np
import scipy.sparse as sp
import torch

from .matrix import CSRMatrix


def poisson_2d(nx: int, ny: int, stencil: int = 5, device="cpu",
               dtype=torch.float64) -> CSRMatrix:
    ex = np.ones(nx)
    ey = np.ones(ny)
    Tx = sp.diags([-ex[:-1], 2 * ex, -ex[:-1]], [-1, 0, 1])
    Ty = sp.diags([-ey[:-1], 2 * ey, -ey[:-1]], [-1, 0, 1])
    Ix, Iy = sp.identity(nx), sp.identity(ny)
    A = sp.kron(Iy, Tx) + sp.kron(Ty, Ix)
    if stencil == 9:
        D = sp.diags([-ex[:-1], np.zeros(nx), -ex[:-1]], [-1, 0, 1])
        Dy = sp.diags([-ey[:-1], np.zeros(ny), -ey[:-1]], [-1, 0, 1])
        A = A + 0.5 * sp.kron(Dy, D)
    A = A.tocsr()
    A.sum_duplicates()
    A.eliminate_zeros()     # kron chains leave explicit zeros off-stencil
    return CSRMatrix.from_scipy(A, device=device, dtype=dtype)


def poisson_3d(nx: int, ny: int, nz: int, device="cpu",
               dtype=torch.float64) -> CSRMatrix:
    """7-point 3D Poisson: diag 6, neighbors -1 (matches
    AMGX_generate_distributed_poisson_7pt semantics, src/amgx_c.cu:4566).
    Direct vectorized stencil assembly (columns sorted per row), so the
    256^3 bench matrix builds in seconds."""
    dev = torch.device(device)
    if dev.type == "cuda":
        # assemble directly on the GPU (milliseconds at 256^3)
        n = nx * ny * nz
        lid = torch.arange(n, dtype=torch.int64, device=dev)
        x = lid % nx
        y = (lid // nx) % ny
        z = lid // (nx * ny)
        mask = torch.empty((n, 7), dtype=torch.bool, device=dev)
        mask[:, 0] = z > 0
        mask[:, 1] = y > 0
        mask[:, 2] = x > 0
        mask[:, 3] = True
        mask[:, 4] = x < nx - 1
        mask[:, 5] = y < ny - 1
        mask[:, 6] = z < nz - 1
        offs = torch.tensor([-nx * ny, -nx, -1, 0, 1, nx, nx * ny],
                            dtype=torch.int64, device=dev)
        ro = torch.zeros(n + 1, dtype=torch.int64, device=dev)
        torch.cumsum(mask.sum(1), 0, out=ro[1:])
        cols = (lid[:, None] + offs[None, :])[mask]
        valrow = torch.tensor([-1., -1., -1., 6., -1., -1., -1.],
                              dtype=dtype, device=dev)
        vals = valrow.expand(n, 7)[mask]
        return CSRMatrix(ro.to(torch.int32), cols.to(torch.int32),
                         vals.contiguous(), n_cols=n)
    ro, cols, vals, _ = poisson_3d_local(nx, ny, nz, 0, 1)
    return CSRMatrix(torch.from_numpy(ro.astype(np.int32)).to(dev),
                     torch.from_numpy(cols.astype(np.int32)).to(dev),
                     torch.from_numpy(vals).to(dtype).to(dev),
                     n_cols=nx * ny * nz)


def poisson_3d_27pt(nx: int, ny: int, nz: int, device="cpu",
                    dtype=torch.float64) -> CSRMatrix:
    """27-point 3D stencil (reference cusp::gallery::poisson27pt used by the
    unit tests): tensor-product of 1D [-1, 2, -1] graphs with all diagonal
    couplings, SPD."""
    def lap1(n):
        e = np.ones(n)
        return sp.diags([-e[:-1], 2 * e, -e[:-1]], [-1, 0, 1])

    def mass1(n):
        e = np.ones(n)
        return sp.diags([e[:-1] * 0.5, e, e[:-1] * 0.5], [-1, 0, 1])

    Lx, Ly, Lz = lap1(nx), lap1(ny), lap1(nz)
    Mx, My, Mz = mass1(nx), mass1(ny), mass1(nz)
    A = (sp.kron(sp.kron(Mz, My), Lx) + sp.kron(sp.kron(Mz, Ly), Mx)
         + sp.kron(sp.kron(Lz, My), Mx)).tocsr()
    A.sum_duplicates()
    A.eliminate_zeros()
    return CSRMatrix.from_scipy(A, device=device, dtype=dtype)


def poisson_3d_local(nx: int, ny: int, nz: int, rank: int, world: int):
    """Rank-local slab rows (z-partition) of a global nx*ny*(nz*world) 7-pt
    Poisson, as (row_offsets, col_indices GLOBAL, values, row_start) numpy.
    One process per GPU partitioning for the distributed path
    (reference AMGX_generate_distributed_poisson_7pt px*py*pz grid)."""
    NZg = nz * world
    n_local = nx * ny * nz
    row_start = rank * n_local
    # vectorized stencil assembly in row-major (n, 7) layout so the boolean
    # select emits rows in order with ascending columns — no transposes
    lid = np.arange(n_local, dtype=np.int64)
    x = lid % nx
    y = (lid // nx) % ny
    gz = lid // (nx * ny) + rank * nz
    gid = lid + row_start
    mask = np.empty((n_local, 7), dtype=bool)
    mask[:, 0] = gz > 0
    mask[:, 1] = y > 0
    mask[:, 2] = x > 0
    mask[:, 3] = True
    mask[:, 4] = x < nx - 1
    mask[:, 5] = y < ny - 1
    mask[:, 6] = gz < NZg - 1
    offs = np.asarray([-nx * ny, -nx, -1, 0, 1, nx, nx * ny], dtype=np.int64)
    deg = mask.sum(axis=1)
    ro = np.zeros(n_local + 1, dtype=np.int64)
    np.cumsum(deg, out=ro[1:])
    cols = (gid[:, None] + offs[None, :])[mask]
    valrow = np.asarray([-1.0, -1.0, -1.0, 6.0, -1.0, -1.0, -1.0])
    vals = np.broadcast_to(valrow, (n_local, 7))[mask]
    return ro, cols, vals, row_start


def block_laplacian(nx: int, ny: int, block_dim: int = 4, device="cpu",
                    dtype=torch.float64, seed: int = 0) -> CSRMatrix:
    """Block-CSR test system: 2D Poisson coupling with SPD random blocks on
    the diagonal (driver config #4 class: block-4 coupled system)."""
    A = poisson_2d(nx, ny).to_scipy().tocsr()
    rng = np.random.RandomState(seed)
    n = A.shape[0]
    nnz = A.nnz
    blocks = np.zeros((nnz, block_dim, block_dim))
    rows = np.repeat(np.arange(n), np.diff(A.indptr))
    for k in range(nnz):
        i, j, v = rows[k], A.indices[k], A.data[k]
        if i == j:
            Q = rng.randn(block_dim, block_dim) * 0.1
            blocks[k] = v * (np.eye(block_dim) + Q @ Q.T)
        else:
            blocks[k] = v * (np.eye(block_dim)
                             + 0.05 * rng.randn(block_dim, block_dim))
    return CSRMatrix.from_bsr(A.indptr.astype(np.int32),
                              A.indices.astype(np.int32), blocks,
                              n_cols=n, device=device, dtype=dtype)


def poisson_3d_local_device(nx: int, ny: int, nz: int, rank: int, world: int,
                            device="cpu", dtype=torch.float64):
    """poisson_3d_local with torch assembly on the target device (the
    multi-GPU bench path: each rank builds its slab on its own GPU).
    Returns (ro, cols_global, vals, row_start) tensors."""
    dev = torch.device(device)
    NZg = nz * world
    n = nx * ny * nz
    row_start = rank * n
    lid = torch.arange(n, dtype=torch.int64, device=dev)
    x = lid % nx
    y = (lid // nx) % ny
    gz = lid // (nx * ny) + rank * nz
    mask = torch.empty((n, 7), dtype=torch.bool, device=dev)
    mask[:, 0] = gz > 0
    mask[:, 1] = y > 0
    mask[:, 2] = x > 0
    mask[:, 3] = True
    mask[:, 4] = x < nx - 1
    mask[:, 5] = y < ny - 1
    mask[:, 6] = gz < NZg - 1
    offs = torch.tensor([-nx * ny, -nx, -1, 0, 1, nx, nx * ny],
                        dtype=torch.int64, device=dev)
    ro = torch.zeros(n + 1, dtype=torch.int64, device=dev)
    torch.cumsum(mask.sum(1), 0, out=ro[1:])
    cols = ((lid + row_start)[:, None] + offs[None, :])[mask]
    valrow = torch.tensor([-1., -1., -1., 6., -1., -1., -1.],
                          dtype=dtype, device=dev)
    vals = valrow.expand(n, 7)[mask].contiguous()
    return ro, cols, vals, row_start


def random_laplacian(n: int, avg_degree: int = 12, seed: int = 11,
                     device="cpu", dtype=torch.float64):
    """Unstructured SPD graph Laplacian (+I): the offline stand-in for the
    SuiteSparse/Florida unstructured matrices the reference benchmarks
    against (BASELINE.json config #5) — irregular row lengths and a
    scattered sparsity pattern stress ILU/halo paths the way a real
    unstructured mesh does."""
    import scipy.sparse as sp

    from .matrix import CSRMatrix
    rng = np.random.RandomState(seed)
    density = avg_degree / float(n)
    G = sp.random(n, n, density=density, random_state=rng, format="csr")
    G = G + G.T
    G.data[:] = np.abs(G.data)
    L = sp.diags(np.asarray(G.sum(1)).ravel()) - G + sp.identity(n)
    L = L.tocsr()
    L.sum_duplicates()
    L.sort_indices()
    L.eliminate_zeros()
    return CSRMatrix.from_scipy(L, dtype=dtype).to(device)


def anisotropic_2d(nx: int, ny: int, eps: float = 0.01, device="cpu",
                   dtype=torch.float64):
    """2D anisotropic diffusion -eps*u_xx - u_yy (5-pt): the classic
    strength-of-connection stress (reference CUSP generator role)."""
    import scipy.sparse as sp
    ex = np.ones(nx)
    ey = np.ones(ny)
    Tx = sp.diags([-ex[:-1] * eps, 2 * eps * ex, -ex[:-1] * eps],
                  [-1, 0, 1], shape=(nx, nx))
    Ty = sp.diags([-ey[:-1], 2 * ey, -ey[:-1]], [-1, 0, 1], shape=(ny, ny))
    L = (sp.kron(sp.identity(ny), Tx) + sp.kron(Ty, sp.identity(nx))).tocsr()
    L.sum_duplicates()
    L.sort_indices()
    L.eliminate_zeros()
    from .matrix import CSRMatrix
    return CSRMatrix.from_scipy(L, dtype=dtype).to(device)


def convection_diffusion_2d(nx: int, ny: int, beta: float = 20.0,
                            device="cpu", dtype=torch.float64):
    """2D convection-diffusion with upwind convection (NONSYMMETRIC): the
    BiCGStab/GMRES/IDR stress problem."""
    import scipy.sparse as sp
    h = 1.0 / (nx + 1)
    n = nx * ny
    rows, cols, vals = [], [], []
    for j in range(ny):
        for i in range(nx):
            r = j * nx + i
            diag = 4.0 + beta * h
            rows.append(r); cols.append(r); vals.append(diag)
            for di, dj, v in ((-1, 0, -1.0 - beta * h), (1, 0, -1.0),
                              (0, -1, -1.0), (0, 1, -1.0)):
                ii, jj = i + di, j + dj
                if 0 <= ii < nx and 0 <= jj < ny:
                    rows.append(r); cols.append(jj * nx + ii); vals.append(v)
    L = sp.csr_matrix((vals, (rows, cols)), shape=(n, n))
    L.sum_duplicates()
    L.sort_indices()
    from .matrix import CSRMatrix
    return CSRMatrix.from_scipy(L, dtype=dtype).to(device)
