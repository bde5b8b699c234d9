import numpy as np
import pytest
import scipy.sparse as sp
import torch

from amgx_amd import CSRMatrix, ops
from amgx_amd.amg.coloring import MatrixColoring
from amgx_amd.config import ConfigScope
from amgx_amd.problems import poisson_2d, poisson_3d, block_laplacian


def rand_csr(n=60, density=0.08, seed=3):
    rng = np.random.RandomState(seed)
    m = sp.random(n, n, density=density, random_state=rng, format="csr")
    m = m + m.T + sp.identity(n) * 4.0
    m = m.tocsr()
    m.sum_duplicates()
    m.sort_indices()
    return CSRMatrix.from_scipy(m)


def test_spmv_matches_scipy():
    A = rand_csr()
    x = torch.rand(A.n_rows, dtype=torch.float64)
    y = ops.spmv(A, x)
    ref = A.to_scipy() @ x.numpy()
    assert np.allclose(y.numpy(), ref)


def test_spmv_alpha_beta_window():
    A = rand_csr()
    x = torch.rand(A.n_rows, dtype=torch.float64)
    y = torch.rand(A.n_rows, dtype=torch.float64)
    y0 = y.clone()
    ops.spmv(A, x, y, alpha=2.0, beta=-0.5, row_begin=10, row_end=30)
    ref = 2.0 * (A.to_scipy() @ x.numpy()) - 0.5 * y0.numpy()
    assert np.allclose(y[10:30].numpy(), ref[10:30])
    assert np.allclose(y[:10].numpy(), y0[:10].numpy())


def test_block_spmv():
    A = block_laplacian(5, 4, block_dim=3)
    x = torch.rand(A.n_rows * 3, dtype=torch.float64)
    y = ops.spmv(A, x)
    ref = A.to_scipy() @ x.numpy()
    assert np.allclose(y.numpy(), ref)


def test_residual_and_norms():
    A = rand_csr()
    x = torch.rand(A.n_rows, dtype=torch.float64)
    b = torch.rand(A.n_rows, dtype=torch.float64)
    r = ops.residual(A, x, b)
    ref = b.numpy() - A.to_scipy() @ x.numpy()
    assert np.allclose(r.numpy(), ref)
    assert np.isclose(ops.nrm2(r), np.linalg.norm(ref))
    assert np.isclose(ops.nrm1(r), np.abs(ref).sum())
    assert np.isclose(ops.nrmmax(r), np.abs(ref).max())


def test_diag_ops():
    A = rand_csr()
    d = A.diagonal()
    assert np.allclose(d.numpy(), A.to_scipy().diagonal())
    dinv = ops.jacobi_dinv(A)
    assert np.allclose(dinv.numpy(), 1.0 / A.to_scipy().diagonal())


def test_coloring_valid():
    A = poisson_2d(12, 9)
    col = MatrixColoring.create(A)
    assert col.validate(A)
    assert 2 <= col.num_colors <= 6
    total = sum(col.rows_of(c).numel() for c in range(col.num_colors))
    assert total == A.n_rows


def test_size2_matching():
    A = poisson_2d(10, 10)
    agg, num = ops.size2_matching(A)
    assert agg.min() >= 0 and agg.max() == num - 1
    # aggregates should be mostly pairs: strictly fewer than n, more than n/4
    assert A.n_rows // 4 <= num <= A.n_rows * 3 // 4


def test_galerkin_aggregation_matches_ptap():
    A = poisson_2d(8, 8)
    agg, num = ops.size2_matching(A)
    Ac = ops.galerkin_aggregation(A, agg, num)
    n = A.n_rows
    P = sp.csr_matrix((np.ones(n), (np.arange(n), agg.numpy())), shape=(n, num))
    ref = (P.T @ A.to_scipy() @ P).toarray()
    assert np.allclose(Ac.to_scipy().toarray(), ref)


def test_restrict_prolongate():
    A = poisson_2d(6, 6)
    agg, num = ops.size2_matching(A)
    r = torch.rand(A.n_rows, dtype=torch.float64)
    rc = ops.restrict_agg(r, agg, num)
    n = A.n_rows
    P = sp.csr_matrix((np.ones(n), (np.arange(n), agg.numpy())), shape=(n, num))
    assert np.allclose(rc.numpy(), P.T @ r.numpy())
    x = torch.zeros(n, dtype=torch.float64)
    xc = torch.rand(num, dtype=torch.float64)
    ops.prolongate_agg(x, xc, agg)
    assert np.allclose(x.numpy(), P @ xc.numpy())


def test_spgemm_transpose_rap():
    A = rand_csr(40)
    B = rand_csr(40, seed=5)
    C = ops.spgemm(A, B)
    assert np.allclose(C.to_scipy().toarray(),
                       (A.to_scipy() @ B.to_scipy()).toarray())
    At = ops.transpose(A)
    assert np.allclose(At.to_scipy().toarray(), A.to_scipy().T.toarray())


def test_truncate_preserves_rowsum():
    A = rand_csr(30)
    T = ops.truncate_rows(A, trunc_factor=0.5)
    rs_old = np.asarray(A.to_scipy().sum(axis=1)).ravel()
    rs_new = np.asarray(T.to_scipy().sum(axis=1)).ravel()
    assert np.allclose(rs_old, rs_new)
    assert T.nnz <= A.nnz


def test_strength_and_pmis():
    A = poisson_2d(10, 10)
    from amgx_amd.ops import cpu
    S = cpu.strength_ahat(A, 0.25, 1.1)
    # poisson: all off-diagonals are strong at theta=0.25
    offdiag = A.nnz - A.n_rows
    assert int(S.sum()) == offdiag
    cf, nc = cpu.pmis_select(A, S)
    assert 0 < nc < A.n_rows
    assert (cf.numpy() >= 0).sum() == nc


def test_interp_d1_partition_of_unity():
    A = poisson_2d(10, 10)
    from amgx_amd.ops import cpu
    S = cpu.strength_ahat(A, 0.25, 1.1)
    cf, nc = cpu.pmis_select(A, S)
    P = cpu.interp_d1(A, S, cf, nc)
    # interior F rows of constant-row-sum Laplacian interpolate to ~1
    rowsum = np.asarray(P.to_scipy().sum(axis=1)).ravel()
    cfn = cf.numpy()
    assert np.allclose(rowsum[cfn >= 0], 1.0)
    assert P.nnz > 0
