"""Property-based invariants over random sparse matrices (hypothesis).

Reference analogue: src/tests/generic_spmv.cu / csr_multiply.cu run against
randomly generated matrices; here the generator space is driven by
hypothesis so shrinking finds minimal failing structures.
"""

import numpy as np
import scipy.sparse as sp
import torch
from hypothesis import given, settings, strategies as st

from amgx_amd import ops
from amgx_amd.matrix import CSRMatrix


def _rand_csr(n, density, seed, spd=False):
    rng = np.random.RandomState(seed)
    m = sp.random(n, n, density=density, random_state=rng, format="csr")
    if spd:
        m = m + m.T + sp.identity(n) * (np.abs(m).sum(1).max() + 1.0)
    m = m.tocsr()
    m.sum_duplicates()
    m.sort_indices()
    return m


@settings(max_examples=25, deadline=None)
@given(n=st.integers(4, 120), density=st.floats(0.01, 0.3),
       seed=st.integers(0, 10_000))
def test_spmv_matches_scipy(n, density, seed):
    m = _rand_csr(n, density, seed)
    A = CSRMatrix.from_scipy(m)
    x = torch.from_numpy(np.random.RandomState(seed + 1).rand(n))
    y = ops.spmv(A, x)
    assert np.allclose(y.numpy(), m @ x.numpy(), rtol=1e-12, atol=1e-12)


@settings(max_examples=25, deadline=None)
@given(n=st.integers(4, 100), density=st.floats(0.02, 0.3),
       seed=st.integers(0, 10_000))
def test_transpose_involution(n, density, seed):
    A = CSRMatrix.from_scipy(_rand_csr(n, density, seed))
    Att = ops.transpose(ops.transpose(A))
    assert torch.equal(Att.row_offsets, A.row_offsets)
    assert torch.equal(Att.col_indices, A.col_indices)
    assert torch.allclose(Att.values, A.values)


@settings(max_examples=20, deadline=None)
@given(n=st.integers(8, 100), density=st.floats(0.02, 0.2),
       seed=st.integers(0, 10_000))
def test_coloring_always_valid(n, density, seed):
    from amgx_amd.amg.coloring import MatrixColoring
    A = CSRMatrix.from_scipy(_rand_csr(n, density, seed, spd=True))
    col = MatrixColoring.create(A)
    assert col.validate(A)
    assert int(col.colors.min()) >= 0


@settings(max_examples=20, deadline=None)
@given(n=st.integers(8, 80), density=st.floats(0.05, 0.3),
       seed=st.integers(0, 10_000),
       factor=st.floats(0.05, 0.9))
def test_truncate_preserves_row_sums(n, density, seed, factor):
    # reference truncate.cu: dropped mass is rescaled into survivors
    A = CSRMatrix.from_scipy(abs(_rand_csr(n, density, seed)) +
                             sp.identity(n))
    T = ops.truncate_rows(A, trunc_factor=factor)
    rs_a = np.asarray(A.to_scipy().sum(1)).ravel()
    rs_t = np.asarray(T.to_scipy().sum(1)).ravel()
    assert np.allclose(rs_a, rs_t, rtol=1e-10, atol=1e-12)
    assert T.nnz <= A.nnz


@settings(max_examples=15, deadline=None)
@given(n=st.integers(10, 80), seed=st.integers(0, 10_000))
def test_pcg_amg_always_converges_on_spd(n, seed):
    from amgx_amd import AMGConfig, create_solver
    from amgx_amd.resources import Resources
    A = CSRMatrix.from_scipy(_rand_csr(n, 0.1, seed, spd=True))
    cfg = {"solver": {"preconditioner": {
        "solver": "AMG", "algorithm": "AGGREGATION",
        "smoother": "MULTICOLOR_GS", "symmetric_GS": 1, "max_iters": 1,
        "min_coarse_rows": 4, "cycle": "V"},
        "solver": "PCG", "max_iters": 300, "monitor_residual": 1,
        "convergence": "RELATIVE_INI", "tolerance": 1e-8}}
    s = create_solver(AMGConfig.from_dict(cfg).root_scope(),
                      resources=Resources("cpu"))
    b = torch.ones(n, dtype=torch.float64)
    x = torch.zeros_like(b)
    s.setup(A)
    st_ = s.solve(b, x, zero_initial_guess=True)
    rel = ops.nrm2(ops.residual(A, x, b)) / ops.nrm2(b)
    assert st_.converged and rel < 1e-6


@settings(max_examples=15, deadline=None)
@given(n=st.integers(4, 80), density=st.floats(0.02, 0.3),
       seed=st.integers(0, 10_000), block=st.sampled_from([1, 2, 4]))
def test_binary_io_roundtrip(n, density, seed, block):
    import os
    import tempfile

    from amgx_amd.io.binary import read_system_binary, write_system_binary
    if block == 1:
        A = CSRMatrix.from_scipy(_rand_csr(n, density, seed))
    else:
        m = _rand_csr(n, density, seed)
        bv = np.random.RandomState(seed).rand(m.nnz, block, block)
        A = CSRMatrix.from_bsr(torch.from_numpy(m.indptr.astype(np.int32)),
                               torch.from_numpy(m.indices.astype(np.int32)),
                               torch.from_numpy(bv))
    b = torch.rand(A.n_rows * block, dtype=torch.float64)
    with tempfile.TemporaryDirectory() as d:
        p = os.path.join(d, "s.bin")
        write_system_binary(p, A, b, None)
        A2, b2, _ = read_system_binary(p)
    assert torch.equal(A2.row_offsets, A.row_offsets)
    assert torch.equal(A2.col_indices, A.col_indices)
    assert torch.allclose(A2.values.reshape(-1), A.values.reshape(-1))
    assert A2.block_dim == block
    assert torch.allclose(b2, b)


@settings(max_examples=15, deadline=None)
@given(n=st.integers(8, 80), density=st.floats(0.03, 0.25),
       seed=st.integers(0, 10_000))
def test_galerkin_matches_scipy_triple_product(n, density, seed):
    m = _rand_csr(n, density, seed, spd=True)
    A = CSRMatrix.from_scipy(m)
    agg, nc = ops.size2_matching(A)
    Ac = ops.galerkin_aggregation(A, agg, nc)
    P = sp.csr_matrix((np.ones(n), (np.arange(n), agg.numpy())),
                      shape=(n, nc))
    ref = (P.T @ m @ P).toarray()
    assert np.allclose(Ac.to_scipy().toarray(), ref, rtol=1e-12, atol=1e-12)


@settings(max_examples=10, deadline=None)
@given(n=st.integers(10, 60), seed=st.integers(0, 10_000),
       scaling=st.sampled_from(["BINORMALIZATION", "DIAGONAL_SYMMETRIC",
                                "NBINORMALIZATION"]))
def test_scaler_lifecycle_restores_matrix(n, seed, scaling):
    from amgx_amd import AMGConfig, create_solver
    from amgx_amd.resources import Resources
    A = CSRMatrix.from_scipy(_rand_csr(n, 0.15, seed, spd=True))
    before = A.values.clone()
    cfg = {"solver": {"solver": "PCG", "preconditioner": "BLOCK_JACOBI",
                      "max_iters": 300, "monitor_residual": 1,
                      "tolerance": 1e-8, "convergence": "RELATIVE_INI",
                      "scaling": scaling}}
    s = create_solver(AMGConfig.from_dict(cfg).root_scope(),
                      resources=Resources("cpu"))
    b = torch.ones(n, dtype=torch.float64)
    x = torch.zeros_like(b)
    s.setup(A)
    st_ = s.solve(b, x, zero_initial_guess=True)
    assert torch.allclose(A.values, before, rtol=1e-12, atol=1e-14)
    assert st_.converged
    rel = ops.nrm2(ops.residual(A, x, b)) / ops.nrm2(b)
    assert rel < 1e-5
