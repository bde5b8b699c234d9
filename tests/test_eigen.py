import os
import numpy as np
import pytest
import torch

from amgx_amd.config import ConfigScope
from amgx_amd.eigensolvers import create_eigensolver, EIGEN_REGISTRY
from amgx_amd.problems import poisson_2d
from amgx_amd.resources import Resources


def scope(d):
    return ConfigScope(None, d)


def ref_extreme_eigs(A):
    dense = A.to_scipy().toarray()
    evals = np.linalg.eigvalsh(dense)
    return evals[0], evals[-1]


def test_power_iteration_largest():
    A = poisson_2d(12, 12)
    lo, hi = ref_extreme_eigs(A)
    s = create_eigensolver(scope({"eig_solver": "POWER_ITERATION",
                                  "eig_max_iters": 2000,
                                  "eig_tolerance": 1e-8}), Resources("cpu"))
    s.setup(A)
    st = s.solve()
    assert st.converged
    assert abs(st.eigenvalues[0] - hi) < 1e-5 * hi


def test_lanczos_largest():
    A = poisson_2d(12, 12)
    lo, hi = ref_extreme_eigs(A)
    s = create_eigensolver(scope({"eig_solver": "LANCZOS",
                                  "eig_max_iters": 80,
                                  "eig_tolerance": 1e-9}), Resources("cpu"))
    s.setup(A)
    st = s.solve()
    assert st.converged
    assert abs(st.eigenvalues[0] - hi) < 1e-6 * hi
    # eigenvector residual
    v = st.eigenvector.numpy()
    r = A.to_scipy() @ v - st.eigenvalues[0] * v
    assert np.linalg.norm(r) < 1e-5 * abs(st.eigenvalues[0])


def test_arnoldi_largest():
    A = poisson_2d(10, 10)
    lo, hi = ref_extreme_eigs(A)
    s = create_eigensolver(scope({"eig_solver": "ARNOLDI",
                                  "eig_max_iters": 80,
                                  "eig_tolerance": 1e-9}), Resources("cpu"))
    s.setup(A)
    st = s.solve()
    assert st.converged and abs(st.eigenvalues[0] - hi) < 1e-6 * hi


def test_lanczos_smallest_via_inverse():
    A = poisson_2d(8, 8)
    lo, hi = ref_extreme_eigs(A)
    s = create_eigensolver(scope({"eig_solver": "LANCZOS",
                                  "eig_max_iters": 60,
                                  "eig_tolerance": 1e-8,
                                  "eig_which": "smallest"}), Resources("cpu"))
    s.setup(A)
    st = s.solve()
    assert abs(st.eigenvalues[0] - lo) < 1e-5 * hi


def test_lobpcg_smallest():
    A = poisson_2d(10, 10)
    lo, hi = ref_extreme_eigs(A)
    s = create_eigensolver(scope({"eig_solver": "LOBPCG",
                                  "eig_max_iters": 300,
                                  "eig_tolerance": 1e-7}), Resources("cpu"))
    s.setup(A)
    st = s.solve()
    assert st.converged
    assert abs(st.eigenvalues[0] - lo) < 1e-4 * hi


def test_subspace_iteration():
    A = poisson_2d(10, 10)
    lo, hi = ref_extreme_eigs(A)
    s = create_eigensolver(scope({"eig_solver": "SUBSPACE_ITERATION",
                                  "eig_max_iters": 500,
                                  "eig_tolerance": 1e-9,
                                  "eig_wanted_count": 2}), Resources("cpu"))
    s.setup(A)
    st = s.solve()
    assert st.converged
    assert abs(st.eigenvalues[0] - hi) < 1e-4 * hi


def test_pagerank():
    """PageRank on a tiny digraph: stationary vector of the Google matrix."""
    import scipy.sparse as sp
    from amgx_amd.matrix import CSRMatrix
    # 4-node graph
    edges = [(0, 1), (0, 2), (1, 2), (2, 0), (3, 2)]
    m = sp.csr_matrix((np.ones(len(edges)),
                       ([e[0] for e in edges], [e[1] for e in edges])),
                      shape=(4, 4))
    A = CSRMatrix.from_scipy(m)
    s = create_eigensolver(scope({"eig_solver": "POWER_ITERATION",
                                  "eig_max_iters": 500,
                                  "eig_tolerance": 1e-10}), Resources("cpu"))
    s.pagerank_setup(A, damping=0.85)
    st = s.solve()
    pr = st.eigenvector / st.eigenvector.sum()
    # dense reference
    d = 0.85
    P = np.zeros((4, 4))
    out = np.asarray(m.sum(axis=1)).ravel()
    for i, j in edges:
        P[j, i] = 1.0 / out[i]
    G = d * P + np.where(out == 0, d / 4.0, 0.0)[None, :] + (1 - d) / 4.0
    evals, evecs = np.linalg.eig(G)
    k = np.argmax(np.real(evals))
    ref = np.real(evecs[:, k])
    ref = ref / ref.sum()
    assert np.allclose(pr.numpy(), ref, atol=1e-6)


def test_shipped_eigen_configs():
    """Every flat config in configs/eigen/ builds a working eigensolver on
    a small SPD system (reference src/configs/eigen_configs)."""
    import glob
    import os

    from amgx_amd.config import AMGConfig
    from amgx_amd.eigensolvers import create_eigensolver
    from amgx_amd.problems import poisson_2d
    here = os.path.join(os.path.dirname(__file__), "..", "configs", "eigen")
    files = sorted(f for f in glob.glob(os.path.join(here, "*"))
                   if os.path.isfile(f))
    assert len(files) >= 8
    A = poisson_2d(10, 10)
    dense = A.to_scipy().toarray()
    evs = np.linalg.eigvalsh(dense)
    for f in files:
        cfg = AMGConfig.from_file(f)
        es = create_eigensolver(cfg.root_scope(), Resources("cpu"))
        if os.path.basename(f) == "PAGERANK":
            continue   # needs a digraph; covered by test_pagerank
        es.setup(A)
        st = es.solve()
        assert st.converged, os.path.basename(f)
        ref = evs[0] if es.which == "smallest" else evs[-1]
        assert abs(st.eigenvalues[-1 if es.which != "smallest" else 0]
                   - ref) < 1e-3 * abs(ref), (os.path.basename(f),
                                              st.eigenvalues, ref)


def test_eig_param_parity_and_knobs():
    """All reference eigensolver parameters (eigensolvers.cu:20-44) are
    registered; eig_subspace_size sets the block width of subspace
    iteration and eig_convergence_check_freq batches the residual check."""
    from amgx_amd.config import PARAM_REGISTRY
    for p in ("eig_solver", "eig_max_iters", "eig_tolerance", "eig_which",
              "eig_shift", "eig_damping_factor", "eig_eigenvector",
              "eig_wanted_count", "eig_subspace_size",
              "eig_convergence_check_freq", "eig_eigenvector_solver"):
        assert p in PARAM_REGISTRY, p
    import torch

    from amgx_amd.config import AMGConfig
    from amgx_amd.eigensolvers import create_eigensolver
    from amgx_amd.problems import poisson_2d
    from amgx_amd.resources import Resources
    A = poisson_2d(12, 12)
    cfg = AMGConfig.from_dict({"eig_solver": "SUBSPACE_ITERATION",
                               "eig_max_iters": 300, "eig_tolerance": 1e-6,
                               "eig_wanted_count": 2,
                               "eig_subspace_size": 6})
    s = create_eigensolver(cfg.root_scope(), Resources("cpu"))
    s.setup(A)
    st = s.solve()
    assert st.converged and len(st.eigenvalues) >= 2
    import scipy.sparse as sp

    from amgx_amd.matrix import CSRMatrix
    D = CSRMatrix.from_scipy(sp.diags([20.0] + [1.0 + 0.1 * i for i in range(49)],
                                      format="csr").tocsr())
    cfg2 = AMGConfig.from_dict({"eig_solver": "POWER_ITERATION",
                                "eig_max_iters": 500, "eig_tolerance": 1e-8,
                                "eig_convergence_check_freq": 5})
    s2 = create_eigensolver(cfg2.root_scope(), Resources("cpu"))
    s2.setup(D)
    st2 = s2.solve()
    assert st2.converged
    assert st2.iterations % 5 == 0 or st2.iterations == 500
    assert abs(st2.eigenvalues[0] - 20.0) < 1e-4


def test_inverse_iteration_and_pagerank_aliases():
    """Reference registers PAGERANK and INVERSE_ITERATION into the
    single-iteration family (eigensolvers.cu:38-43); INVERSE_ITERATION
    defaults to the smallest eigenpair."""
    import scipy.sparse.linalg as spla

    from amgx_amd.config import AMGConfig
    from amgx_amd.eigensolvers import EIGEN_REGISTRY, create_eigensolver
    from amgx_amd.problems import poisson_2d
    from amgx_amd.resources import Resources
    assert "PAGERANK" in EIGEN_REGISTRY and "INVERSE_ITERATION" in EIGEN_REGISTRY
    A = poisson_2d(12, 12)
    cfg = AMGConfig.from_dict({"eig_solver": "INVERSE_ITERATION",
                               "eig_max_iters": 200, "eig_tolerance": 1e-6})
    s = create_eigensolver(cfg.root_scope(), Resources("cpu"))
    s.setup(A)
    st = s.solve()
    truth = spla.eigsh(A.to_scipy(), k=1, which="SM",
                       return_eigenvectors=False)[0]
    assert st.converged and abs(st.eigenvalues[0] - truth) < 1e-4


def test_eigensolver_zero_initial_vector():
    """A zero x0 (the C API's freshly-created solution vector) must not
    annihilate the iteration: solvers fall back to the seeded random
    start. Regression: INVERSE_FGMRES through the C API returned
    lam=inf."""
    from amgx_amd import capi as C
    C.AMGX_initialize()
    rc, cfg = C.AMGX_config_create_from_file(
        os.path.join(os.path.dirname(__file__), os.pardir, "configs",
                     "eigen", "INVERSE_FGMRES"))
    assert rc == C.RC_OK
    rc, res = C.AMGX_resources_create_simple(cfg)
    rc, m = C.AMGX_matrix_create(res, "hDDI")
    rc, b = C.AMGX_vector_create(res, "hDDI")
    rc, x = C.AMGX_vector_create(res, "hDDI")
    C.AMGX_generate_distributed_poisson_7pt(m, b, x, 1, 1, 8, 8, 8)
    rc, es = C.AMGX_eigensolver_create(res, "hDDI", cfg)
    assert C.AMGX_eigensolver_setup(es, m) == C.RC_OK
    assert C.AMGX_eigensolver_solve(es, x) == C.RC_OK
    st = es.status
    assert st.converged and abs(st.eigenvalues[0] - 0.3618) < 1e-3
