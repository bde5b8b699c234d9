"""AMGX_* API surface tests (reference examples/amgx_capi.c flow)."""

import os
import tempfile

import numpy as np
import pytest
import torch

from amgx_amd import capi
from amgx_amd.problems import poisson_2d

CFG = """{
  "config_version": 2,
  "solver": {
    "solver": "FGMRES",
    "gmres_n_restart": 10,
    "max_iters": 100,
    "monitor_residual": 1,
    "convergence": "RELATIVE_INI",
    "tolerance": 1e-06,
    "preconditioner": {
      "solver": "AMG", "algorithm": "AGGREGATION", "selector": "SIZE_2",
      "smoother": "MULTICOLOR_DILU", "presweeps": 0, "postsweeps": 3,
      "coarse_solver": "DENSE_LU_SOLVER", "max_iters": 1,
      "min_coarse_rows": 32, "relaxation_factor": 0.75, "cycle": "V"
    }
  }
}"""


def test_capi_full_flow():
    assert capi.AMGX_initialize() == capi.RC_OK
    rc, (major, minor) = capi.AMGX_get_api_version()[0], capi.AMGX_get_api_version()[1:]
    rc, cfg = capi.AMGX_config_create(CFG)
    assert rc == capi.RC_OK
    rc, res = capi.AMGX_resources_create_simple(cfg)
    assert rc == capi.RC_OK
    mode = "hDDI"
    rc, A = capi.AMGX_matrix_create(res, mode)
    assert rc == capi.RC_OK
    rc, b = capi.AMGX_vector_create(res, mode)
    rc, x = capi.AMGX_vector_create(res, mode)
    # upload a Poisson matrix
    P = poisson_2d(16, 16)
    n, nnz = P.n_rows, P.nnz
    rc = capi.AMGX_matrix_upload_all(A, n, nnz, 1, 1,
                                     P.row_offsets.numpy(),
                                     P.col_indices.numpy(),
                                     P.values.numpy())
    assert rc == capi.RC_OK
    rc, rows, bx, by = capi.AMGX_matrix_get_size(A)
    assert rows == n and bx == 1
    capi.AMGX_vector_upload(b, n, 1, np.ones(n))
    capi.AMGX_vector_set_zero(x, n, 1)
    rc, s = capi.AMGX_solver_create(res, mode, cfg)
    assert rc == capi.RC_OK
    assert capi.AMGX_solver_setup(s, A) == capi.RC_OK
    assert capi.AMGX_solver_solve(s, b, x, True) == capi.RC_OK
    rc, status = capi.AMGX_solver_get_status(s)
    assert status == 0
    rc, iters = capi.AMGX_solver_get_iterations_number(s)
    assert 0 < iters <= 40
    rc, resid = capi.AMGX_solver_get_iteration_residual(s)
    rc, xs = capi.AMGX_vector_download(x)
    r = np.ones(n) - P.to_scipy() @ xs
    assert np.linalg.norm(r) < 1e-4
    for h, d in ((s, capi.AMGX_solver_destroy), (x, capi.AMGX_vector_destroy),
                 (b, capi.AMGX_vector_destroy), (A, capi.AMGX_matrix_destroy),
                 (res, capi.AMGX_resources_destroy),
                 (cfg, capi.AMGX_config_destroy)):
        assert d(h) == capi.RC_OK
    assert capi.AMGX_finalize() == capi.RC_OK


def test_capi_read_write_system(tmp_path):
    capi.AMGX_initialize()
    rc, cfg = capi.AMGX_config_create('{"config_version": 2, "solver": {"solver": "CG"}}')
    rc, res = capi.AMGX_resources_create_simple(cfg)
    rc, A = capi.AMGX_matrix_create(res, "hDDI")
    rc, b = capi.AMGX_vector_create(res, "hDDI")
    rc, x = capi.AMGX_vector_create(res, "hDDI")
    P = poisson_2d(6, 6)
    capi.AMGX_matrix_upload_all(A, P.n_rows, P.nnz, 1, 1,
                                P.row_offsets.numpy(), P.col_indices.numpy(),
                                P.values.numpy())
    capi.AMGX_vector_upload(b, P.n_rows, 1, np.arange(P.n_rows, dtype=float))
    path = str(tmp_path / "sys.mtx")
    assert capi.AMGX_write_system(A, b, None, path) == capi.RC_OK
    rc, A2 = capi.AMGX_matrix_create(res, "hDDI")
    rc, b2 = capi.AMGX_vector_create(res, "hDDI")
    rc, x2 = capi.AMGX_vector_create(res, "hDDI")
    assert capi.AMGX_read_system(A2, b2, x2, path) == capi.RC_OK
    assert np.allclose(A2.A.to_scipy().toarray(), P.to_scipy().toarray())
    rc, b2v = capi.AMGX_vector_download(b2)
    assert np.allclose(b2v, np.arange(P.n_rows, dtype=float))


def test_capi_graceful_failure():
    """reference src/tests/capi_graceful_failure.cu: errors come back as RC
    codes, never exceptions."""
    rc = capi.AMGX_matrix_upload_all(None, 1, 1, 1, 1, [0, 1], [0], [1.0])
    assert rc != capi.RC_OK
    rc_and = capi.AMGX_config_create("this is { not json =")
    assert rc_and != capi.RC_OK or rc_and[0] != capi.RC_OK
    rc, cfg = capi.AMGX_config_create('{"config_version": 2, "solver": {"solver": "NOPE"}}')
    rc, res = capi.AMGX_resources_create_simple(cfg)
    rc, m = capi.AMGX_matrix_create(res, "hDDI")
    P = poisson_2d(4, 4)
    capi.AMGX_matrix_upload_all(m, P.n_rows, P.nnz, 1, 1,
                                P.row_offsets.numpy(), P.col_indices.numpy(),
                                P.values.numpy())
    rc, s = capi.AMGX_solver_create(res, "hDDI", cfg)
    assert capi.AMGX_solver_setup(s, m) != capi.RC_OK   # unknown solver name
    assert capi.AMGX_matrix_create(res, "zZZZ") != capi.RC_OK


def test_capi_symmetry_checks():
    capi.AMGX_initialize()
    rc, cfg = capi.AMGX_config_create('{"solver": {"solver": "CG"}}')
    rc, res = capi.AMGX_resources_create_simple(cfg)
    rc, A = capi.AMGX_matrix_create(res, "hDDI")
    P = poisson_2d(5, 5)
    capi.AMGX_matrix_upload_all(A, P.n_rows, P.nnz, 1, 1,
                                P.row_offsets.numpy(), P.col_indices.numpy(),
                                P.values.numpy())
    rc, struct_sym, sym = capi.AMGX_matrix_check_symmetry(A)
    assert struct_sym and sym
    rc, dd = capi.AMGX_matrix_check_diag_dominant(A)
    assert dd
