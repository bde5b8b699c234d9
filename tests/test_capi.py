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


# ------------------------------------------------- binary IO + API additions
def test_binary_io_roundtrip(tmp_path):
    """Binary system format round-trips matrix+rhs+sol (reference
    ReadNVAMGBinary, src/readers.cu:1676)."""
    import numpy as np
    import torch

    from amgx_amd.io.binary import (is_binary_file, read_system_binary,
                                    write_system_binary)
    from amgx_amd.problems import poisson_3d
    A = poisson_3d(4, 4, 4)
    b = torch.rand(A.n_rows, dtype=torch.float64)
    x = torch.rand(A.n_rows, dtype=torch.float64)
    p = str(tmp_path / "sys.bin")
    write_system_binary(p, A, b, x)
    assert is_binary_file(p)
    A2, b2, x2 = read_system_binary(p)
    assert torch.equal(A2.row_offsets, A.row_offsets)
    assert torch.equal(A2.col_indices, A.col_indices)
    assert torch.allclose(A2.values, A.values)
    assert torch.allclose(b2, b) and torch.allclose(x2, x)


def test_nvamg_binary_roundtrip(tmp_path):
    """Reference NVAMGBinary format interop (src/readers.cu:1676 header
    '%%NVAMGBinary' + 9 u32): our writer emits the upstream layout and our
    dispatching reader loads it; block + external-diag variant included."""
    import torch

    from amgx_amd.io.binary import (is_binary_file, read_system_any,
                                    read_system_nvamg, write_system_nvamg)
    from amgx_amd.problems import block_laplacian, poisson_3d
    A = poisson_3d(4, 4, 4)
    b = torch.rand(A.n_rows, dtype=torch.float64)
    x = torch.rand(A.n_rows, dtype=torch.float64)
    p = str(tmp_path / "sys.nvamgb")
    write_system_nvamg(p, A, b, x)
    assert is_binary_file(p)
    with open(p, "rb") as f:
        assert f.read(14) == b"%%NVAMGBinary\n"
    A2, b2, x2 = read_system_nvamg(p)
    assert torch.equal(A2.row_offsets, A.row_offsets)
    assert torch.equal(A2.col_indices, A.col_indices)
    assert torch.allclose(A2.values, A.values)
    assert torch.allclose(b2, b) and torch.allclose(x2, x)
    # dispatching reader handles both magics
    A3, _, _ = read_system_any(p)
    assert torch.allclose(A3.values, A.values)
    # block matrix
    Ab = block_laplacian(4, 4, block_dim=3, seed=3)
    pb = str(tmp_path / "blk.nvamgb")
    write_system_nvamg(pb, Ab)
    Ab2, _, _ = read_system_nvamg(pb)
    assert Ab2.block_dim == 3
    assert torch.allclose(Ab2.values, Ab.values)


def test_l1_scaled_blocked_norm():
    """L1_SCALED on a block matrix must not silently fall into the L2
    default (advisor finding): check it equals per-component L1 / length."""
    import torch

    from amgx_amd.problems import block_laplacian
    from amgx_amd.solvers import create_solver
    from amgx_amd.config import AMGConfig
    from amgx_amd.resources import Resources
    A = block_laplacian(5, 5, block_dim=2, seed=1)
    cfg = AMGConfig.from_dict({
        "config_version": 2,
        "solver": {"solver": "BLOCK_JACOBI", "max_iters": 2,
                   "norm": "L1_SCALED", "monitor_residual": 1,
                   "convergence": "RELATIVE_INI", "tolerance": 1e-30}})
    s = create_solver(cfg.root_scope(), resources=Resources("cpu"))
    s.setup(A)
    r = torch.rand(A.n_rows * 2, dtype=torch.float64)
    got = s.compute_norm(r)
    per = r.reshape(-1, 2).abs().sum(0) / A.n_rows
    assert abs(got - float(per.max())) < 1e-12


def test_capi_read_write_binary(tmp_path):
    import torch

    from amgx_amd import capi as C
    C.AMGX_initialize()
    rc, cfg = C.AMGX_config_create("config_version=2, solver=CG, max_iters=40,"
                                   " tolerance=1e-8, convergence=RELATIVE_INI,"
                                   " monitor_residual=1")
    assert rc == C.RC_OK
    rc, res = C.AMGX_resources_create_simple(cfg)
    rc, m = C.AMGX_matrix_create(res, "hDDI")
    rc, rhs = C.AMGX_vector_create(res, "hDDI")
    rc, sol = C.AMGX_vector_create(res, "hDDI")
    rc = C.AMGX_generate_distributed_poisson_7pt(m, rhs, sol, 1, 1, 5, 5, 5)
    assert rc == C.RC_OK
    p = str(tmp_path / "sys.bin")
    rc = C.AMGX_write_system(m, rhs, sol, p)
    assert rc == C.RC_OK
    rc, m2 = C.AMGX_matrix_create(res, "hDDI")
    rc, rhs2 = C.AMGX_vector_create(res, "hDDI")
    rc, sol2 = C.AMGX_vector_create(res, "hDDI")
    rc = C.AMGX_read_system(m2, rhs2, sol2, p)
    assert rc == C.RC_OK
    assert m2.A.n_rows == 125
    rc, nnz = C.AMGX_matrix_get_nnz(m2)
    assert rc == C.RC_OK and nnz == m.A.nnz
    # solve from the reloaded system
    rc, s = C.AMGX_solver_create(res, "hDDI", cfg)
    assert C.AMGX_solver_setup(s, m2) == C.RC_OK
    assert C.AMGX_solver_solve(s, rhs2, sol2) == C.RC_OK
    rc, nrm = C.AMGX_solver_calculate_residual_norm(s, m2, rhs2, sol2)
    assert rc == C.RC_OK and nrm < 1e-6 * 125


def test_capi_matrix_vector_multiply():
    import numpy as np
    import torch

    from amgx_amd import capi as C
    C.AMGX_initialize()
    rc, cfg = C.AMGX_config_create("config_version=2, solver=CG")
    rc, res = C.AMGX_resources_create_simple(cfg)
    rc, m = C.AMGX_matrix_create(res, "hDDI")
    rc, x = C.AMGX_vector_create(res, "hDDI")
    rc, y = C.AMGX_vector_create(res, "hDDI")
    n = 4
    ro = [0, 1, 2, 3, 4]
    ci = [0, 1, 2, 3]
    va = [2.0, 3.0, 4.0, 5.0]
    assert C.AMGX_matrix_upload_all(m, n, 4, 1, 1, ro, ci, va) == C.RC_OK
    C.AMGX_vector_upload(x, n, 1, [1.0, 1.0, 1.0, 1.0])
    assert C.AMGX_matrix_vector_multiply(m, x, y) == C.RC_OK
    assert np.allclose(y.v.numpy(), [2, 3, 4, 5])


def test_capi_attach_geometry_and_coloring():
    import numpy as np

    from amgx_amd import capi as C
    from amgx_amd.problems import poisson_3d
    C.AMGX_initialize()
    rc, cfg = C.AMGX_config_create("config_version=2, solver=CG")
    rc, res = C.AMGX_resources_create_simple(cfg)
    rc, m = C.AMGX_matrix_create(res, "hDDI")
    A = poisson_3d(3, 3, 3)
    m.A = A
    g = np.random.rand(27, 3)
    assert C.AMGX_matrix_attach_geometry(m, g) == C.RC_OK
    assert "geometry" in A._cache
    colors = np.arange(27) % 2
    # invalid colorings are accepted like the reference (user responsibility)
    assert C.AMGX_matrix_attach_coloring(m, colors) == C.RC_OK
    assert A._cache["coloring"].num_colors == 2
    assert C.AMGX_get_error_string(C.RC_OK) == "No error"
    assert "configuration" in C.AMGX_get_error_string(
        C.RC_BAD_CONFIGURATION).lower()


def test_capi_eigensolver():
    """AMGX_eigensolver_* flow (reference include/amgx_eig_c.h:16-26)."""
    import torch

    from amgx_amd import capi as C
    C.AMGX_initialize()
    rc, cfg = C.AMGX_config_create(
        "config_version=2, eig_solver=POWER_ITERATION, eig_max_iters=200,"
        " eig_tolerance=1e-8")
    assert rc == C.RC_OK
    rc, res = C.AMGX_resources_create_simple(cfg)
    rc, m = C.AMGX_matrix_create(res, "hDDI")
    n = 3
    ro = [0, 1, 2, 3]
    ci = [0, 1, 2]
    va = [3.0, 2.0, 1.0]
    assert C.AMGX_matrix_upload_all(m, n, 3, 1, 1, ro, ci, va) == C.RC_OK
    rc, es = C.AMGX_eigensolver_create(res, "hDDI", cfg)
    assert rc == C.RC_OK
    assert C.AMGX_eigensolver_setup(es, m) == C.RC_OK
    rc, v0 = C.AMGX_vector_create(res, "hDDI")
    C.AMGX_vector_upload(v0, n, 1, [1.0, 1.0, 1.0])
    assert C.AMGX_eigensolver_solve(es, v0) == C.RC_OK
    assert abs(es.status.eigenvalues[-1] - 3.0) < 1e-6
    assert C.AMGX_eigensolver_destroy(es) == C.RC_OK


def test_upload_with_external_diag():
    """DIAG-property upload (block-DIA-CSR, reference include/matrix.h:24-26):
    the external diagonal is folded into the CSR so every kernel sees one
    canonical matrix; solve equals the plain upload."""
    import numpy as np
    import torch

    from amgx_amd import capi as C
    from amgx_amd.problems import poisson_2d
    C.AMGX_initialize()
    rc, cfg = C.AMGX_config_create(
        "config_version=2, solver=CG, max_iters=200, tolerance=1e-10,"
        " convergence=RELATIVE_INI, monitor_residual=1")
    rc, res = C.AMGX_resources_create_simple(cfg)
    A_full = poisson_2d(7, 6).to_scipy().tocsr()
    n = A_full.shape[0]
    # split: off-diagonal CSR + external diagonal
    offd = A_full.copy().tolil()
    diag = A_full.diagonal().copy()
    offd.setdiag(0.0)
    offd = offd.tocsr()
    offd.eliminate_zeros()
    rc, m = C.AMGX_matrix_create(res, "hDDI")
    assert C.AMGX_matrix_upload_all(
        m, n, offd.nnz, 1, 1, offd.indptr, offd.indices, offd.data,
        diag_data=diag) == C.RC_OK
    # folded matrix equals the original
    got = m.A.to_scipy().tocsr()
    assert (abs(got - A_full)).nnz == 0
    rc, bh = C.AMGX_vector_create(res, "hDDI")
    rc, xh = C.AMGX_vector_create(res, "hDDI")
    C.AMGX_vector_upload(bh, n, 1, np.ones(n))
    C.AMGX_vector_set_zero(xh, n, 1)
    rc, s = C.AMGX_solver_create(res, "hDDI", cfg)
    assert C.AMGX_solver_setup(s, m) == C.RC_OK
    assert C.AMGX_solver_solve(s, bh, xh) == C.RC_OK
    rc, nrm = C.AMGX_solver_calculate_residual_norm(s, m, bh, xh)
    assert nrm < 1e-8
    # replace coefficients with the same split structure, scaled by 2
    assert C.AMGX_matrix_replace_coefficients(
        m, n, offd.nnz, offd.data * 2.0, diag_data=diag * 2.0) == C.RC_OK
    got2 = m.A.to_scipy().tocsr()
    assert np.allclose(got2.toarray(), 2.0 * A_full.toarray())


def test_upload_all_global_single_process():
    """upload_all_global without torch.distributed = the 1-rank communicator
    case (reference runs on MPI_COMM_SELF): serial matrix, full solve."""
    import numpy as np
    from amgx_amd import capi as C
    C.AMGX_initialize()
    rc, cfg = C.AMGX_config_create(
        "config_version=2, solver=PCG, preconditioner=BLOCK_JACOBI,"
        " max_iters=500, tolerance=1e-10, convergence=RELATIVE_INI,"
        " monitor_residual=1")
    rc, res = C.AMGX_resources_create(cfg, comm=None, device_num=0)
    rc, A = C.AMGX_matrix_create(res, "hDDI")
    n = 50
    ro, cols, vals = [0], [], []
    for i in range(n):
        for j, v in ((i - 1, -1.0), (i, 2.0), (i + 1, -1.0)):
            if 0 <= j < n:
                cols.append(j)
                vals.append(v)
        ro.append(len(cols))
    rc = C.AMGX_matrix_upload_all_global(
        A, n, n, len(cols), 1, 1, np.asarray(ro),
        np.asarray(cols, dtype=np.int64), np.asarray(vals))
    assert rc == C.RC_OK
    rc, b = C.AMGX_vector_create(res, "hDDI")
    rc, x = C.AMGX_vector_create(res, "hDDI")
    C.AMGX_vector_upload(b, n, 1, np.ones(n))
    C.AMGX_vector_set_zero(x, n, 1)
    rc, s = C.AMGX_solver_create(res, "hDDI", cfg)
    assert C.AMGX_solver_setup(s, A) == C.RC_OK
    assert C.AMGX_solver_solve(s, b, x) == C.RC_OK
    rc, st = C.AMGX_solver_get_status(s)
    assert st == C.AMGX_SOLVE_SUCCESS
    C.AMGX_finalize()


def test_read_system_block_convert_and_rhs_from_a():
    """Resources-config IO knobs (reference src/amgx_c.cu:5008-5019):
    block_convert=4 regroups a scalar file system into 4x4 block-CSR and
    rhs_from_a=1 generates b = A*[1..1]^T when the file has no RHS."""
    from amgx_amd import capi as C
    from amgx_amd.io.matrix_market import write_matrix_market
    A = poisson_2d(8, 8)   # 64 rows -> 16 block-4 rows
    tmp = tempfile.mkdtemp()
    path = os.path.join(tmp, "m.mtx")
    write_matrix_market(path, A)
    C.AMGX_initialize()
    rc, cfg = C.AMGX_config_create(
        "config_version=2, solver=PCG, max_iters=200, tolerance=1e-8,"
        " convergence=RELATIVE_INI, monitor_residual=1, block_convert=4,"
        " rhs_from_a=1")
    assert rc == C.RC_OK
    rc, res = C.AMGX_resources_create_simple(cfg)
    rc, m = C.AMGX_matrix_create(res, "hDDI")
    rc, b = C.AMGX_vector_create(res, "hDDI")
    rc, x = C.AMGX_vector_create(res, "hDDI")
    assert C.AMGX_read_system(m, b, x, path) == C.RC_OK
    assert m.A.block_dim == 4 and m.A.n_rows == 16
    ref = torch.from_numpy(A.to_scipy() @ np.ones(64))
    assert torch.allclose(b.v, ref)
    rc, s = C.AMGX_solver_create(res, "hDDI", cfg)
    assert C.AMGX_solver_setup(s, m) == C.RC_OK
    assert C.AMGX_solver_solve(s, b, x) == C.RC_OK
    resid = ref - torch.from_numpy(A.to_scipy() @ x.v.numpy())
    assert float(resid.norm() / ref.norm()) < 1e-7


def test_print_callback_captures_all_output():
    """AMGX_register_print_callback receives EVERY library print —
    residual table, grid stats, timings (reference src/misc.cu
    amgx_output indirection)."""
    from amgx_amd import capi as C
    captured = []
    C.AMGX_initialize()
    C.AMGX_register_print_callback(lambda m: captured.append(m))
    try:
        rc, cfg = C.AMGX_config_create(
            "config_version=2, solver=PCG, max_iters=50, tolerance=1e-6,"
            " convergence=RELATIVE_INI, monitor_residual=1,"
            " print_solve_stats=1, obtain_timings=1")
        rc, res = C.AMGX_resources_create_simple(cfg)
        rc, m = C.AMGX_matrix_create(res, "hDDI")
        rc, b = C.AMGX_vector_create(res, "hDDI")
        rc, x = C.AMGX_vector_create(res, "hDDI")
        C.AMGX_generate_distributed_poisson_7pt(m, b, x, 1, 1, 6, 6, 6)
        rc, s = C.AMGX_solver_create(res, "hDDI", cfg)
        C.AMGX_solver_setup(s, m)
        C.AMGX_solver_solve(s, b, x)
    finally:
        C.AMGX_register_print_callback(None)
    joined = "".join(captured)
    assert "Total Iterations" in joined
    assert "Total Time" in joined


def test_solver_level_print_callback():
    """AMGX_solver_register_print_callback redirects THAT solver's output
    (registration before setup included)."""
    from amgx_amd import capi as C
    C.AMGX_initialize()
    rc, cfg = C.AMGX_config_create(
        "config_version=2, solver=PCG, max_iters=30, tolerance=1e-6,"
        " convergence=RELATIVE_INI, monitor_residual=1, print_solve_stats=1")
    rc, res = C.AMGX_resources_create_simple(cfg)
    rc, m = C.AMGX_matrix_create(res, "hDDI")
    rc, b = C.AMGX_vector_create(res, "hDDI")
    rc, x = C.AMGX_vector_create(res, "hDDI")
    C.AMGX_generate_distributed_poisson_7pt(m, b, x, 1, 1, 6, 6, 6)
    rc, s = C.AMGX_solver_create(res, "hDDI", cfg)
    got = []
    C.AMGX_solver_register_print_callback(s, lambda msg: got.append(msg))
    C.AMGX_solver_setup(s, m)
    C.AMGX_solver_solve(s, b, x)
    assert any("Total Iterations" in g for g in got)
