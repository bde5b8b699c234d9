import numpy as np
import pytest
import torch

from amgx_amd import AMGConfig, create_solver
from amgx_amd.config import ConfigScope
from amgx_amd.resources import Resources
from amgx_amd import ops
from amgx_amd.problems import poisson_2d, poisson_3d, block_laplacian


def make(cfg_dict):
    cfg = AMGConfig.from_dict(cfg_dict)
    return create_solver(cfg.root_scope(), resources=Resources("cpu"))


def solve_poisson(solver, n=None, A=None, tol_check=1e-6):
    if A is None:
        A = poisson_2d(16, 16)
    b = torch.ones(A.n_rows * A.block_dim, dtype=torch.float64)
    x = torch.zeros_like(b)
    solver.setup(A)
    st = solver.solve(b, x, zero_initial_guess=True)
    r = ops.residual(A, x, b)
    rel = ops.nrm2(r) / ops.nrm2(b)
    return st, rel


def test_cg_poisson():
    s = make({"solver": "CG", "max_iters": 400, "monitor_residual": 1,
              "tolerance": 1e-8, "convergence": "RELATIVE_INI"})
    st, rel = solve_poisson(s)
    assert st.converged and rel < 1e-7


def test_pcg_jacobi_poisson():
    s = make({"solver": "PCG", "preconditioner": "BLOCK_JACOBI",
              "max_iters": 400, "monitor_residual": 1, "tolerance": 1e-8,
              "convergence": "RELATIVE_INI"})
    st, rel = solve_poisson(s)
    assert st.converged and rel < 1e-7


def test_bicgstab_poisson():
    s = make({"solver": "PBICGSTAB", "preconditioner": "BLOCK_JACOBI",
              "max_iters": 400, "monitor_residual": 1, "tolerance": 1e-8,
              "convergence": "RELATIVE_INI"})
    st, rel = solve_poisson(s)
    assert st.converged and rel < 1e-6


def test_fgmres_poisson():
    s = make({"solver": "FGMRES", "preconditioner": "BLOCK_JACOBI",
              "gmres_n_restart": 20, "max_iters": 300, "monitor_residual": 1,
              "tolerance": 1e-8, "convergence": "RELATIVE_INI"})
    st, rel = solve_poisson(s)
    assert st.converged and rel < 1e-6


def test_idr_poisson():
    s = make({"solver": "IDR", "subspace_dim_s": 4, "max_iters": 300,
              "monitor_residual": 1, "tolerance": 1e-8,
              "convergence": "RELATIVE_INI"})
    st, rel = solve_poisson(s)
    assert st.converged and rel < 1e-6


def test_jacobi_smoother_reduces_residual():
    A = poisson_2d(12, 12)
    s = make({"solver": "BLOCK_JACOBI", "max_iters": 20})
    b = torch.ones(A.n_rows, dtype=torch.float64)
    x = torch.zeros_like(b)
    s.setup(A)
    r0 = ops.nrm2(ops.residual(A, x, b))
    s.solve(b, x, zero_initial_guess=False)
    r1 = ops.nrm2(ops.residual(A, x, b))
    assert r1 < 0.7 * r0


def test_multicolor_gs_smoother():
    A = poisson_2d(12, 12)
    s = make({"solver": "MULTICOLOR_GS", "max_iters": 10})
    b = torch.ones(A.n_rows, dtype=torch.float64)
    x = torch.zeros_like(b)
    s.setup(A)
    r0 = ops.nrm2(ops.residual(A, x, b))
    s.solve(b, x)
    r1 = ops.nrm2(ops.residual(A, x, b))
    assert r1 < 0.75 * r0


def test_chebyshev_smoother():
    A = poisson_2d(12, 12)
    s = make({"solver": "CHEBYSHEV", "max_iters": 10})
    b = torch.ones(A.n_rows, dtype=torch.float64)
    x = torch.zeros_like(b)
    s.setup(A)
    r0 = ops.nrm2(ops.residual(A, x, b))
    s.solve(b, x)
    r1 = ops.nrm2(ops.residual(A, x, b))
    assert r1 < 0.75 * r0


def test_dilu_smoother_scalar():
    A = poisson_2d(12, 12)
    s = make({"solver": "MULTICOLOR_DILU", "max_iters": 10,
              "relaxation_factor": 1.0})
    b = torch.ones(A.n_rows, dtype=torch.float64)
    x = torch.zeros_like(b)
    s.setup(A)
    r0 = ops.nrm2(ops.residual(A, x, b))
    s.solve(b, x)
    r1 = ops.nrm2(ops.residual(A, x, b))
    assert r1 < 0.6 * r0


def test_ilu0_smoother():
    A = poisson_2d(10, 10)
    s = make({"solver": "MULTICOLOR_ILU", "max_iters": 6})
    b = torch.ones(A.n_rows, dtype=torch.float64)
    x = torch.zeros_like(b)
    s.setup(A)
    r0 = ops.nrm2(ops.residual(A, x, b))
    s.solve(b, x)
    r1 = ops.nrm2(ops.residual(A, x, b))
    assert r1 < 0.7 * r0


def test_block_jacobi_on_block_matrix():
    A = block_laplacian(8, 8, block_dim=4)
    s = make({"solver": "PCG", "preconditioner": "BLOCK_JACOBI",
              "max_iters": 500, "monitor_residual": 1, "tolerance": 1e-8,
              "convergence": "RELATIVE_INI"})
    st, rel = solve_poisson(s, A=A)
    assert st.converged and rel < 1e-6


def test_dense_lu_direct():
    A = poisson_2d(6, 6)
    s = make({"solver": "DENSE_LU_SOLVER"})
    b = torch.rand(A.n_rows, dtype=torch.float64)
    x = torch.zeros_like(b)
    s.setup(A)
    s.solve(b, x)
    r = ops.residual(A, x, b)
    assert ops.nrm2(r) < 1e-10


def test_block4_dilu_smoother_and_fgmres():
    """BASELINE config #4 shape: block-4 coupled system, multicolor DILU
    (reference multicolor_dilu_solver.cu NxN block path)."""
    A = block_laplacian(8, 8, block_dim=4)
    s = make({"solver": "MULTICOLOR_DILU", "max_iters": 8,
              "relaxation_factor": 1.0})
    n = A.n_rows * A.block_dim
    b = torch.ones(n, dtype=torch.float64)
    x = torch.zeros_like(b)
    s.setup(A)
    r0 = ops.nrm2(ops.residual(A, x, b))
    s.solve(b, x)
    r1 = ops.nrm2(ops.residual(A, x, b))
    assert r1 < 0.5 * r0, f"{r1} !< 0.5*{r0}"
    # outer FGMRES + DILU preconditioner on the block system
    s2 = make({"solver": "FGMRES", "max_iters": 200, "gmres_n_restart": 30,
               "monitor_residual": 1, "tolerance": 1e-8,
               "convergence": "RELATIVE_INI",
               "preconditioner": {"solver": "MULTICOLOR_DILU",
                                  "max_iters": 2,
                                  "relaxation_factor": 1.0}})
    x2 = torch.zeros_like(b)
    s2.setup(A)
    st = s2.solve(b, x2, zero_initial_guess=True)
    rel = float(ops.nrm2(ops.residual(A, x2, b)) / ops.nrm2(b))
    assert st.converged and rel < 1e-7


def test_block_sizes_2_to_5():
    """Block sizes 2..5 through block-Jacobi+PCG (reference
    smoother_blocksizes.cu covers 2-10)."""
    for bd in (2, 3, 4, 5):
        A = block_laplacian(6, 6, block_dim=bd)
        s = make({"solver": "PCG", "preconditioner": "BLOCK_JACOBI",
                  "max_iters": 600, "monitor_residual": 1,
                  "tolerance": 1e-8, "convergence": "RELATIVE_INI"})
        n = A.n_rows * bd
        b = torch.ones(n, dtype=torch.float64)
        x = torch.zeros_like(b)
        s.setup(A)
        st = s.solve(b, x, zero_initial_guess=True)
        rel = float(ops.nrm2(ops.residual(A, x, b)) / ops.nrm2(b))
        assert st.converged and rel < 1e-6, f"bd={bd}: {st} rel={rel}"


def test_mixed_precision_mode():
    """dDFI-analog: float32 matrix x float64 vectors (reference mixed mode
    dDFI, include/amgx_config.h:79-120)."""
    from amgx_amd import capi as C
    C.AMGX_initialize()
    rc, cfg = C.AMGX_config_create(
        "config_version=2, solver=PCG, preconditioner=BLOCK_JACOBI,"
        " max_iters=400, tolerance=1e-6, convergence=RELATIVE_INI,"
        " monitor_residual=1")
    assert rc == C.RC_OK
    rc, res = C.AMGX_resources_create_simple(cfg)
    rc, m = C.AMGX_matrix_create(res, "hDFI")
    rc, b = C.AMGX_vector_create(res, "hDFI")
    rc, x = C.AMGX_vector_create(res, "hDFI")
    assert C.AMGX_generate_distributed_poisson_7pt(
        m, b, x, 1, 1, 8, 8, 8) == C.RC_OK
    assert m.A.values.dtype == torch.float32      # matrix precision F
    assert b.v.dtype == torch.float64             # vector precision D
    rc, s = C.AMGX_solver_create(res, "hDFI", cfg)
    assert C.AMGX_solver_setup(s, m) == C.RC_OK
    assert C.AMGX_solver_solve(s, b, x) == C.RC_OK
    assert s.status.converged
    rc, nrm = C.AMGX_solver_calculate_residual_norm(s, m, b, x)
    assert nrm < 1e-3


def test_complex_mode_cg_and_gmres():
    """hZZI complex-double mode (reference complex modes,
    include/amgx_config.h:79-120): Hermitian positive-definite system solved
    by CG (conjugated dots) and GMRES on the host backend."""
    import numpy as np
    import scipy.sparse as sp

    from amgx_amd import AMGConfig, create_solver
    from amgx_amd.matrix import CSRMatrix
    from amgx_amd.resources import Resources
    rng = np.random.RandomState(11)
    n = 60
    B = sp.random(n, n, density=0.06, random_state=rng, format="csr")
    C = sp.random(n, n, density=0.06, random_state=rng, format="csr")
    M = (B + 1j * C)
    H = (M + M.conj().T) * 0.5 + sp.identity(n) * 8.0   # Hermitian PD
    A = CSRMatrix.from_scipy(H.tocsr(), dtype=torch.complex128)
    xref = torch.from_numpy(rng.randn(n) + 1j * rng.randn(n))
    b = torch.from_numpy(H @ xref.numpy())
    for solver_name in ("CG", "GMRES"):
        cfg = AMGConfig.from_dict({"solver": {
            "solver": solver_name, "max_iters": 300,
            "gmres_n_restart": 40, "monitor_residual": 1,
            "convergence": "RELATIVE_INI", "tolerance": 1e-10,
        }})
        s = create_solver(cfg.root_scope(), resources=Resources("cpu"))
        x = torch.zeros(n, dtype=torch.complex128)
        s.setup(A)
        st = s.solve(b, x, zero_initial_guess=True)
        err = float(torch.linalg.vector_norm(x - xref)
                    / torch.linalg.vector_norm(xref))
        assert st.converged, f"{solver_name}: {st}"
        assert err < 1e-7, f"{solver_name}: err={err}"


def test_complex_mode_capi():
    """hZZI through the AMGX_* API with block-Jacobi preconditioning."""
    import numpy as np

    from amgx_amd import capi as C
    C.AMGX_initialize()
    rc, cfg = C.AMGX_config_create(
        "config_version=2, solver=PCG, preconditioner=BLOCK_JACOBI,"
        " max_iters=300, tolerance=1e-10, convergence=RELATIVE_INI,"
        " monitor_residual=1")
    rc, res = C.AMGX_resources_create_simple(cfg)
    rc, m = C.AMGX_matrix_create(res, "hZZI")
    n = 4
    ro = [0, 2, 4, 6, 8]
    ci = [0, 1, 0, 1, 2, 3, 2, 3]
    va = np.asarray([4.0, 1 - 1j, 1 + 1j, 4.0, 5.0, 2j, -2j, 5.0],
                    dtype=np.complex128)
    assert C.AMGX_matrix_upload_all(m, n, 8, 1, 1, ro, ci, va) == C.RC_OK
    assert m.A.values.dtype == torch.complex128
    rc, bh = C.AMGX_vector_create(res, "hZZI")
    rc, xh = C.AMGX_vector_create(res, "hZZI")
    C.AMGX_vector_upload(bh, n, 1, np.ones(n, dtype=np.complex128))
    C.AMGX_vector_set_zero(xh, n, 1)
    rc, s = C.AMGX_solver_create(res, "hZZI", cfg)
    assert C.AMGX_solver_setup(s, m) == C.RC_OK
    assert C.AMGX_solver_solve(s, bh, xh) == C.RC_OK
    rc, nrm = C.AMGX_solver_calculate_residual_norm(s, m, bh, xh)
    assert rc == C.RC_OK and nrm < 1e-8


def test_ilu_k_levels():
    """ILU(1) (level-1 fill, reference csr_sparsity_ilu1 + ilu_sparsity_level)
    preconditions better than ILU(0)."""
    from amgx_amd.solvers.ilu import extended_sparsity
    A = poisson_2d(14, 14)
    E = extended_sparsity(A, 1)
    assert E.nnz > A.nnz            # fill added
    assert E.n_rows == A.n_rows

    def iters(level):
        s = make({"solver": "FGMRES", "max_iters": 200,
                  "gmres_n_restart": 40, "monitor_residual": 1,
                  "tolerance": 1e-10, "convergence": "RELATIVE_INI",
                  "preconditioner": {"solver": "MULTICOLOR_ILU",
                                     "max_iters": 1,
                                     "ilu_sparsity_level": level}})
        b = torch.ones(A.n_rows, dtype=torch.float64)
        x = torch.zeros_like(b)
        s.setup(A)
        st = s.solve(b, x, zero_initial_guess=True)
        rel = float(ops.nrm2(ops.residual(A, x, b)) / ops.nrm2(b))
        assert st.converged and rel < 1e-8, f"ILU({level}): {st} rel={rel}"
        return st.iterations

    it0, it1 = iters(0), iters(1)
    assert it1 <= it0, f"ILU(1) {it1} !<= ILU(0) {it0}"


def test_min_iters_and_l1_scaled_norm():
    """min_iters forces extra iterations; L1_SCALED divides by length
    (reference norm machinery, src/solvers/solver.cu:209-260)."""
    A = poisson_2d(8, 8)
    s = make({"solver": "CG", "max_iters": 60, "min_iters": 12,
              "monitor_residual": 1, "tolerance": 1e-3,
              "convergence": "RELATIVE_INI"})
    b = torch.ones(A.n_rows, dtype=torch.float64)
    x = torch.zeros_like(b)
    s.setup(A)
    st = s.solve(b, x, zero_initial_guess=True)
    assert st.converged and st.iterations >= 12
    s2 = make({"solver": "CG", "max_iters": 200, "monitor_residual": 1,
               "tolerance": 1e-10, "convergence": "ABSOLUTE",
               "norm": "L1_SCALED"})
    x2 = torch.zeros_like(b)
    s2.setup(A)
    st2 = s2.solve(b, x2, zero_initial_guess=True)
    assert st2.converged
    r = ops.residual(A, x2, b)
    assert float(ops.nrm1(r)) / A.n_rows < 1e-9


def test_truncated_fgmres():
    """gmres_krylov_dim truncates the orthogonalization window (reference
    src/core.cu:391) — still solves SPD Poisson with AMG preconditioning."""
    A = poisson_3d(8, 8, 8)
    s = make({"solver": "FGMRES", "max_iters": 150, "gmres_n_restart": 50,
              "gmres_krylov_dim": 4, "monitor_residual": 1,
              "tolerance": 1e-8, "convergence": "RELATIVE_INI",
              "preconditioner": {"solver": "AMG",
                                 "algorithm": "AGGREGATION",
                                 "smoother": "BLOCK_JACOBI", "max_iters": 1,
                                 "min_coarse_rows": 16, "scope": "amg"}})
    b = torch.ones(A.n_rows, dtype=torch.float64)
    x = torch.zeros_like(b)
    s.setup(A)
    st = s.solve(b, x, zero_initial_guess=True)
    rel = float(ops.nrm2(ops.residual(A, x, b)) / ops.nrm2(b))
    assert st.converged and rel < 1e-7


def test_fixcolor_gs():
    """FIXCOLOR_GS: the fixed 2x2x2 lattice coloring is valid on a cube and
    smooths like multicolor GS; non-cubes fall back to a computed coloring."""
    from amgx_amd.problems import poisson_3d, poisson_2d
    from amgx_amd.config import ConfigScope
    for prob in (poisson_3d(8, 8, 8), poisson_2d(9, 7)):
        s = create_solver(ConfigScope(None, {"solver": "FIXCOLOR_GS",
                                             "max_iters": 30}),
                          resources=Resources("cpu"))
        b = torch.ones(prob.n_rows, dtype=torch.float64)
        x = torch.zeros_like(b)
        s.setup(prob)
        r0 = ops.nrm2(ops.residual(prob, x, b))
        s.solve(b, x)
        r1 = ops.nrm2(ops.residual(prob, x, b))
        assert r1 < 0.2 * r0, (prob.n_rows, r1 / r0)
    # the cube case really used the lattice: exactly 8 colors
    A = poisson_3d(8, 8, 8)
    s = create_solver(ConfigScope(None, {"solver": "FIXCOLOR_GS",
                                         "max_iters": 1}),
                      resources=Resources("cpu"))
    s.setup(A)
    assert A.coloring.num_colors == 8
    assert A.coloring.validate(A)


def test_blocked_norms():
    """Block matrices use per-block-component norms by default (reference
    src/norm.cu); use_scalar_norm=1 restores the flat norm."""
    import numpy as np
    from amgx_amd.config import ConfigScope
    from amgx_amd.problems import block_laplacian
    A = block_laplacian(6, 6, block_dim=3)
    n = A.n_rows * 3
    g = torch.Generator().manual_seed(5)
    r = torch.rand(n, generator=g, dtype=torch.float64)
    comp = r.reshape(-1, 3).numpy()
    for norm, ref in (
            ("L2", np.linalg.norm(comp, axis=0).max()),
            ("L1", np.abs(comp).sum(0).max()),
            ("LMAX", np.abs(comp).max(0).max())):
        s = create_solver(ConfigScope(None, {"solver": "BLOCK_JACOBI",
                                             "max_iters": 1, "norm": norm}),
                          resources=Resources("cpu"))
        s.A = A
        assert abs(s.compute_norm(r) - ref) < 1e-12, norm
        s2 = create_solver(ConfigScope(None, {"solver": "BLOCK_JACOBI",
                                              "max_iters": 1, "norm": norm,
                                              "use_scalar_norm": 1}),
                           resources=Resources("cpu"))
        s2.A = A
        flat = {"L2": np.linalg.norm(r.numpy()),
                "L1": np.abs(r.numpy()).sum(),
                "LMAX": np.abs(r.numpy()).max()}[norm]
        assert abs(s2.compute_norm(r) - flat) < 1e-12, norm
    # and a full block solve still converges under the blocked norm
    cfg = {"solver": {"solver": "PCG", "preconditioner": "BLOCK_JACOBI",
                      "max_iters": 500, "monitor_residual": 1,
                      "tolerance": 1e-8, "convergence": "RELATIVE_INI"}}
    from amgx_amd import AMGConfig
    s = create_solver(AMGConfig.from_dict(cfg).root_scope(),
                      resources=Resources("cpu"))
    b = torch.ones(n, dtype=torch.float64)
    x = torch.zeros_like(b)
    s.setup(A)
    st = s.solve(b, x, zero_initial_guess=True)
    assert st.converged


def test_unstructured_random_laplacian():
    """FGMRES+ILU(0) and PCG+AMG on an unstructured random graph Laplacian
    (the offline SuiteSparse stand-in, irregular degrees)."""
    from amgx_amd import AMGConfig
    from amgx_amd.problems import random_laplacian
    A = random_laplacian(400, avg_degree=10, seed=3)
    for cfg in (
        {"solver": {"preconditioner": {"solver": "MULTICOLOR_ILU",
                                       "max_iters": 1, "scope": "i"},
                    "solver": "FGMRES", "max_iters": 200,
                    "gmres_n_restart": 30, "monitor_residual": 1,
                    "convergence": "RELATIVE_INI", "tolerance": 1e-8}},
        {"solver": {"preconditioner": {"solver": "AMG",
                                       "algorithm": "AGGREGATION",
                                       "smoother": "MULTICOLOR_GS",
                                       "symmetric_GS": 1, "max_iters": 1,
                                       "min_coarse_rows": 10, "cycle": "V"},
                    "solver": "PCG", "max_iters": 200,
                    "monitor_residual": 1,
                    "convergence": "RELATIVE_INI", "tolerance": 1e-8}},
    ):
        s = create_solver(AMGConfig.from_dict(cfg).root_scope(),
                          resources=Resources("cpu"))
        b = torch.ones(A.n_rows, dtype=torch.float64)
        x = torch.zeros_like(b)
        s.setup(A)
        st = s.solve(b, x, zero_initial_guess=True)
        rel = ops.nrm2(ops.residual(A, x, b)) / ops.nrm2(b)
        assert st.converged and rel < 1e-7, st


def test_anisotropic_and_nonsymmetric():
    """Anisotropic diffusion under classical AMG+PCG and nonsymmetric
    convection-diffusion under BiCGStab/GMRES/IDR+ILU(0) (reference CUSP
    generator roles; nonsymmetric Krylov coverage)."""
    from amgx_amd import AMGConfig
    from amgx_amd.problems import anisotropic_2d, convection_diffusion_2d
    A = anisotropic_2d(30, 30, eps=0.01)
    cfg = {"solver": {"preconditioner": {
        "solver": "AMG", "algorithm": "CLASSICAL",
        "smoother": "MULTICOLOR_GS", "symmetric_GS": 1, "max_iters": 1,
        "min_coarse_rows": 10, "cycle": "V"},
        "solver": "PCG", "max_iters": 200, "monitor_residual": 1,
        "convergence": "RELATIVE_INI", "tolerance": 1e-8}}
    s = create_solver(AMGConfig.from_dict(cfg).root_scope(),
                      resources=Resources("cpu"))
    b = torch.ones(A.n_rows, dtype=torch.float64)
    x = torch.zeros_like(b)
    s.setup(A)
    st = s.solve(b, x, zero_initial_guess=True)
    assert st.converged and st.iterations < 40, st

    An = convection_diffusion_2d(30, 30, beta=20.0)
    for outer in ("BICGSTAB", "GMRES", "IDR"):
        cfg = {"solver": {"preconditioner": {"solver": "MULTICOLOR_ILU",
                                             "max_iters": 1, "scope": "i"},
               "solver": outer, "max_iters": 300, "gmres_n_restart": 40,
               "monitor_residual": 1, "convergence": "RELATIVE_INI",
               "tolerance": 1e-8}}
        s = create_solver(AMGConfig.from_dict(cfg).root_scope(),
                          resources=Resources("cpu"))
        b = torch.ones(An.n_rows, dtype=torch.float64)
        x = torch.zeros_like(b)
        s.setup(An)
        st = s.solve(b, x, zero_initial_guess=True)
        rel = ops.nrm2(ops.residual(An, x, b)) / ops.nrm2(b)
        assert st.converged and rel < 1e-6, (outer, st)


def test_degenerate_matrices():
    """Reference zero_{in,off}_diagonal / zero_values handling tests:
    diagonal-only operators, zero diagonal entries and explicit zeros must
    not produce NaN/inf anywhere in setup or solve."""
    import numpy as np
    import scipy.sparse as sp
    from amgx_amd import AMGConfig
    from amgx_amd.matrix import CSRMatrix

    # 1. diagonal-only matrix: no strong connections, AMG degenerates cleanly
    D = CSRMatrix.from_scipy(sp.diags(np.arange(1.0, 41.0)).tocsr())
    cfg = {"solver": {"preconditioner": {
        "solver": "AMG", "algorithm": "AGGREGATION",
        "smoother": "BLOCK_JACOBI", "max_iters": 1, "min_coarse_rows": 4,
        "cycle": "V"},
        "solver": "PCG", "max_iters": 60, "monitor_residual": 1,
        "convergence": "RELATIVE_INI", "tolerance": 1e-10}}
    s = create_solver(AMGConfig.from_dict(cfg).root_scope(),
                      resources=Resources("cpu"))
    b = torch.ones(40, dtype=torch.float64)
    x = torch.zeros_like(b)
    s.setup(D)
    st = s.solve(b, x, zero_initial_guess=True)
    assert st.converged and torch.isfinite(x).all()
    assert torch.allclose(x, 1.0 / torch.arange(1.0, 41.0,
                                                dtype=torch.float64),
                          atol=1e-8)

    # 2. one zero diagonal entry: Jacobi dinv must stay finite
    m = sp.diags([np.full(19, -1.0), np.full(20, 4.0), np.full(19, -1.0)],
                 [-1, 0, 1]).tocsr().astype(float)
    m = m.tolil()
    m[7, 7] = 0.0
    A = CSRMatrix.from_scipy(m.tocsr())
    dinv = ops.jacobi_dinv(A)
    assert torch.isfinite(dinv).all()
    dinv = ops.jacobi_dinv(A, l1=True)
    assert torch.isfinite(dinv).all()

    # 3. explicit zeros in a user-provided pattern are PRESERVED (the
    # structure is the replace_coefficients contract) and ops handle them
    m2 = sp.csr_matrix((np.array([4.0, 0.0, -1.0, 4.0]),
                        np.array([0, 1, 0, 1]), np.array([0, 2, 4])),
                       shape=(2, 2))
    A2 = CSRMatrix.from_scipy(m2)
    assert A2.nnz == 4
    y = ops.spmv(A2, torch.tensor([1.0, 2.0], dtype=torch.float64))
    assert torch.allclose(y, torch.tensor([4.0, 7.0], dtype=torch.float64))


def test_mixed_precision_amg_cycle():
    """hDFI through the full AMG stack: fp32 matrix/hierarchy, fp64 vectors
    (reference mixed mode dDFI; cycle scratch re-keys to the rhs dtype)."""
    from amgx_amd import AMGConfig
    from amgx_amd.matrix import CSRMatrix
    from amgx_amd.problems import poisson_3d
    A64 = poisson_3d(10, 10, 10)
    A = CSRMatrix(A64.row_offsets, A64.col_indices,
                  A64.values.to(torch.float32), n_cols=A64.n_cols)
    cfg = {"solver": {"preconditioner": {
        "solver": "AMG", "algorithm": "AGGREGATION",
        "smoother": "MULTICOLOR_DILU", "max_iters": 1,
        "min_coarse_rows": 10, "cycle": "V"},
        "solver": "FGMRES", "max_iters": 150, "gmres_n_restart": 30,
        "monitor_residual": 1,
        "convergence": "RELATIVE_INI", "tolerance": 1e-6}}
    s = create_solver(AMGConfig.from_dict(cfg).root_scope(),
                      resources=Resources("cpu"))
    b = torch.ones(A.n_rows, dtype=torch.float64)
    x = torch.zeros_like(b)
    s.setup(A)
    st = s.solve(b, x, zero_initial_guess=True)
    assert x.dtype == torch.float64
    rel = ops.nrm2(ops.residual(A64, x, b)) / ops.nrm2(b)
    assert st.converged and rel < 1e-5, (st, rel)


def test_block_ilu0():
    """Block ILU(0) on the host path (reference multicolor_ilu_solver.cu
    bxb setup_LU): smooths a block-4 system and preconditions FGMRES."""
    for bd in (2, 4):
        A = block_laplacian(8, 8, block_dim=bd)
        n = A.n_rows * bd
        s = make({"solver": "MULTICOLOR_ILU", "max_iters": 6})
        b = torch.ones(n, dtype=torch.float64)
        x = torch.zeros_like(b)
        s.setup(A)
        r0 = ops.nrm2(ops.residual(A, x, b))
        s.solve(b, x)
        r1 = ops.nrm2(ops.residual(A, x, b))
        assert r1 < 0.6 * r0, (bd, r1 / r0)
    A = block_laplacian(10, 10, block_dim=4)
    s = make({"solver": "FGMRES", "max_iters": 200, "gmres_n_restart": 30,
              "monitor_residual": 1, "tolerance": 1e-8,
              "convergence": "RELATIVE_INI",
              "preconditioner": {"solver": "MULTICOLOR_ILU",
                                 "max_iters": 1}})
    n = A.n_rows * 4
    b = torch.ones(n, dtype=torch.float64)
    x = torch.zeros_like(b)
    s.setup(A)
    st = s.solve(b, x, zero_initial_guess=True)
    rel = float(ops.nrm2(ops.residual(A, x, b)) / ops.nrm2(b))
    assert st.converged and rel < 1e-7, (st, rel)


def test_user_solver_hook():
    """USER solver plugin (reference src/solvers/user_solver.cu): each
    iteration invokes the registered callback(A, b, x); convergence is
    monitored by the standard machinery."""
    from amgx_amd.solvers import UserSolver
    A = poisson_2d(16, 16)
    calls = []

    import scipy.sparse.linalg as spla
    As = A.to_scipy().tocsc()
    lu = spla.splu(As)

    def direct_cb(M, b, x):
        calls.append(1)
        x.copy_(torch.from_numpy(lu.solve(b.numpy())))

    UserSolver.set_callback(direct_cb)
    try:
        s = make({"solver": "USER", "max_iters": 50, "monitor_residual": 1,
                  "convergence": "RELATIVE_INI", "tolerance": 1e-3})
        b = torch.ones(A.n_rows, dtype=torch.float64)
        x = torch.zeros_like(b)
        s.setup(A)
        st = s.solve(b, x, zero_initial_guess=True)
        assert calls, "callback never invoked"
        rel = ops.nrm2(ops.residual(A, x, b)) / ops.nrm2(b)
        assert st.converged and rel < 1e-3, (st.iterations, rel)
    finally:
        UserSolver.set_callback(None)
    # unset callback raises loudly
    s2 = make({"solver": "USER", "max_iters": 2})
    s2.setup(A)
    with pytest.raises(RuntimeError):
        s2.solve(torch.ones(A.n_rows, dtype=torch.float64),
                 torch.zeros(A.n_rows, dtype=torch.float64))


def test_cf_jacobi_mode_default():
    """cf_smoothing_mode defaults to 0 = CF ordering (reference
    src/core.cu:416); the parameter is registered so an unset config does
    not silently pick the FC ordering."""
    from amgx_amd.config import default_of
    assert default_of("cf_smoothing_mode") == 0
    s = make({"solver": "CF_JACOBI", "max_iters": 3})
    assert s.mode == 0


def test_ilu_dilu_equivalence():
    """DILU == ILU(0) on triangle-free graphs (5-pt Poisson): identical
    iterates after 10 sweeps (reference src/tests/ilu_dilu_equivalence.cu,
    which checks the same to 1e-10 on a florida matrix)."""
    A = poisson_2d(16, 16)
    b = torch.ones(A.n_rows, dtype=torch.float64)
    xs = {}
    for name, extra in (("DILU", {"solver": "MULTICOLOR_DILU"}),
                        ("ILU", {"solver": "MULTICOLOR_ILU",
                                 "ilu_sparsity_level": 0})):
        cfg = {"max_iters": 10, "monitor_residual": 1,
               "max_uncolored_percentage": 0.0, "coloring_level": 1,
               "relaxation_factor": 1.0, "weight": 1.0}
        cfg.update(extra)
        s = make(cfg)
        x = torch.zeros_like(b)
        s.setup(A)
        s.solve(b, x, zero_initial_guess=True)
        xs[name] = x
    assert torch.allclose(xs["DILU"], xs["ILU"], atol=1e-10), \
        (xs["DILU"] - xs["ILU"]).abs().max()


def test_block_sizes_6_to_10_spmv_and_jacobi():
    """Block sizes 6..10: SpMV matches scipy and block-Jacobi smoothing
    reduces the residual (reference smoother_blocksizes.cu covers 2-10)."""
    for bd in (6, 7, 8, 9, 10):
        A = block_laplacian(5, 5, block_dim=bd)
        n = A.n_rows * bd
        x = torch.rand(n, dtype=torch.float64)
        assert np.allclose(ops.spmv(A, x).numpy(), A.to_scipy() @ x.numpy())
        s = make({"solver": "BLOCK_JACOBI", "max_iters": 8})
        b = torch.ones(n, dtype=torch.float64)
        xz = torch.zeros_like(b)
        s.setup(A)
        r0 = ops.nrm2(ops.residual(A, xz, b))
        s.solve(b, xz)
        r1 = ops.nrm2(ops.residual(A, xz, b))
        assert np.isfinite(r1) and r1 < 0.9 * r0, (bd, r1 / r0)


def test_smoother_random_matrix_stays_finite():
    """Smoothers on a random diagonally-heavy unsymmetric matrix never
    produce non-finite values (reference smoother_nan_random.cu)."""
    rng = np.random.default_rng(5)
    n = 120
    import scipy.sparse as sp
    M = sp.random(n, n, density=0.06, random_state=7, format="csr")
    M = M + sp.identity(n) * (np.abs(M).sum(axis=1).max() + 1.0)
    from amgx_amd.matrix import CSRMatrix
    A = CSRMatrix.from_scipy(M.tocsr())
    b = torch.from_numpy(rng.standard_normal(n))
    for name in ("BLOCK_JACOBI", "JACOBI_L1", "GS", "MULTICOLOR_GS",
                 "MULTICOLOR_DILU", "MULTICOLOR_ILU", "KACZMARZ",
                 "CHEBYSHEV", "POLYNOMIAL"):
        s = make({"solver": name, "max_iters": 5})
        x = torch.zeros(n, dtype=torch.float64)
        s.setup(A)
        s.solve(b, x)
        assert torch.isfinite(x).all(), name


def test_chebyshev_lambda_modes():
    """All four reference lambda-estimate modes (cheb_solver.cu:180-213):
    0/1 power iteration, 2 preconditioned-spectrum assumption (lmax=0.9),
    3 user-provided bounds."""
    A = poisson_2d(12, 12)
    for mode, extra in ((0, {}), (1, {}), (2, {}),
                        (3, {"cheby_max_lambda": 1.9,
                             "cheby_min_lambda": 0.1})):
        cfg = {"solver": "CHEBYSHEV", "max_iters": 10,
               "chebyshev_lambda_estimate_mode": mode}
        cfg.update(extra)
        s = make(cfg)
        b = torch.ones(A.n_rows, dtype=torch.float64)
        x = torch.zeros_like(b)
        s.setup(A)
        if mode == 2:
            assert s.lmax == 0.9
        if mode == 3:
            assert s.lmax == 1.9 and s.lmin == 0.1
        r0 = ops.nrm2(ops.residual(A, x, b))
        s.solve(b, x)
        r1 = ops.nrm2(ops.residual(A, x, b))
        assert np.isfinite(r1) and r1 < r0, (mode, r1 / r0)
