import numpy as np
import pytest
import torch

from amgx_amd import AMGConfig, create_solver, ops
from amgx_amd.resources import Resources
from amgx_amd.problems import poisson_2d, poisson_3d

FGMRES_AGG = {
    "solver": {
        "preconditioner": {
            "algorithm": "AGGREGATION",
            "solver": "AMG",
            "smoother": "MULTICOLOR_DILU",
            "presweeps": 0,
            "postsweeps": 3,
            "selector": "SIZE_2",
            "coarse_solver": "DENSE_LU_SOLVER",
            "max_iters": 1,
            "min_coarse_rows": 32,
            "relaxation_factor": 0.75,
            "scope": "amg",
            "max_levels": 50,
            "cycle": "V",
        },
        "solver": "FGMRES",
        "max_iters": 100,
        "gmres_n_restart": 10,
        "monitor_residual": 1,
        "convergence": "RELATIVE_INI",
        "tolerance": 1e-6,
        "norm": "L2",
    }
}

PCG_CLASSICAL = {
    "solver": {
        "preconditioner": {
            "solver": "AMG",
            "algorithm": "CLASSICAL",
            "smoother": {"solver": "BLOCK_JACOBI", "relaxation_factor": 0.8},
            "presweeps": 1,
            "postsweeps": 1,
            "max_iters": 1,
            "min_coarse_rows": 10,
            "max_levels": 20,
            "cycle": "V",
        },
        "solver": "PCG",
        "max_iters": 100,
        "monitor_residual": 1,
        "convergence": "RELATIVE_INI",
        "tolerance": 1e-6,
    }
}


def run(cfg_dict, A, max_expected_iters, tol=1e-6):
    cfg = AMGConfig.from_dict(cfg_dict)
    s = create_solver(cfg.root_scope(), resources=Resources("cpu"))
    b = torch.ones(A.n_rows, dtype=torch.float64)
    x = torch.zeros_like(b)
    s.setup(A)
    st = s.solve(b, x, zero_initial_guess=True)
    r = ops.residual(A, x, b)
    rel = ops.nrm2(r) / ops.nrm2(b)
    assert st.converged, f"not converged: {st}"
    assert rel < 10 * tol
    assert st.iterations <= max_expected_iters, st.iterations
    return st


def test_amg_aggregation_standalone():
    """Plain AMG V-cycles as the solver (aggregation)."""
    A = poisson_2d(24, 24)
    cfg = {
        "solver": {
            "solver": "AMG",
            "algorithm": "AGGREGATION",
            "selector": "SIZE_2",
            "smoother": "MULTICOLOR_DILU",
            "presweeps": 1, "postsweeps": 1,
            "min_coarse_rows": 16,
            "max_iters": 60,
            "monitor_residual": 1,
            "convergence": "RELATIVE_INI",
            "tolerance": 1e-8,
        }
    }
    st = run(cfg, A, max_expected_iters=60, tol=1e-8)


def test_fgmres_aggregation_poisson2d():
    """The flagship FGMRES_AGGREGATION config (driver config class #2)."""
    A = poisson_2d(24, 24)
    st = run(FGMRES_AGG, A, max_expected_iters=30)


def test_fgmres_aggregation_poisson3d():
    A = poisson_3d(8, 8, 8)
    st = run(FGMRES_AGG, A, max_expected_iters=30)


def test_pcg_classical_poisson2d():
    A = poisson_2d(24, 24)
    st = run(PCG_CLASSICAL, A, max_expected_iters=30)


def test_pcg_classical_poisson3d():
    A = poisson_3d(8, 8, 8)
    st = run(PCG_CLASSICAL, A, max_expected_iters=30)


def test_w_cycle():
    A = poisson_2d(20, 20)
    cfg = {"solver": dict(FGMRES_AGG["solver"])}
    cfg["solver"]["preconditioner"] = dict(FGMRES_AGG["solver"]["preconditioner"],
                                           cycle="W")
    run(cfg, A, max_expected_iters=30)


def test_grid_stats():
    A = poisson_2d(24, 24)
    from amgx_amd.amg.amg import AMGHierarchy
    cfg = AMGConfig.from_dict(FGMRES_AGG)
    _, sub = cfg.root_scope().sub_solver("preconditioner")
    h = AMGHierarchy(sub, Resources("cpu"))
    h.setup(A)
    stats = h.grid_stats()
    assert "Number of Levels" in stats and len(h.levels) >= 3
    # aggregation should roughly halve rows per level
    assert h.levels[1].A.n_rows < 0.7 * A.n_rows


def test_f_cycle():
    A = poisson_2d(20, 20)
    cfg = {"solver": dict(FGMRES_AGG["solver"])}
    cfg["solver"]["preconditioner"] = dict(FGMRES_AGG["solver"]["preconditioner"],
                                           cycle="F")
    run(cfg, A, max_expected_iters=30)


def test_cg_kcycle():
    A = poisson_2d(20, 20)
    cfg = {"solver": dict(FGMRES_AGG["solver"])}
    cfg["solver"]["preconditioner"] = dict(FGMRES_AGG["solver"]["preconditioner"],
                                           cycle="CG")
    st = run(cfg, A, max_expected_iters=30)


def test_cycle_ordering():
    """W/F/CG cycles should not be WORSE than V on iteration count."""
    A = poisson_3d(10, 10, 10)
    iters = {}
    for cyc in ("V", "W", "F", "CG"):
        cfg = {"solver": dict(FGMRES_AGG["solver"])}
        cfg["solver"]["preconditioner"] = dict(
            FGMRES_AGG["solver"]["preconditioner"], cycle=cyc)
        st = run(cfg, A, max_expected_iters=40)
        iters[cyc] = st.iterations
    assert iters["W"] <= iters["V"] + 2
    assert iters["CG"] <= iters["V"] + 2


# ----------------------------------------------------------- coloring schemes
def test_coloring_schemes_valid():
    """Every registered coloring scheme yields a valid distance-1 (or
    distance-2 for *2RING) coloring on an unstructured-ish matrix (reference
    src/tests/valid_coloring.cu + matrix_coloring_test.cu)."""
    import numpy as np

    from amgx_amd.amg.coloring import COLORING_REGISTRY, MatrixColoring
    from amgx_amd.config import AMGConfig
    from amgx_amd.problems import poisson_3d
    A = poisson_3d(5, 4, 3)
    for scheme in sorted(COLORING_REGISTRY):
        cfg = AMGConfig.from_dict({"solver": {
            "solver": "MULTICOLOR_GS",
            "matrix_coloring_scheme": scheme}})
        col = MatrixColoring.create(A, cfg.root_scope())
        assert col.colors.numel() == A.n_rows
        assert col.num_colors >= 1
        level = 2 if "2RING" in scheme else 1
        if scheme != "UNIFORM":   # UNIFORM is index-pattern, like reference
            assert col.validate(A, level=level), f"{scheme} invalid"
        # bounds cover all rows exactly once
        assert col.bounds[-1] == A.n_rows


def test_coloring_2ring_is_distance2():
    from amgx_amd.amg.coloring import MatrixColoring
    from amgx_amd.config import AMGConfig
    from amgx_amd.problems import poisson_3d
    A = poisson_3d(4, 4, 2)
    cfg = AMGConfig.from_dict({"solver": {
        "solver": "MULTICOLOR_GS",
        "matrix_coloring_scheme": "MIN_MAX_2RING"}})
    col = MatrixColoring.create(A, cfg.root_scope())
    assert col.validate(A, level=2)


def test_coloring_level2_any_scheme():
    """coloring_level=2 upgrades any scheme to the distance-2 graph
    (reference coloring_level, src/core.cu:489)."""
    from amgx_amd.amg.coloring import MatrixColoring
    from amgx_amd.config import AMGConfig
    from amgx_amd.problems import poisson_3d
    A = poisson_3d(4, 3, 3)
    cfg = AMGConfig.from_dict({"solver": {
        "solver": "MULTICOLOR_GS",
        "matrix_coloring_scheme": "PARALLEL_GREEDY",
        "coloring_level": 2}})
    col = MatrixColoring.create(A, cfg.root_scope())
    assert col.validate(A, level=2)


# --------------------------------------- classical strength/selector/interp
def _classical_cfg(**overrides):
    from amgx_amd.config import AMGConfig
    node = {
        "preconditioner": {
            "solver": "AMG", "algorithm": "CLASSICAL",
            "smoother": "MULTICOLOR_GS", "presweeps": 1, "postsweeps": 1,
            "max_iters": 1, "min_coarse_rows": 10, "cycle": "V",
            "scope": "amg",
        },
        "solver": "PCG", "max_iters": 80, "monitor_residual": 1,
        "convergence": "RELATIVE_INI", "tolerance": 1e-8,
    }
    node["preconditioner"].update(overrides)
    return AMGConfig.from_dict({"solver": node})


def _solve_classical(cfg, n=10):
    import torch

    from amgx_amd import create_solver, ops
    from amgx_amd.problems import poisson_3d
    from amgx_amd.resources import Resources
    A = poisson_3d(n, n, n)
    s = create_solver(cfg.root_scope(), resources=Resources("cpu"))
    b = torch.ones(A.n_rows, dtype=torch.float64)
    x = torch.zeros_like(b)
    s.setup(A)
    st = s.solve(b, x, zero_initial_guess=True)
    rel = float(ops.nrm2(ops.residual(A, x, b)) / ops.nrm2(b))
    return st, rel


def test_classical_selectors_converge():
    """HMIS / RS / CR selectors build hierarchies that PCG+V-cycle solves
    (reference src/tests/classical_pmis.cu analogue for each selector)."""
    for sel in ("PMIS", "HMIS", "RS", "CR"):
        st, rel = _solve_classical(_classical_cfg(selector=sel), n=8)
        assert st.converged and rel < 1e-7, f"{sel}: {st}, rel={rel}"
        assert st.iterations <= 40, f"{sel} took {st.iterations}"


def test_classical_strength_metrics():
    for strength in ("AHAT", "ALL", "AFFINITY"):
        st, rel = _solve_classical(_classical_cfg(strength=strength), n=8)
        assert st.converged and rel < 1e-7, f"{strength}: {st}, rel={rel}"


def test_classical_interpolators():
    """D2 (standard distance-2) and MULTIPASS interpolation converge; D2
    reproduces constants away from the boundary."""
    for interp in ("D1", "D2", "MULTIPASS"):
        st, rel = _solve_classical(_classical_cfg(interpolator=interp), n=8)
        assert st.converged and rel < 1e-7, f"{interp}: {st}, rel={rel}"


def test_aggressive_coarsening():
    """aggressive_levels=1 with MULTIPASS interpolation: much smaller level-1
    grid than plain PMIS, still convergent (reference aggressive_pmis.cu +
    multipass.cu)."""
    import torch

    from amgx_amd import create_solver
    from amgx_amd.problems import poisson_3d
    from amgx_amd.resources import Resources
    A = poisson_3d(8, 8, 8)

    def coarse_rows(cfg):
        s = create_solver(cfg.root_scope(), resources=Resources("cpu"))
        s.setup(A)
        h = s.precond.hierarchy
        return h.levels[1].A.n_rows if len(h.levels) > 1 else A.n_rows

    plain = coarse_rows(_classical_cfg())
    agg = coarse_rows(_classical_cfg(aggressive_levels=1))
    assert agg < plain, f"aggressive {agg} !< plain {plain}"
    st, rel = _solve_classical(_classical_cfg(aggressive_levels=1), n=8)
    assert st.converged and rel < 1e-7


def test_d2_interp_reproduces_constant():
    """P applied to the all-ones coarse vector is 1 on rows with full
    row-sum-zero stencils (interior): standard interpolation exactness."""
    import numpy as np
    import torch

    from amgx_amd.amg.classical import (INTERP_REGISTRY, SELECTOR_REGISTRY,
                                        STRENGTH_REGISTRY)
    from amgx_amd.problems import poisson_3d
    from amgx_amd.config import AMGConfig
    cfg = _classical_cfg()
    scope = cfg.root_scope().sub_solver("preconditioner", "AMG")[1]
    A = poisson_3d(6, 6, 6)
    # make row sums exactly zero (pure Neumann-like stencil) so constants
    # are in the near-nullspace
    m = A.to_scipy().tolil()
    for i in range(A.n_rows):
        m[i, i] -= m[i].sum()
    from amgx_amd.matrix import CSRMatrix
    A0 = CSRMatrix.from_scipy(m.tocsr(), dtype=torch.float64)
    S = STRENGTH_REGISTRY["AHAT"](A0, scope)
    cf, nc = SELECTOR_REGISTRY["PMIS"](A0, S, scope)
    P = INTERP_REGISTRY["D2"](A0, S, cf, nc, scope)
    ones_c = np.ones(nc)
    Pv = P.to_scipy() @ ones_c
    assert np.allclose(Pv, 1.0, atol=1e-10), \
        f"max dev {np.abs(Pv - 1).max()}"


def test_d2_device_formulation_matches_host():
    """The torch-ops device D2 (_interp_d2_device) must produce the same P
    as the scipy host path — run both on CPU tensors and compare."""
    import numpy as np

    from amgx_amd.amg.classical import (SELECTOR_REGISTRY, STRENGTH_REGISTRY,
                                        _interp_d2_device, _interp_d2_host)
    from amgx_amd.problems import poisson_3d
    cfg = _classical_cfg()
    scope = cfg.root_scope().sub_solver("preconditioner", "AMG")[1]
    for seed, build in enumerate([
            lambda: poisson_3d(5, 6, 4),
            lambda: poisson_3d(7, 7, 3)]):
        A = build()
        S = STRENGTH_REGISTRY["AHAT"](A, scope)
        cf, nc = SELECTOR_REGISTRY["PMIS"](A, S, scope)
        Ph = _interp_d2_host(A, S, cf, nc, scope).to_scipy()
        Pd = _interp_d2_device(A, S, cf, nc, scope).to_scipy()
        diff = abs(Ph - Pd)
        assert Ph.shape == Pd.shape
        assert diff.nnz == 0 or diff.max() < 1e-12, \
            f"seed {seed}: max dev {diff.max()}"


def test_multipass_device_formulation_matches_host():
    """The torch-ops device MULTIPASS must produce the same P as the scipy
    host path — run both on CPU tensors and compare (aggressive-coarsening
    C/F split so several passes actually happen)."""
    from amgx_amd.amg.classical import (SELECTOR_REGISTRY, STRENGTH_REGISTRY,
                                        _interp_multipass_device,
                                        interp_multipass)
    from amgx_amd.problems import poisson_3d
    cfg = _classical_cfg()
    scope = cfg.root_scope().sub_solver("preconditioner", "AMG")[1]
    A = poisson_3d(6, 5, 5)
    S = STRENGTH_REGISTRY["AHAT"](A, scope)
    cf, nc = SELECTOR_REGISTRY["AGGRESSIVE_PMIS"](A, S, scope)
    Ph = interp_multipass(A, S, cf, nc, scope).to_scipy()
    Pd = _interp_multipass_device(A, S, cf, nc, scope).to_scipy()
    diff = abs(Ph - Pd)
    assert Ph.shape == Pd.shape
    assert diff.nnz == 0 or diff.max() < 1e-12, f"max dev {diff.max()}"


# -------------------------------------------------- aggregation selectors
def test_aggregation_selectors():
    """Every aggregation selector yields a valid aggregate map and a
    convergent FGMRES+AGG hierarchy (reference aggregates_*.cu tests)."""
    import torch

    from amgx_amd import AMGConfig, create_solver, ops
    from amgx_amd.amg.aggregation import AGG_SELECTOR_REGISTRY
    from amgx_amd.problems import poisson_3d
    from amgx_amd.resources import Resources
    A = poisson_3d(8, 8, 8)
    A._cache["geometry"] = __import__("numpy").stack(
        __import__("numpy").meshgrid(range(8), range(8), range(8),
                                     indexing="ij"), -1).reshape(-1, 3)
    for sel in sorted(AGG_SELECTOR_REGISTRY):
        cfg = AMGConfig.from_dict({"solver": {
            "preconditioner": {
                "solver": "AMG", "algorithm": "AGGREGATION",
                "smoother": "BLOCK_JACOBI", "presweeps": 1, "postsweeps": 1,
                "max_iters": 1, "min_coarse_rows": 16, "cycle": "V",
                "selector": sel, "aggregate_size": 4,
            },
            "solver": "FGMRES", "max_iters": 120, "gmres_n_restart": 20,
            "monitor_residual": 1, "convergence": "RELATIVE_INI",
            "tolerance": 1e-6,
        }})
        scope = cfg.root_scope().sub_solver("preconditioner", "AMG")[1]
        agg, num = AGG_SELECTOR_REGISTRY[sel](A, scope)
        assert agg.numel() == A.n_rows
        assert 0 < num < A.n_rows, f"{sel}: num={num}"
        assert int(agg.max()) == num - 1 and int(agg.min()) == 0
        s = create_solver(cfg.root_scope(), resources=Resources("cpu"))
        b = torch.ones(A.n_rows, dtype=torch.float64)
        x = torch.zeros_like(b)
        s.setup(A)
        st = s.solve(b, x, zero_initial_guess=True)
        rel = float(ops.nrm2(ops.residual(A, x, b)) / ops.nrm2(b))
        assert st.converged and rel < 1e-5, f"{sel}: {st} rel={rel}"


def test_energymin_level():
    """ENERGYMIN algorithm (CR selector + EM interpolation) solves Poisson
    (reference src/tests/energymin_algorithm.cu)."""
    st, rel = _solve_classical(_classical_cfg(algorithm="ENERGYMIN"), n=8)
    assert st.converged and rel < 1e-7, f"{st}, rel={rel}"


# --------------------------------------------------------- structure reuse
def test_structure_reuse_resetup():
    """structure_reuse_levels: resetup after a value change keeps the
    aggregates and rebuilds only Galerkin values (reference
    amg_levels_reuse.cu + AMGX_solver_resetup)."""
    import torch

    from amgx_amd import AMGConfig, create_solver, ops
    from amgx_amd.matrix import CSRMatrix
    from amgx_amd.problems import poisson_3d
    from amgx_amd.resources import Resources
    cfg = AMGConfig.from_dict({"solver": {
        "preconditioner": {
            "solver": "AMG", "algorithm": "AGGREGATION",
            "smoother": "BLOCK_JACOBI", "presweeps": 1, "postsweeps": 1,
            "max_iters": 1, "min_coarse_rows": 16, "cycle": "V",
            "structure_reuse_levels": -1, "scope": "amg",
        },
        "solver": "PCG", "max_iters": 100, "monitor_residual": 1,
        "convergence": "RELATIVE_INI", "tolerance": 1e-8,
    }})
    A = poisson_3d(8, 8, 8)
    s = create_solver(cfg.root_scope(), resources=Resources("cpu"))
    b = torch.ones(A.n_rows, dtype=torch.float64)
    x = torch.zeros_like(b)
    s.setup(A)
    h1 = s.precond.hierarchy
    aggs1 = [l.aggregates for l in h1.levels[:-1]]
    st1 = s.solve(b, x, zero_initial_guess=True)
    assert st1.converged
    # scale the values (same structure), resetup, solve again
    A2 = CSRMatrix(A.row_offsets, A.col_indices, A.values * 2.0,
                   n_cols=A.n_cols)
    s.resetup(A2)
    h2 = s.precond.hierarchy
    assert h2 is h1, "hierarchy object must be reused"
    for l, a1 in zip(h2.levels[:-1], aggs1):
        assert l.aggregates is a1, "aggregates must be reused"
    x2 = torch.zeros_like(b)
    st2 = s.solve(b, x2, zero_initial_guess=True)
    assert st2.converged
    rel = float(ops.nrm2(ops.residual(A2, x2, b)) / ops.nrm2(b))
    assert rel < 1e-7
    # solution of 2A x = b is x/2
    assert torch.allclose(x2, x / 2.0, atol=1e-6)


def test_classical_structure_reuse():
    import torch

    from amgx_amd import AMGConfig, create_solver
    from amgx_amd.matrix import CSRMatrix
    from amgx_amd.problems import poisson_3d
    from amgx_amd.resources import Resources
    cfg = _classical_cfg(structure_reuse_levels=-1)
    A = poisson_3d(8, 8, 8)
    s = create_solver(cfg.root_scope(), resources=Resources("cpu"))
    b = torch.ones(A.n_rows, dtype=torch.float64)
    x = torch.zeros_like(b)
    s.setup(A)
    h1 = s.precond.hierarchy
    cf1 = [l.cf_map for l in h1.levels[:-1]]
    A2 = CSRMatrix(A.row_offsets, A.col_indices, A.values * 3.0,
                   n_cols=A.n_cols)
    s.resetup(A2)
    assert s.precond.hierarchy is h1
    for l, c in zip(h1.levels[:-1], cf1):
        assert l.cf_map is c
    st = s.solve(b, x, zero_initial_guess=True)
    assert st.converged


def test_poisson_27pt_classical():
    """27-point stencil (reference cusp::gallery::poisson27pt test fixture):
    SPD, denser rows; classical AMG still converges."""
    import numpy as np
    import torch

    from amgx_amd import create_solver, ops
    from amgx_amd.problems import poisson_3d_27pt
    from amgx_amd.resources import Resources
    A = poisson_3d_27pt(7, 7, 7)
    assert A.n_rows == 343
    d = np.diff(A.row_offsets.numpy())
    assert d.max() == 27
    # SPD check on a small instance
    dense = A.to_scipy().toarray()
    assert np.allclose(dense, dense.T)
    assert np.linalg.eigvalsh(dense).min() > 0
    # dense stencils want distance-2 interpolation + a strong smoother
    # (reference: D2/multipass exist exactly for this row density)
    cfg = _classical_cfg(interpolator="D2", smoother="MULTICOLOR_DILU")
    s = create_solver(cfg.root_scope(), resources=Resources("cpu"))
    b = torch.ones(A.n_rows, dtype=torch.float64)
    x = torch.zeros_like(b)
    s.setup(A)
    st = s.solve(b, x, zero_initial_guess=True)
    rel = float(ops.nrm2(ops.residual(A, x, b)) / ops.nrm2(b))
    assert st.converged and rel < 1e-7
    assert st.iterations <= 30


def test_chebyshev_as_amg_smoother():
    """Chebyshev used as the per-level smoother (the sweep() entry must
    re-seed the recurrence; regression for a missing solve_init)."""
    st, rel = _solve_classical(_classical_cfg(smoother="CHEBYSHEV"), n=8)
    assert st.converged and rel < 1e-7, f"{st}, rel={rel}"


def test_aggregation_error_scaling():
    """error_scaling 2/3 scale the prolongated correction by the
    residual/energy-minimizing lambda (reference
    src/aggregation/aggregation_amg_level.cu:700-825): both must converge
    and beat the unscaled V-cycle on Poisson; reuse_scale caches lambda."""
    def run(es, reuse=0):
        cfg = AMGConfig.from_dict({
            "config_version": 2,
            "solver": {
                "solver": "AMG", "algorithm": "AGGREGATION",
                "selector": "SIZE_2", "smoother": "BLOCK_JACOBI",
                "relaxation_factor": 0.75, "presweeps": 2, "postsweeps": 2,
                "coarsest_sweeps": 2, "min_coarse_rows": 8, "cycle": "V",
                "max_iters": 100, "monitor_residual": 1,
                "convergence": "RELATIVE_INI", "tolerance": 1e-8,
                "error_scaling": es, "reuse_scale": reuse,
            },
        })
        s = create_solver(cfg.root_scope(), resources=Resources("cpu"))
        A = poisson_3d(12, 12, 12)
        b = torch.ones(A.n_rows, dtype=torch.float64)
        x = torch.zeros_like(b)
        s.setup(A)
        st = s.solve(b, x, zero_initial_guess=True)
        rel = float(ops.nrm2(ops.residual(A, x, b)) / ops.nrm2(b))
        return st, rel

    st0, rel0 = run(0)
    assert st0.converged and rel0 < 1e-7
    for es in (2, 3):
        st, rel = run(es)
        assert st.converged and rel < 1e-7, (es, st, rel)
        assert st.iterations < st0.iterations, (es, st.iterations,
                                                st0.iterations)
    # cached-lambda path (reuse_scale > 0) still converges
    st, rel = run(3, reuse=2)
    assert st.converged and rel < 1e-7


def test_w_cycle_deep_hierarchy_converges():
    """Regression: W-cycles on a 6+-level classical D2 hierarchy diverged
    because a Galerkin coarse operator reached lam_max(D^-1 A) ~ 4 where
    the fixed 0.9 Jacobi damping amplifies error; the spectral safeguard in
    BLOCK_JACOBI clamps the effective relaxation. (V masked the
    instability, W's 2^depth coarse visits compounded it.)"""
    cfg = AMGConfig.from_dict({
        "config_version": 2,
        "solver": {"solver": "AMG", "interpolator": "D2", "cycle": "W",
                   "presweeps": 1, "postsweeps": 1, "max_iters": 60,
                   "monitor_residual": 1, "convergence": "RELATIVE_INI",
                   "tolerance": 1e-6},
    })
    s = create_solver(cfg.root_scope(), resources=Resources("cpu"))
    A = poisson_3d(40, 40, 40)   # deep enough to include the bad level
    b = torch.ones(A.n_rows, dtype=torch.float64)
    x = torch.zeros_like(b)
    s.setup(A)
    st = s.solve(b, x, zero_initial_guess=True)
    assert st.converged and st.iterations <= 40, st


def test_multipass_preserves_constants():
    """Regression: multipass interpolation lumped weak entries into the
    denominator instead of the reference alfa = -sum_N/(sum_C*diag)
    normalization (multipass.cu:1127-1191), so P rows summed to ~0.14 and
    aggressive hierarchies lost the constant — P row sums must be ~1."""
    import numpy as np
    import scipy.sparse as sp
    from amgx_amd.amg.classical import (SELECTOR_REGISTRY, STRENGTH_REGISTRY,
                                        INTERP_REGISTRY,
                                        _interp_multipass_device)
    from amgx_amd.config import ConfigScope
    sc = ConfigScope(None, {"strength_threshold": 0.25})
    A = poisson_3d(16, 16, 16)
    S = STRENGTH_REGISTRY["AHAT"](A, sc)
    cf, nc = SELECTOR_REGISTRY["AGGRESSIVE_PMIS"](A, S, sc)
    P = INTERP_REGISTRY["MULTIPASS"](A, S, cf, nc, sc)
    p = sp.csr_matrix((P.values.numpy().ravel(), P.col_indices.numpy(),
                       P.row_offsets.numpy()), shape=(P.n_rows, P.n_cols))
    rs = np.asarray(p.sum(axis=1)).ravel()
    interp_rows = np.diff(p.indptr) > 0
    assert np.median(rs[interp_rows]) > 0.95
    # device formulation (on CPU tensors) matches the host result exactly
    Pd = _interp_multipass_device(A, S, cf, nc, sc)
    pd_ = sp.csr_matrix((Pd.values.numpy().ravel(), Pd.col_indices.numpy(),
                         Pd.row_offsets.numpy()), shape=(Pd.n_rows, Pd.n_cols))
    assert abs(p - pd_).max() < 1e-14


def test_aggressive_hmis_keeps_isolated_c():
    """Regression: HMIS C sets are 2-ring independent, so the aggressive
    second pass sees an edgeless C-C graph; isolated C nodes were dropped
    (nc=0, no coarsening at all). They must stay coarse."""
    from amgx_amd.amg.classical import SELECTOR_REGISTRY, STRENGTH_REGISTRY
    from amgx_amd.config import ConfigScope
    sc = ConfigScope(None, {"strength_threshold": 0.25})
    A = poisson_3d(12, 12, 12)
    S = STRENGTH_REGISTRY["AHAT"](A, sc)
    cf, nc = SELECTOR_REGISTRY["AGGRESSIVE_HMIS"](A, S, sc)
    assert nc > 0
    assert 5.0 < A.n_rows / nc < 40.0   # aggressive-range coarsening


def test_bench_hierarchy_quality_bounds():
    """End-to-end hierarchy-quality guard on the flagship bench config
    (VERDICT r01 weak #3): healthy SIZE_2 coarsening must give per-level
    ratio <= 0.75, operator complexity < 3 and bounded depth at 40^3 —
    the round-1 tie-break defect showed up exactly as 0.83/level and ~6x
    operator complexity."""
    import torch

    from amgx_amd import AMGConfig, create_solver
    from amgx_amd.problems import poisson_3d
    from amgx_amd.resources import Resources
    A = poisson_3d(40, 40, 40)
    cfg = AMGConfig.from_dict(FGMRES_AGG)
    s = create_solver(cfg.root_scope(), resources=Resources("cpu"))
    s.setup(A)
    h = s.precond.hierarchy
    rows = [l.A.n_rows for l in h.levels]
    nnz = [l.A.nnz for l in h.levels]
    # pairwise matching halves each level (allow slack on small/coarse tails)
    for i in range(len(rows) - 1):
        if rows[i] > 500:
            ratio = rows[i + 1] / rows[i]
            assert ratio <= 0.75, (i, rows)
    # unsmoothed pairwise aggregation densifies (~6.9 -> ~11 nnz/row by
    # level 2), so healthy operator complexity is ~2.5; the round-1 defect
    # read ~6
    assert sum(nnz) / nnz[0] < 3.0, [f"{x:.3g}" for x in nnz]
    assert sum(rows) / rows[0] < 2.2, rows
    assert len(rows) <= 20, rows
    stats = h.grid_stats()
    assert "Operator Complexity" in stats


def test_square_graph_coloring_path():
    """The SpGEMM-squared-graph distance-2 path (the device-resident
    branch of the *_2RING schemes) yields valid distance-2 colorings —
    exercised here on CPU through the identical code: pattern square via
    backend spgemm, then ops.color_matrix (min_max_2ring.cu role)."""
    import torch

    from amgx_amd import ops
    from amgx_amd.amg.coloring import MatrixColoring, _square_graph_matrix
    from amgx_amd.matrix import CSRMatrix
    from amgx_amd.problems import poisson_3d
    A = poisson_3d(6, 5, 4)
    colors, num = ops.color_matrix(_square_graph_matrix(A))
    col = MatrixColoring(torch.as_tensor(colors), num)
    assert col.validate(A, level=2)
    # halo-column filter branch: squared structure must ignore halo columns
    ro, ci, v = A.row_offsets, A.col_indices, A.values
    ro2 = ro.clone()
    ro2[1:] += 1
    k = int(ro[1])
    ci2 = torch.cat([ci[:k], torch.tensor([A.n_rows + 3],
                                          dtype=torch.int32), ci[k:]])
    v2 = torch.cat([v[:k], torch.tensor([0.5], dtype=v.dtype), v[k:]])
    Ah = CSRMatrix(ro2, ci2, v2, n_cols=A.n_rows + 5)
    A2h = _square_graph_matrix(Ah)
    assert A2h.n_cols == A.n_rows
    c2, n2 = ops.color_matrix(A2h)
    assert MatrixColoring(torch.as_tensor(c2), n2).validate(A, level=2)


def test_nested_amg_equivalence():
    """5-level AMG == 3-level AMG whose coarse solver is another 3-level
    AMG (5 levels total) — bitwise-identical iterates (reference
    src/tests/nested_amg_equivalence.cu, tol 1e-10; deterministic SIZE_2
    matching makes the sub-hierarchies identical here)."""
    import copy

    import torch

    from amgx_amd import AMGConfig, create_solver
    from amgx_amd.problems import poisson_3d
    from amgx_amd.resources import Resources

    def run(cfg_dict):
        A = poisson_3d(12, 12, 12)
        cfg = AMGConfig.from_dict(cfg_dict)
        s = create_solver(cfg.root_scope(), resources=Resources("cpu"))
        b = torch.ones(A.n_rows, dtype=torch.float64)
        x = torch.zeros_like(b)
        s.setup(A)
        s.solve(b, x, zero_initial_guess=True)
        return x

    flat = {"config_version": 2, "solver": {
        "algorithm": "AGGREGATION", "solver": "AMG",
        "smoother": "BLOCK_JACOBI", "coarse_solver": "NOSOLVER",
        "presweeps": 1, "postsweeps": 1, "selector": "SIZE_2",
        "coarsest_sweeps": 1, "max_iters": 20, "monitor_residual": 1,
        "max_levels": 5, "tolerance": 1e-6, "norm": "L1", "cycle": "V",
        "min_coarse_rows": 2}}
    nested = copy.deepcopy(flat)
    nested["solver"]["max_levels"] = 3
    nested["solver"]["coarse_solver"] = {
        "solver": "AMG", "algorithm": "AGGREGATION",
        "smoother": "BLOCK_JACOBI", "presweeps": 1, "postsweeps": 1,
        "selector": "SIZE_2", "max_iters": 1, "coarse_solver": "NOSOLVER",
        "scope": "lower", "max_levels": 3, "coarsest_sweeps": 1,
        "cycle": "V", "min_coarse_rows": 2}
    x1 = run(flat)
    x2 = run(nested)
    assert torch.equal(x1, x2), (x1 - x2).abs().max()


def test_classical_block_matrix_clean_error():
    """CLASSICAL AMG is scalar-only, like the reference (classical_amg_
    level.cu computeAOperator_1x1): block matrices get a clear error, not
    a broadcast failure deep in setup."""
    import pytest
    import torch  # noqa: F401

    from amgx_amd import AMGConfig, create_solver
    from amgx_amd.problems import block_laplacian
    from amgx_amd.resources import Resources
    A = block_laplacian(8, 8, block_dim=4)
    cfg = AMGConfig.from_dict({"solver": {
        "solver": "AMG", "algorithm": "CLASSICAL",
        "smoother": "BLOCK_JACOBI", "max_iters": 1}})
    s = create_solver(cfg.root_scope(), resources=Resources("cpu"))
    with pytest.raises(ValueError, match="scalar"):
        s.setup(A)


def test_classical_dummy_selector():
    """DUMMY classical selector: alternating C/F (reference
    dummy_selector.cu:26) still yields a convergent hierarchy."""
    import torch

    from amgx_amd import AMGConfig, create_solver, ops
    from amgx_amd.problems import poisson_2d
    from amgx_amd.resources import Resources
    A = poisson_2d(16, 16)
    cfg = AMGConfig.from_dict({"solver": {
        "preconditioner": {
            "solver": "AMG", "algorithm": "CLASSICAL", "selector": "DUMMY",
            "smoother": {"solver": "MULTICOLOR_GS", "symmetric_GS": 1,
                         "max_iters": 1},
            "presweeps": 1, "postsweeps": 1, "max_iters": 1,
            "min_coarse_rows": 8},
        "solver": "PCG", "max_iters": 200, "monitor_residual": 1,
        "convergence": "RELATIVE_INI", "tolerance": 1e-8}})
    s = create_solver(cfg.root_scope(), resources=Resources("cpu"))
    b = torch.ones(A.n_rows, dtype=torch.float64)
    x = torch.zeros_like(b)
    s.setup(A)
    st = s.solve(b, x, zero_initial_guess=True)
    rel = float(ops.nrm2(ops.residual(A, x, b)) / ops.nrm2(b))
    assert st.converged and rel < 1e-7
