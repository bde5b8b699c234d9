import json

from amgx_amd.config import AMGConfig, write_parameters_description


FGMRES_AGG = {
    "config_version": 2,
    "solver": {
        "preconditioner": {
            "algorithm": "AGGREGATION",
            "solver": "AMG",
            "smoother": "MULTICOLOR_DILU",
            "presweeps": 0,
            "selector": "SIZE_2",
            "coarse_solver": "DENSE_LU_SOLVER",
            "max_iters": 1,
            "postsweeps": 3,
            "min_coarse_rows": 32,
            "relaxation_factor": 0.75,
            "scope": "amg",
            "max_levels": 50,
            "cycle": "V",
        },
        "use_scalar_norm": 1,
        "solver": "FGMRES",
        "max_iters": 100,
        "monitor_residual": 1,
        "gmres_n_restart": 10,
        "convergence": "RELATIVE_INI",
        "scope": "main",
        "tolerance": 1e-06,
        "norm": "L2",
    },
}


def test_json_roundtrip():
    cfg = AMGConfig.from_dict(FGMRES_AGG)
    root = cfg.root_scope()
    assert root.get("solver") == "FGMRES"
    assert root.get("tolerance") == 1e-6
    assert root.get("gmres_n_restart") == 10
    name, sub = root.sub_solver("preconditioner")
    assert name == "AMG"
    assert sub.get("algorithm") == "AGGREGATION"
    assert sub.get("postsweeps") == 3
    assert sub.get("relaxation_factor") == 0.75
    sm_name, sm = sub.sub_solver("smoother")
    assert sm_name == "MULTICOLOR_DILU"
    # registry default fallback
    assert root.get("presweeps") == 1
    assert sub.get("presweeps") == 0


def test_parse_json_string():
    cfg = AMGConfig.parse(json.dumps(FGMRES_AGG))
    assert cfg.root_scope().get("solver") == "FGMRES"


def test_flat_string():
    cfg = AMGConfig.parse(
        "config_version=2, solver(main)=PCG, main:max_iters=42, "
        "main:preconditioner(amg)=AMG, amg:presweeps=2")
    root = cfg.root_scope()
    # main scope node holds the solver params
    main = root.node["solver"]
    assert main["solver"] == "PCG"
    assert main["max_iters"] == 42
    assert main["preconditioner"]["solver"] == "AMG"
    assert main["preconditioner"]["presweeps"] == 2


def test_registry_dump():
    d = json.loads(write_parameters_description())
    assert "tolerance" in d and d["strength_threshold"]["default"] == 0.25


def test_all_shipped_configs_solve():
    """Every JSON config in configs/ drives a working solve on a small
    Poisson system (reference ships its configs/ as the contract)."""
    import glob
    import os

    import torch

    from amgx_amd import AMGConfig, create_solver, ops
    from amgx_amd.problems import poisson_3d
    from amgx_amd.resources import Resources
    here = os.path.join(os.path.dirname(__file__), "..", "configs")
    files = sorted(glob.glob(os.path.join(here, "*.json")))
    assert len(files) >= 12
    for f in files:
        cfg = AMGConfig.from_file(f)
        s = create_solver(cfg.root_scope(), resources=Resources("cpu"))
        A = poisson_3d(8, 8, 8)
        b = torch.ones(A.n_rows, dtype=torch.float64)
        x = torch.zeros_like(b)
        s.setup(A)
        st = s.solve(b, x, zero_initial_guess=True)
        rel = ops.nrm2(ops.residual(A, x, b)) / ops.nrm2(b)
        assert st.converged and rel < 1e-4, (os.path.basename(f), st, rel)


def test_unknown_flat_parameter_rejected():
    """parseParameterString rejects unregistered names (reference
    src/amg_config.cu 'Variable not registered')."""
    import pytest

    from amgx_amd.config import AMGConfig
    with pytest.raises(KeyError):
        AMGConfig.parse("not_a_real_parameter=3")
    AMGConfig.parse("max_iters=10")   # registered names parse
