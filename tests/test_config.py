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
