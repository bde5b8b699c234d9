#!/usr/bin/env python3
"""Single-process example through the AMGX_* API (role-equivalent of
reference examples/amgx_capi.c): read a MatrixMarket system (or generate a
Poisson problem), build a solver from a JSON config, solve, report.

    python examples/amgx_capi.py [-m matrix.mtx] [-c configs/FGMRES_AGGREGATION.json]
"""

import argparse
import os
import sys

sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))

from amgx_amd import capi as C  # noqa: E402

HERE = os.path.dirname(os.path.abspath(__file__))


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("-m", "--matrix", default=None,
                    help="MatrixMarket or AMGXAMDB binary system file")
    ap.add_argument("-c", "--config",
                    default=os.path.join(HERE, "..", "configs",
                                         "FGMRES_AGGREGATION.json"))
    ap.add_argument("--mode", default=None, help="dDDI (GPU) or hDDI (host)")
    args = ap.parse_args()

    import torch
    mode = args.mode or ("dDDI" if torch.cuda.is_available() else "hDDI")

    assert C.AMGX_initialize() == C.RC_OK
    C.AMGX_register_print_callback(lambda s: print(s, end=""))
    rc, major, minor = C.AMGX_get_api_version()
    print(f"AMGX-AMD api {major}.{minor}, mode {mode}")

    rc, cfg = C.AMGX_config_create_from_file(args.config)
    assert rc == C.RC_OK, C.AMGX_get_error_string(rc)
    rc, res = C.AMGX_resources_create_simple(cfg)
    rc, A = C.AMGX_matrix_create(res, mode)
    rc, b = C.AMGX_vector_create(res, mode)
    rc, x = C.AMGX_vector_create(res, mode)

    if args.matrix:
        assert C.AMGX_read_system(A, b, x, args.matrix) == C.RC_OK
    else:
        assert C.AMGX_generate_distributed_poisson_7pt(
            A, b, x, 1, 1, 32, 32, 32) == C.RC_OK
    rc, n, bx, by = C.AMGX_matrix_get_size(A)
    print(f"system: n={n} block {bx}x{by}")

    rc, solver = C.AMGX_solver_create(res, mode, cfg)
    assert C.AMGX_solver_setup(solver, A) == C.RC_OK
    C.AMGX_vector_set_zero(x)
    assert C.AMGX_solver_solve(solver, b, x) == C.RC_OK
    rc, status = C.AMGX_solver_get_status(solver)
    rc, iters = C.AMGX_solver_get_iterations_number(solver)
    rc, res_norm = C.AMGX_solver_calculate_residual_norm(solver, A, b, x)
    print(f"status={status} iterations={iters} |r|={res_norm:.3e}")

    for h in (solver, A, b, x):
        pass
    C.AMGX_solver_destroy(solver)
    C.AMGX_matrix_destroy(A)
    C.AMGX_vector_destroy(b)
    C.AMGX_vector_destroy(x)
    C.AMGX_resources_destroy(res)
    C.AMGX_config_destroy(cfg)
    C.AMGX_finalize()


if __name__ == "__main__":
    main()
