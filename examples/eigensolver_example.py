#!/usr/bin/env python3
"""Eigensolver example (role-equivalent of reference
examples/eigensolver.c): largest eigenpair of a Poisson operator by power
iteration / Lanczos through the AMGX_eigensolver_* API."""

import os
import sys

sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))

import torch  # noqa: E402

from amgx_amd import capi as C  # noqa: E402


def main():
    mode = "dDDI" if torch.cuda.is_available() else "hDDI"
    C.AMGX_initialize()
    rc, cfg = C.AMGX_config_create(
        "config_version=2, eig_solver=LANCZOS, eig_max_iters=200,"
        " eig_tolerance=1e-8, eig_eigenvector=1")
    rc, res = C.AMGX_resources_create_simple(cfg)
    rc, A = C.AMGX_matrix_create(res, mode)
    rc, b = C.AMGX_vector_create(res, mode)
    rc, x = C.AMGX_vector_create(res, mode)
    assert C.AMGX_generate_distributed_poisson_7pt(
        A, b, x, 1, 1, 16, 16, 16) == C.RC_OK

    rc, eig = C.AMGX_eigensolver_create(res, mode, cfg)
    assert C.AMGX_eigensolver_setup(eig, A) == C.RC_OK
    rc, v0 = C.AMGX_vector_create(res, mode)
    C.AMGX_vector_set_random(v0, A.A.n_rows)
    assert C.AMGX_eigensolver_solve(eig, v0) == C.RC_OK
    st = eig.status
    print(f"lambda_max ~= {st.eigenvalues[-1]:.8f} "
          f"({st.iterations} iterations, converged={st.converged})")
    C.AMGX_eigensolver_destroy(eig)
    C.AMGX_finalize()


if __name__ == "__main__":
    main()
