#!/usr/bin/env python3
"""Distributed upload example (role-equivalent of reference
examples/amgx_mpi_capi.c): each rank assembles its own rows of a global
system with GLOBAL column indices and uploads via
AMGX_matrix_upload_all_global — the general ingestion path for external
applications (CFD codes etc.), as opposed to the built-in Poisson generator.

    python -m torch.distributed.run --nnodes=1 --nproc-per-node 2 \
        --master-addr 127.0.0.1 examples/amgx_mpi_capi.py
"""

import os
import sys

sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))

import numpy as np  # noqa: E402
import torch  # noqa: E402
import torch.distributed as dist  # noqa: E402

from amgx_amd import capi as C  # noqa: E402


def assemble_my_rows(n_global, lo, hi):
    """1D Laplacian rows [lo, hi) with global column ids."""
    rows = []
    cols = []
    vals = []
    ro = [0]
    for i in range(lo, hi):
        if i > 0:
            cols.append(i - 1)
            vals.append(-1.0)
        cols.append(i)
        vals.append(2.0)
        if i < n_global - 1:
            cols.append(i + 1)
            vals.append(-1.0)
        ro.append(len(cols))
    return (np.asarray(ro), np.asarray(cols, dtype=np.int64),
            np.asarray(vals))


def main():
    rank = int(os.environ.get("RANK", "0"))
    world = int(os.environ.get("WORLD_SIZE", "1"))
    local_rank = int(os.environ.get("LOCAL_RANK", "0"))
    use_cuda = torch.cuda.is_available()
    if world > 1:
        dist.init_process_group("nccl" if use_cuda else "gloo",
                                rank=rank, world_size=world)
    if use_cuda:
        torch.cuda.set_device(local_rank)
    mode = "dDDI" if use_cuda else "hDDI"

    C.AMGX_initialize()
    rc, cfg = C.AMGX_config_create(
        "config_version=2, solver=PCG, preconditioner=BLOCK_JACOBI,"
        " max_iters=500, tolerance=1e-10, convergence=RELATIVE_INI,"
        " monitor_residual=1")
    rc, res = C.AMGX_resources_create(cfg, comm=None, device_num=local_rank)
    rc, A = C.AMGX_matrix_create(res, mode)
    rc, b = C.AMGX_vector_create(res, mode)
    rc, x = C.AMGX_vector_create(res, mode)

    n_global = 1000 * world
    per = n_global // world
    lo = rank * per
    hi = n_global if rank == world - 1 else lo + per
    ro, cols, vals = assemble_my_rows(n_global, lo, hi)
    assert C.AMGX_matrix_upload_all_global(
        A, n_global, hi - lo, len(cols), 1, 1, ro, cols, vals) == C.RC_OK
    C.AMGX_vector_bind(b, A)
    C.AMGX_vector_bind(x, A)
    C.AMGX_vector_upload(b, hi - lo, 1, np.ones(hi - lo))
    C.AMGX_vector_set_zero(x, hi - lo, 1)

    rc, solver = C.AMGX_solver_create(res, mode, cfg)
    assert C.AMGX_solver_setup(solver, A) == C.RC_OK
    assert C.AMGX_solver_solve(solver, b, x) == C.RC_OK
    rc, iters = C.AMGX_solver_get_iterations_number(solver)
    rc, nrm = C.AMGX_solver_calculate_residual_norm(solver, A, b, x)
    if rank == 0:
        print(f"{world} ranks, n={n_global}: {iters} iterations, "
              f"|r| = {nrm:.3e}")
    C.AMGX_finalize()
    if world > 1:
        dist.destroy_process_group()


if __name__ == "__main__":
    main()
