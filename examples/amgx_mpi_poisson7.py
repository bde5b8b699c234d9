#!/usr/bin/env python3
"""Multi-GPU distributed example (role-equivalent of reference
examples/amgx_mpi_poisson7.c): every rank owns a z-slab of a 3D 7-point
Poisson problem; solve with classical AMG + PCG over RCCL/xGMI (gloo on
host).

    python -m torch.distributed.run --nnodes=1 --nproc-per-node 2 \
        --master-addr 127.0.0.1 examples/amgx_mpi_poisson7.py -n 64
"""

import argparse
import os
import sys

sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))

import torch  # noqa: E402
import torch.distributed as dist  # noqa: E402

from amgx_amd import capi as C  # noqa: E402

HERE = os.path.dirname(os.path.abspath(__file__))


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("-n", type=int, default=32, help="per-rank cube edge")
    ap.add_argument("-c", "--config",
                    default=os.path.join(HERE, "..", "configs",
                                         "PCG_CLASSICAL_V_JACOBI.json"))
    args = ap.parse_args()

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
    rc, cfg = C.AMGX_config_create_from_file(args.config)
    rc, res = C.AMGX_resources_create(cfg, comm=None, device_num=local_rank)
    rc, A = C.AMGX_matrix_create(res, mode)
    rc, b = C.AMGX_vector_create(res, mode)
    rc, x = C.AMGX_vector_create(res, mode)

    n = args.n
    assert C.AMGX_generate_distributed_poisson_7pt(
        A, b, x, 1, 1, n, n, n) == C.RC_OK

    rc, solver = C.AMGX_solver_create(res, mode, cfg)
    assert C.AMGX_solver_setup(solver, A) == C.RC_OK
    assert C.AMGX_solver_solve(solver, b, x) == C.RC_OK
    rc, iters = C.AMGX_solver_get_iterations_number(solver)
    rc, nrm = C.AMGX_solver_calculate_residual_norm(solver, A, b, x)
    if rank == 0:
        print(f"{world} ranks x {n}^3 rows: {iters} iterations, "
              f"|r| = {nrm:.3e}")

    C.AMGX_solver_destroy(solver)
    C.AMGX_finalize()
    if world > 1:
        dist.destroy_process_group()


if __name__ == "__main__":
    main()
