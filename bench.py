#!/usr/bin/env python3
"""Driver benchmark contract: AMG setup+solve on 3D 7-pt Poisson fp64 with
FGMRES + aggregation-AMG (the reference FGMRES_AGGREGATION.json config;
BASELINE.json metric "AMG setup+solve sec & iters, 3D Poisson 256^3 fp64,
FGMRES+AGG, 1/2/4/8 MI355X").

Weak scaling: every rank owns a 256^3 subdomain (z-slab of a global
256 x 256 x 256*N cube). One step = one full setup + solve-to-convergence
(tolerance 1e-6, RELATIVE_INI) from scratch. Rank 0 prints ONE JSON line.

    python bench.py --gpus N --steps K --warmup W [--size 256] [--config ...]
"""

import argparse
import json
import os
import sys
import time

import numpy as np
import torch

sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))

from amgx_amd import AMGConfig, create_solver  # noqa: E402
from amgx_amd.resources import Resources  # noqa: E402

FGMRES_AGG = {
    "config_version": 2,
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
    },
}


# Secondary measurable configs (BASELINE.json configs #3/#4/#5 shapes);
# the driver contract always uses the default fgmres_agg.
# symmetric_GS: ascending+descending color sweeps make the smoother (and
# with equal pre/post sweeps, the whole V-cycle) symmetric — PCG needs an
# SPD preconditioner (plain one-direction multicolor GS is not adjoint to
# itself, which stalls PCG at scale)
PCG_CLASSICAL = {
    "config_version": 2,
    "solver": {
        "preconditioner": {
            "algorithm": "CLASSICAL", "solver": "AMG",
            "interpolator": "D2",      # PMIS needs distance-2 interpolation
            "interp_truncation_factor": 0.25,   # keep the D2 operator sparse
            "smoother": "MULTICOLOR_GS", "symmetric_GS": 1,
            "presweeps": 1, "postsweeps": 1,
            "max_iters": 1, "min_coarse_rows": 32, "scope": "amg",
            "cycle": "V"},
        "solver": "PCG", "max_iters": 100, "monitor_residual": 1,
        "convergence": "RELATIVE_INI", "tolerance": 1e-6, "norm": "L2"},
}
FGMRES_ILU0 = {
    "config_version": 2,
    "solver": {
        "preconditioner": {"solver": "MULTICOLOR_ILU", "max_iters": 1,
                           "scope": "ilu"},
        "solver": "FGMRES", "max_iters": 500, "gmres_n_restart": 30,
        "monitor_residual": 1, "convergence": "RELATIVE_INI",
        "tolerance": 1e-6},
}
FGMRES_BLOCK4_DILU = {
    "config_version": 2,
    "solver": {
        "preconditioner": {"solver": "MULTICOLOR_DILU", "max_iters": 2,
                           "relaxation_factor": 1.0, "scope": "dilu"},
        "solver": "FGMRES", "max_iters": 200, "gmres_n_restart": 30,
        "monitor_residual": 1, "convergence": "RELATIVE_INI",
        "tolerance": 1e-6},
}
CONFIGS = {
    "fgmres_agg": (FGMRES_AGG, "poisson"),
    "classical_pcg": (PCG_CLASSICAL, "poisson"),
    "ilu0": (FGMRES_ILU0, "unstructured"),
    "block4_dilu": (FGMRES_BLOCK4_DILU, "block4"),
}


def build_local_matrix(size, rank, world, device, problem="poisson"):
    """Rank-local z-slab of the global size x size x (size*world) 7-pt
    Poisson (or a block-4 coupled system for block4_dilu). For world=1 this
    is exactly the 256^3 single-GPU config."""
    if problem == "block4":
        import numpy as np

        from amgx_amd.problems import block_laplacian
        side = max(int(round(size ** 1.5 / world ** 0.5)), 8)
        Afull = block_laplacian(side, side * world, block_dim=4, seed=9)
        # BASELINE config #4 is fp32/fp64 MIXED: fp32 matrix storage
        # (half the SpMV bytes), fp64 vectors end to end
        from amgx_amd.matrix import CSRMatrix
        Afull = CSRMatrix(Afull.row_offsets, Afull.col_indices,
                          Afull.values.to(torch.float32),
                          n_cols=Afull.n_cols, block_dim=4)
        if world == 1:
            return Afull.to(device), None
        from amgx_amd.distributed.manager import DistributedManager
        n = Afull.n_rows
        per = n // world
        lo = rank * per
        hi = n if rank == world - 1 else lo + per
        ro = Afull.row_offsets.numpy().astype(np.int64)
        s0, s1 = ro[lo], ro[hi]
        A = DistributedManager.upload_global_csr(
            ro[lo:hi + 1] - s0, Afull.col_indices.numpy()[s0:s1],
            Afull.values.numpy().reshape(Afull.nnz, 16)[s0:s1],
            hi - lo, lo, n, device=device, block_dim=4,
            dtype=torch.float32)
        return A, A.manager
    if problem == "unstructured":
        # scrambled Poisson: random node relabeling destroys the banded
        # locality (offline SuiteSparse stand-in, BASELINE config #5) while
        # keeping the spectrum — irregular halo + gather patterns
        import numpy as np
        import scipy.sparse as sp

        from amgx_amd.matrix import CSRMatrix
        from amgx_amd.problems import poisson_3d
        n = size ** 3 * world
        Ah = poisson_3d(size, size, size * world).to_scipy()
        rng = np.random.RandomState(7)
        perm = rng.permutation(n)
        Pm = sp.csr_matrix((np.ones(n), (perm, np.arange(n))), shape=(n, n))
        As = (Pm @ Ah @ Pm.T).tocsr()
        As.sum_duplicates()
        As.sort_indices()
        if world == 1:
            return CSRMatrix.from_scipy(As).to(device), None
        from amgx_amd.distributed.manager import DistributedManager
        per = n // world
        lo = rank * per
        hi = n if rank == world - 1 else lo + per
        s0, s1 = As.indptr[lo], As.indptr[hi]
        A = DistributedManager.upload_global_csr(
            As.indptr[lo:hi + 1] - s0, As.indices[s0:s1], As.data[s0:s1],
            hi - lo, lo, n, device=device)
        return A, A.manager
    if world == 1:
        from amgx_amd.problems import poisson_3d
        return poisson_3d(size, size, size, device=device), None
    from amgx_amd.distributed.manager import DistributedManager
    from amgx_amd.problems import poisson_3d_local_device
    ro, cols, vals, row_start = poisson_3d_local_device(
        size, size, size, rank, world, device=device)
    n_local = int(ro.numel() - 1)
    A = DistributedManager.upload_global_csr(
        ro, cols, vals, n_local, row_start, size * size * size * world,
        device=device)
    return A, A.manager


def run_step(A, cfg, res, b):
    x = torch.zeros_like(b)
    solver = create_solver(cfg.root_scope(), resources=res)
    t0 = time.perf_counter()
    solver.setup(A)
    t1 = time.perf_counter()
    st = solver.solve(b, x, zero_initial_guess=True)
    t2 = time.perf_counter()
    return {
        "setup_s": t1 - t0,
        "solve_s": t2 - t1,
        "total_s": t2 - t0,
        "iterations": st.iterations,
        "converged": bool(st.converged),
        "residual": st.residuals[-1] / st.residuals[0] if st.residuals else None,
    }


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--gpus", type=int, default=1)
    ap.add_argument("--steps", type=int, default=3)
    ap.add_argument("--warmup", type=int, default=1)
    ap.add_argument("--size", type=int, default=None,
                    help="per-rank subdomain edge (size^3 rows per GPU); "
                         "defaults: 256 (fgmres_agg/classical_pcg/block4), "
                         "128 (ilu0 — the unstructured ILU(0) iteration "
                         "count grows with size)")
    ap.add_argument("--device", default=None)
    ap.add_argument("--config", default="fgmres_agg", choices=sorted(CONFIGS),
                    help="solver composition (driver contract: fgmres_agg)")
    args = ap.parse_args()

    world = int(os.environ.get("WORLD_SIZE", "1"))
    rank = int(os.environ.get("RANK", "0"))
    local_rank = int(os.environ.get("LOCAL_RANK", "0"))
    n_gpus = max(args.gpus, world)

    use_cuda = torch.cuda.is_available() if args.device is None \
        else args.device.startswith("cuda")
    device = args.device or (f"cuda:{local_rank}" if use_cuda else "cpu")
    distributed = world > 1
    if use_cuda:
        torch.cuda.set_device(local_rank)
    res = Resources(device, distributed=distributed)
    dist = None
    if distributed:
        import torch.distributed as tdist
        dist = tdist

    if args.size is None:
        args.size = 128 if args.config == "ilu0" else 256
    cfg_dict, problem = CONFIGS[args.config]
    cfg = AMGConfig.from_dict(cfg_dict)
    A, manager = build_local_matrix(args.size, rank, world, device, problem)
    n_local = A.n_rows * A.block_dim
    g = torch.Generator().manual_seed(1234 + rank)
    b_user = torch.rand(n_local, generator=g, dtype=torch.float64).to(device)
    # distributed solves run on halo-extended vectors in internal ordering
    b = manager.permute_in(b_user) if manager is not None else b_user

    def barrier_sync():
        if dist is not None:
            dist.barrier()
        if use_cuda:
            torch.cuda.synchronize()

    for _ in range(args.warmup):
        run_step(A, cfg, res, b)

    barrier_sync()
    t_start = time.perf_counter()
    stats = [run_step(A, cfg, res, b) for _ in range(args.steps)]
    barrier_sync()
    t_total = time.perf_counter() - t_start

    # max over ranks of the timed window
    if dist is not None:
        tt = torch.tensor([t_total], dtype=torch.float64,
                          device=device if res.is_cuda else "cpu")
        dist.all_reduce(tt, op=dist.ReduceOp.MAX)
        t_total = float(tt.item())

    sec_per_step = t_total / args.steps
    last = stats[-1]
    dof = n_local * n_gpus   # actual per-rank unknowns (exact for block/unstructured too)
    if rank == 0:
        line = {
            "metric": ("amg_setup_solve_seconds_poisson256_fgmres_agg"
                       if args.config == "fgmres_agg" else
                       f"amg_setup_solve_seconds_{args.config}"),
            "value": sec_per_step,
            "unit": "s",
            "n_gpus": n_gpus,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": sec_per_step * 1e3,
            "higher_is_better": False,
            "scaling": "weak",
            "vs_baseline": None,
            "dtype": "fp32/fp64 mixed" if args.config == "block4_dilu"
                     else "fp64",
            "data": "synthetic",
            "config": {
                "model": {
                    "fgmres_agg": "3D 7-pt Poisson, FGMRES + aggregation-AMG"
                                  " V-cycle (FGMRES_AGGREGATION.json), tol"
                                  " 1e-6 RELATIVE_INI",
                    "classical_pcg": "3D 7-pt Poisson, PCG + classical"
                                     " Ruge-Stueben AMG V-cycle, tol 1e-6",
                    "ilu0": "scrambled-Poisson unstructured Laplacian"
                            " (SuiteSparse stand-in), FGMRES +"
                            " multicolor ILU(0), tol 1e-6",
                    "block4_dilu": "block-4 coupled system, FGMRES +"
                                   " multicolor DILU, tol 1e-6",
                }[args.config],
                "global_batch": dof,
                "seq_len": args.size,
                "parallelism": f"dd{n_gpus}",
                "rows_per_gpu": n_local,
                "iterations": last["iterations"],
                "converged": last["converged"],
                "setup_s": last["setup_s"],
                "solve_s": last["solve_s"],
                "final_rel_residual": last["residual"],
            },
        }
        print(json.dumps(line))
    if dist is not None:
        dist.destroy_process_group()


if __name__ == "__main__":
    main()
