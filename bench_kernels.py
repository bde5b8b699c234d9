#!/usr/bin/env python3
"""Per-kernel microbenchmarks on the current device (MI355X when available).

Times the solve-path kernels (csrmv, fused residual, DILU sweeps, color-GS,
reductions, axpy) and the setup kernels (coloring, matching, Galerkin,
SpGEMM, transpose) on a 3D Poisson problem, reporting ms and achieved GB/s
against the analytic bytes-moved model. One JSON line per kernel to stdout;
human table to stderr.

    python bench_kernels.py [--size 192] [--iters 20]
"""

import argparse
import json
import sys
import time

import torch


def bench(fn, iters, sync):
    fn()                       # warmup
    sync()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    sync()
    return (time.perf_counter() - t0) / iters


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--size", type=int, default=192)
    ap.add_argument("--iters", type=int, default=20)
    args = ap.parse_args()
    sys.path.insert(0, ".")
    from amgx_amd import ops
    from amgx_amd.amg.coloring import MatrixColoring
    from amgx_amd.config import ConfigScope
    from amgx_amd.problems import poisson_3d
    from amgx_amd.solvers.dilu import dilu_setup, dilu_solve

    dev = "cuda:0" if torch.cuda.is_available() else "cpu"
    sync = (lambda: torch.cuda.synchronize()) if dev != "cpu" else (lambda: 0)
    n3 = args.size
    A = poisson_3d(n3, n3, n3, device=dev)
    n, nnz = A.n_rows, A.nnz
    x = torch.rand(n, dtype=torch.float64, device=dev)
    y = torch.zeros_like(x)
    b = torch.ones_like(x)

    rows_bytes = nnz * 12 + (n + 1) * 4            # vals+cols+offsets
    spmv_bytes = rows_bytes + n * 8 * 2            # + x read, y write (approx)

    results = {}

    def rec(name, sec, bytes_moved=None):
        gbps = bytes_moved / sec / 1e9 if bytes_moved else None
        results[name] = {"ms": sec * 1e3, "GBps": gbps}
        print(json.dumps({"kernel": name, "ms": sec * 1e3, "GBps": gbps}))

    rec("csrmv", bench(lambda: ops.spmv(A, x, y), args.iters, sync),
        spmv_bytes)
    from amgx_amd.matrix import CSRMatrix
    A32 = CSRMatrix(A.row_offsets, A.col_indices,
                    A.values.to(torch.float32), n_cols=A.n_cols)
    rec("csrmv_mixed_f32A", bench(lambda: ops.spmv(A32, x, y),
                                  args.iters, sync),
        nnz * 8 + (n + 1) * 4 + n * 16)   # fp32 vals: half the value bytes
    r = torch.zeros_like(x)
    rec("residual_fused", bench(lambda: ops.residual(A, x, b, r),
                                args.iters, sync), spmv_bytes + n * 8)
    rec("dot", bench(lambda: ops.dot(x, y), args.iters, sync), n * 16)
    rec("nrm2", bench(lambda: ops.nrm2(x), args.iters, sync), n * 8)
    rec("axpy", bench(lambda: ops.axpy(y, x, 0.5), args.iters, sync), n * 24)

    col = MatrixColoring.create(A, ConfigScope(None, {}))
    dinv = ops.jacobi_dinv(A)
    rec("jacobi_sweep", bench(
        lambda: ops.jacobi_smooth(A, dinv, b, x, y, 0.9), args.iters, sync),
        spmv_bytes + n * 24)
    rec("gs_color_sweep", bench(
        lambda: ops.gs_sweep(A, dinv, b, x.clone(), col, 0.9),
        max(args.iters // 4, 2), sync), spmv_bytes + n * 24)
    einv = dilu_setup(A, col)
    rec("dilu_apply", bench(
        lambda: dilu_solve(A, einv, col, r, 0.75, y), args.iters, sync),
        2 * rows_bytes + n * 40)

    # block-4 kernels: MFMA wave path vs the scalar baseline
    from amgx_amd.problems import block_laplacian
    side = max(args.size, 64)          # side^2 block rows
    A4 = block_laplacian(side, side, block_dim=4, seed=3).to(dev)
    n4 = A4.n_rows
    x4 = torch.rand(n4 * 4, dtype=torch.float64, device=dev)
    y4 = torch.zeros_like(x4)
    b4_bytes = A4.nnz * (16 * 8 + 4) + (n4 + 1) * 4 + n4 * 4 * 16
    rec("bsrmv_b4_mfma", bench(lambda: ops.spmv(A4, x4, y4),
                               args.iters, sync), b4_bytes)
    if dev != "cpu":
        from amgx_amd import _core
        rec("bsrmv_b4_scalar_base", bench(
            lambda: _core.bsrmv_generic(A4.row_offsets, A4.col_indices,
                                        A4.values.reshape(-1), 4,
                                        x4, y4, 1.0, 0.0),
            args.iters, sync), b4_bytes)
    col4 = MatrixColoring.create(A4, ConfigScope(None, {}))
    e4 = dilu_setup(A4, col4)
    r4 = torch.rand_like(x4)
    rec("dilu_b4_apply_mfma", bench(
        lambda: dilu_solve(A4, e4, col4, r4, 1.0, y4), args.iters, sync),
        2 * A4.nnz * 16 * 8 + n4 * 4 * 40)
    rec("dilu_b4_setup_mfma", bench(
        lambda: dilu_setup(A4, col4), max(args.iters // 4, 2), sync))

    # setup-path
    rec("coloring_minmax", bench(
        lambda: ops.color_matrix(A), max(args.iters // 4, 2), sync))
    rec("size2_matching", bench(
        lambda: ops.size2_matching(A), max(args.iters // 8, 1), sync))
    agg, num = ops.size2_matching(A)
    agg = agg.to(A.row_offsets.device)
    rec("galerkin_agg", bench(
        lambda: ops.galerkin_aggregation(A, agg, num),
        max(args.iters // 8, 1), sync))
    rec("transpose", bench(lambda: ops.transpose(A),
                           max(args.iters // 8, 1), sync))
    print("kernel              ms         GB/s", file=sys.stderr)
    for k, v in results.items():
        gb = f"{v['GBps']:.0f}" if v["GBps"] else "-"
        print(f"{k:18s} {v['ms']:9.3f}  {gb}", file=sys.stderr)


if __name__ == "__main__":
    main()
