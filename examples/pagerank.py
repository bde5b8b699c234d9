#!/usr/bin/env python3
"""PageRank via the eigensolver C API (role-equivalent of the reference
eigen examples with configs/eigen/PAGERANK): power iteration on the Google
matrix of a small web graph. Works serial or distributed (torchrun)."""

import os
import sys

sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))

import numpy as np  # noqa: E402
import scipy.sparse as sp  # noqa: E402
import torch  # noqa: E402

from amgx_amd import capi as C  # noqa: E402

HERE = os.path.dirname(os.path.abspath(__file__))


def main():
    n = 500
    rng = np.random.RandomState(42)
    M = sp.random(n, n, density=0.02, random_state=rng, format="csr")
    M.setdiag(0)
    M.eliminate_zeros()
    M.data[:] = 1.0

    C.AMGX_initialize()
    rc, cfg = C.AMGX_config_create_from_file(
        os.path.join(HERE, "..", "configs", "eigen", "PAGERANK"))
    rc, res = C.AMGX_resources_create_simple(cfg)
    mode = "dDDI" if torch.cuda.is_available() else "hDDI"
    rc, A = C.AMGX_matrix_create(res, mode)
    C.AMGX_matrix_upload_all(A, n, M.nnz, 1, 1, M.indptr, M.indices, M.data,
                             None)
    rc, es = C.AMGX_eigensolver_create(res, mode, cfg)
    assert C.AMGX_eigensolver_pagerank_setup(es, A) == C.RC_OK
    rc, x = C.AMGX_vector_create(res, mode)
    assert C.AMGX_eigensolver_solve(es, x) == C.RC_OK
    st = es.solver.status
    pr = st.eigenvector / st.eigenvector.sum()
    top = torch.topk(pr, 5)
    print(f"converged in {st.iterations} iterations; "
          f"lambda = {st.eigenvalues[0]:.6f}")
    print("top-5 pages:", [(int(i), float(v))
                           for i, v in zip(top.indices, top.values)])
    C.AMGX_finalize()


if __name__ == "__main__":
    main()
