"""AMGX_* API parity layer (reference include/amgx_c.h:150-603,
src/amgx_c.cu).

The reference exposes ~75 flat C entry points over opaque handles. This
module reproduces that surface 1:1 in Python (same names, same handle
lifecycle, same RC error-code discipline: every call returns RC_OK or an
error code, never raises across the boundary — reference AMGX_TRIES/
AMGX_CATCHES, src/amgx_c.cu). Outputs are returned after the RC
(Pythonic substitute for C out-pointers).

Mode strings follow the reference bit-packed modes (include/amgx_config.h:
79-120): e.g. "dDDI" = device memory space, double vector, double matrix,
int index; "hDDI" = host. Mixed precision "dDFI" = double vector x float
matrix.
"""

from __future__ import annotations

import threading
from typing import Optional

import numpy as np
import torch

from . import ops
from .config import AMGConfig, write_parameters_description
from .matrix import CSRMatrix
from .resources import Resources
from .solvers import create_solver

# ----------------------------------------------------------------- RC codes
RC_OK = 0
RC_BAD_PARAMETERS = 1
RC_UNKNOWN = 2
RC_NOT_SUPPORTED_TARGET = 3
RC_NOT_SUPPORTED_BLOCKSIZE = 4
RC_CUDA_FAILURE = 5
RC_IO_ERROR = 6
RC_BAD_MODE = 7
RC_CORE = 8
RC_PLUGIN = 9
RC_BAD_CONFIGURATION = 10
RC_NOT_IMPLEMENTED = 11
RC_LICENSE_NOT_FOUND = 12
RC_INTERNAL = 13

AMGX_DIST_PARTITION_VECTOR = 0
AMGX_DIST_PARTITION_OFFSETS = 1

_API_VERSION = (2, 4)

_lock = threading.RLock()
_initialized = False
_print_callback = None


def _amgx_try(fn):
    def wrapper(*args, **kwargs):
        global _print_callback
        with _lock:
            try:
                return fn(*args, **kwargs)
            except Exception as e:  # noqa: BLE001 - API boundary
                msg = f"AMGX error in {fn.__name__}: {e}\n"
                if _print_callback is not None:
                    try:
                        _print_callback(msg)
                    except Exception:
                        pass
                rc = RC_BAD_PARAMETERS if isinstance(e, (TypeError, ValueError,
                                                         KeyError, AssertionError)) \
                    else RC_UNKNOWN
                return rc
    wrapper.__name__ = fn.__name__
    return wrapper


def _parse_mode(mode: str):
    if not isinstance(mode, str) or len(mode) != 4:
        raise ValueError(f"bad mode {mode!r}")
    mem = {"d": "cuda", "h": "cpu"}[mode[0]]
    prec = {"D": torch.float64, "F": torch.float32,
            "Z": torch.complex128, "C": torch.complex64}  # Z/C: complex modes
    vec = prec[mode[1]]
    mat = prec[mode[2]]
    assert mode[3] == "I"
    return mem, vec, mat


# ----------------------------------------------------------------- handles
class _ConfigHandle:
    def __init__(self, cfg: AMGConfig):
        self.cfg = cfg


class _ResourcesHandle:
    def __init__(self, res: Resources, cfg=None):
        self.res = res
        # the resources-level config governs IO knobs (block_convert,
        # rhs_from_a) exactly like the reference (src/amgx_c.cu:5016-5019)
        self.cfg = cfg


class _DistributionHandle:
    def __init__(self):
        self.kind = AMGX_DIST_PARTITION_OFFSETS
        self.data = None


class _MatrixHandle:
    def __init__(self, res: Resources, mode: str):
        self.res = res
        self.mode = mode
        self.A: Optional[CSRMatrix] = None


class _VectorHandle:
    def __init__(self, res: Resources, mode: str):
        self.res = res
        self.mode = mode
        self.v: Optional[torch.Tensor] = None
        self.bound: Optional[_MatrixHandle] = None
        self.block_dim = 1
        self.n = 0


class _SolverHandle:
    def __init__(self, res: Resources, mode: str, cfg: AMGConfig):
        self.res = res
        self.mode = mode
        self.cfg = cfg
        self.solver = None
        self.status = None


# ----------------------------------------------------------------- lifecycle
@_amgx_try
def AMGX_initialize():
    global _initialized
    _initialized = True
    return RC_OK


@_amgx_try
def AMGX_initialize_plugins():
    return RC_OK


@_amgx_try
def AMGX_finalize():
    global _initialized
    _initialized = False
    return RC_OK


@_amgx_try
def AMGX_finalize_plugins():
    return RC_OK


def AMGX_abort(res, err):  # pragma: no cover - mirrors reference abort
    raise SystemExit(err)


@_amgx_try
def AMGX_get_api_version():
    return RC_OK, _API_VERSION[0], _API_VERSION[1]


@_amgx_try
def AMGX_get_build_info_strings():
    from . import __version__
    return RC_OK, f"amgx_amd {__version__}", "MI355X gfx950", "rocm"


@_amgx_try
def AMGX_install_signal_handler():
    import faulthandler
    faulthandler.enable()
    return RC_OK


@_amgx_try
def AMGX_reset_signal_handler():
    import faulthandler
    faulthandler.disable()
    return RC_OK


@_amgx_try
def AMGX_register_print_callback(cb):
    global _print_callback
    _print_callback = cb
    from . import output
    output.print_callback = cb
    return RC_OK


@_amgx_try
def AMGX_write_parameters_description(path: str):
    with open(path, "w") as f:
        f.write(write_parameters_description())
    return RC_OK


# ----------------------------------------------------------------- config
@_amgx_try
def AMGX_config_create(options: str):
    return RC_OK, _ConfigHandle(AMGConfig.parse(options))


@_amgx_try
def AMGX_config_create_from_file(path: str):
    return RC_OK, _ConfigHandle(AMGConfig.from_file(path))


@_amgx_try
def AMGX_config_add_parameters(cfg: _ConfigHandle, options: str):
    extra = AMGConfig.parse(options)
    node = cfg.cfg.tree
    node.update(extra.tree)
    return RC_OK


@_amgx_try
def AMGX_config_get_default_number_of_rings(cfg: _ConfigHandle):
    """Reference semantics (src/amgx_c.cu): aggregation-style compositions
    ask for 2 import rings, everything else 1."""
    def _has_agg(node):
        if not isinstance(node, dict):
            return False
        if node.get("algorithm") == "AGGREGATION":
            return True
        return any(_has_agg(v) for v in node.values())
    return RC_OK, 2 if _has_agg(cfg.cfg.tree) else 1


@_amgx_try
def AMGX_config_destroy(cfg: _ConfigHandle):
    cfg.cfg = None
    return RC_OK


# ----------------------------------------------------------------- resources
@_amgx_try
def AMGX_resources_create_simple(cfg: _ConfigHandle):
    return RC_OK, _ResourcesHandle(Resources(), cfg)


@_amgx_try
def AMGX_resources_create(cfg: _ConfigHandle, comm=None, device_num: int = 0,
                          devices=None):
    dev = f"cuda:{device_num}" if torch.cuda.is_available() else "cpu"
    distributed = comm is not None
    return RC_OK, _ResourcesHandle(Resources(dev, distributed=distributed),
                                   cfg)


@_amgx_try
def AMGX_resources_destroy(res: _ResourcesHandle):
    res.res = None
    return RC_OK


# ----------------------------------------------------------------- distribution
@_amgx_try
def AMGX_distribution_create(cfg: _ConfigHandle = None):
    return RC_OK, _DistributionHandle()


@_amgx_try
def AMGX_distribution_set_partition_data(dist: _DistributionHandle, kind,
                                         data):
    dist.kind = kind
    dist.data = np.asarray(data, dtype=np.int64)
    return RC_OK


@_amgx_try
def AMGX_distribution_destroy(dist: _DistributionHandle):
    dist.data = None
    return RC_OK


# ----------------------------------------------------------------- matrix
@_amgx_try
def AMGX_matrix_create(res: _ResourcesHandle, mode: str):
    _parse_mode(mode)
    h = _MatrixHandle(res.res, mode)
    h.res_cfg = getattr(res, "cfg", None)
    return RC_OK, h


@_amgx_try
def AMGX_matrix_destroy(m: _MatrixHandle):
    m.A = None
    return RC_OK


def _fold_external_diag(n, b, ro, ci, va, diag):
    """Merge the DIAG-property external diagonal into the CSR structure
    (reference block-DIA-CSR, include/matrix.h:24-26). The gfx950 kernels
    take one folded CSR — the external layout is an input format here."""
    deg = np.diff(ro)
    rows = np.repeat(np.arange(n, dtype=np.int64), deg)
    all_rows = np.concatenate([rows, np.arange(n, dtype=np.int64)])
    all_cols = np.concatenate([ci.astype(np.int64),
                               np.arange(n, dtype=np.int64)])
    vab = va.reshape(len(ci), -1)
    db = diag.reshape(n, -1)
    all_vals = np.concatenate([vab, db], axis=0)
    order = np.lexsort((all_cols, all_rows))
    ro2 = np.zeros(n + 1, dtype=np.int64)
    np.cumsum(np.bincount(all_rows, minlength=n), out=ro2[1:])
    cols2 = all_cols[order]
    vals2 = all_vals[order]
    # sum duplicates (a CSR that already stores an explicit diagonal)
    ncols = int(max(cols2.max() + 1 if cols2.size else 1, n))
    key = all_rows[order] * ncols + cols2
    uniq, first = np.unique(key, return_index=True)
    if uniq.size != key.size:
        sums = np.add.reduceat(vals2, first, axis=0)
        cols2 = cols2[first]
        vals2 = sums
        counts = np.bincount((uniq // ncols).astype(np.int64), minlength=n)
        ro2 = np.zeros(n + 1, dtype=np.int64)
        np.cumsum(counts, out=ro2[1:])
    return ro2, cols2, vals2


@_amgx_try
def AMGX_matrix_upload_all(m: _MatrixHandle, n, nnz, block_dimx, block_dimy,
                           row_ptrs, col_indices, data, diag_data=None):
    mem, _, matprec = _parse_mode(m.mode)
    assert block_dimx == block_dimy, "rectangular blocks unsupported"
    device = mem if mem == "cpu" else m.res.device
    b = int(block_dimx)
    ro_np = np.asarray(row_ptrs, dtype=np.int64)
    ci_np = np.asarray(col_indices, dtype=np.int64)
    va_np = np.asarray(data)
    if diag_data is not None:
        dg = np.asarray(diag_data)
        ro_np, ci_np, va_np = _fold_external_diag(int(n), b, ro_np, ci_np,
                                                  va_np, dg)
        m._upload_structure = (np.asarray(row_ptrs, dtype=np.int64),
                               np.asarray(col_indices, dtype=np.int64))
    ro = torch.as_tensor(ro_np, dtype=torch.int32)
    ci = torch.as_tensor(ci_np, dtype=torch.int32)
    va = torch.as_tensor(np.ascontiguousarray(va_np)).to(matprec)
    va = va.reshape(-1, b, b) if b > 1 else va.reshape(-1)
    m.A = CSRMatrix(ro.to(device), ci.to(device), va.to(device),
                    n_cols=n, block_dim=b)
    return RC_OK


@_amgx_try
def AMGX_matrix_upload_all_global(m: _MatrixHandle, n_global, n, nnz,
                                  block_dimx, block_dimy, row_ptrs,
                                  col_indices_global, data, diag_data=None,
                                  allocated_halo_depth=1, num_import_rings=1,
                                  partition_vector=None):
    """Distributed upload with GLOBAL column indices (reference
    src/amgx_c.cu:4615 / matrix_upload_distributed:1739)."""
    import torch.distributed as tdist
    from .distributed.manager import DistributedManager
    mem, _, matprec = _parse_mode(m.mode)
    assert block_dimx == block_dimy
    if not tdist.is_initialized():
        # single-process global upload (reference: 1-rank communicator) —
        # global columns ARE local columns, no halo structure needed
        assert int(n) == int(n_global), \
            "upload_all_global without torch.distributed needs n == n_global"
        from .matrix import CSRMatrix
        device = mem if mem == "cpu" else m.res.device
        ro_t = torch.as_tensor(np.asarray(row_ptrs)).to(torch.int32)
        ci_t = torch.as_tensor(
            np.asarray(col_indices_global)).to(torch.int32)
        vals = torch.as_tensor(np.asarray(data)).to(matprec)
        if block_dimx > 1:
            m.A = CSRMatrix.from_bsr(
                ro_t, ci_t, vals.reshape(nnz, block_dimx, block_dimx),
                n_cols=int(n)).to(device)
        else:
            m.A = CSRMatrix(ro_t, ci_t, vals, n_cols=int(n)).to(device)
        return RC_OK
    rank = tdist.get_rank()
    if partition_vector is not None:
        pv = np.asarray(partition_vector)
        row_start = int(np.sum([(pv[:] == r).sum() for r in range(rank)]))
        assert (np.sort(np.nonzero(pv == rank)[0])
                == np.arange(row_start, row_start + n)).all(), \
            "non-contiguous partition vectors unsupported"
    else:
        counts = [None] * tdist.get_world_size()
        tdist.all_gather_object(counts, int(n))
        row_start = int(sum(counts[:rank]))
    device = mem if mem == "cpu" else m.res.device
    vals = np.asarray(data)
    if block_dimx > 1:
        vals = vals.reshape(nnz, block_dimx * block_dimx)
    m.A = DistributedManager.upload_global_csr(
        np.asarray(row_ptrs), np.asarray(col_indices_global), vals,
        n, row_start, n_global, device=device, block_dim=int(block_dimx),
        dtype=matprec)
    return RC_OK


@_amgx_try
def AMGX_matrix_upload_distributed(m, n_global, n, nnz, block_dimx,
                                   block_dimy, row_ptrs, col_indices_global,
                                   data, diag_data, dist: _DistributionHandle):
    pv = None
    if dist is not None and dist.kind == AMGX_DIST_PARTITION_VECTOR:
        pv = dist.data
    return AMGX_matrix_upload_all_global(
        m, n_global, n, nnz, block_dimx, block_dimy, row_ptrs,
        col_indices_global, data, diag_data, partition_vector=pv)


@_amgx_try
def AMGX_matrix_replace_coefficients(m: _MatrixHandle, n, nnz, data,
                                     diag_data=None):
    _, _, matprec = _parse_mode(m.mode)
    b = m.A.block_dim
    if diag_data is not None:
        # re-fold through the ORIGINAL upload structure so positions match
        ro0, ci0 = m._upload_structure
        _, _, va_np = _fold_external_diag(int(n), b, ro0, ci0,
                                          np.asarray(data),
                                          np.asarray(diag_data))
        va = torch.as_tensor(np.ascontiguousarray(va_np)).to(matprec)
        va = va.reshape(-1, b, b) if b > 1 else va.reshape(-1)
        m.A.replace_coefficients(va)
        return RC_OK
    va = torch.as_tensor(np.asarray(data)).to(matprec)
    if b > 1:
        va = va.reshape(-1, b, b)
    m.A.replace_coefficients(va)
    return RC_OK


@_amgx_try
def AMGX_matrix_get_size(m: _MatrixHandle):
    return RC_OK, m.A.n_rows, m.A.block_dim, m.A.block_dim


@_amgx_try
def AMGX_matrix_download_all(m: _MatrixHandle):
    A = m.A
    return (RC_OK, A.row_offsets.cpu().numpy(), A.col_indices.cpu().numpy(),
            A.values.cpu().numpy(),
            A.diag.cpu().numpy() if A.diag is not None else None)


@_amgx_try
def AMGX_matrix_check_symmetry(m: _MatrixHandle):
    s = m.A.to_scipy()
    diff = abs(s - s.T)
    structurally = (s != 0).multiply((s.T != 0) != (s != 0)).nnz == 0
    sym = diff.nnz == 0 or diff.max() < 1e-12
    return RC_OK, bool(structurally or sym), bool(sym)


@_amgx_try
def AMGX_matrix_check_diag_dominant(m: _MatrixHandle):
    s = m.A.to_scipy().tocsr()
    d = np.abs(s.diagonal())
    off = np.asarray(abs(s).sum(axis=1)).ravel() - d
    return RC_OK, bool((d >= off).all())


# ----------------------------------------------------------------- vector
@_amgx_try
def AMGX_vector_create(res: _ResourcesHandle, mode: str):
    _parse_mode(mode)
    return RC_OK, _VectorHandle(res.res, mode)


@_amgx_try
def AMGX_vector_destroy(v: _VectorHandle):
    v.v = None
    return RC_OK


@_amgx_try
def AMGX_vector_bind(v: _VectorHandle, m: _MatrixHandle):
    """Attach to a (possibly distributed) matrix (reference
    AMGX_vector_bind: vector inherits the matrix's comm pattern)."""
    v.bound = m
    return RC_OK


def _vec_device(v: _VectorHandle):
    mem, vecprec, _ = _parse_mode(v.mode)
    dev = mem if mem == "cpu" else v.res.device
    return dev, vecprec


@_amgx_try
def AMGX_vector_upload(v: _VectorHandle, n, block_dim, data):
    dev, prec = _vec_device(v)
    t = torch.as_tensor(np.asarray(data)).to(prec).reshape(-1)
    v.n, v.block_dim = int(n), int(block_dim)
    A = v.bound.A if v.bound is not None else None
    mgr = getattr(A, "manager", None) if A is not None else None
    if mgr is not None:
        full = mgr.new_ext_vec(prec)
        full[:mgr.owned_size] = t.to(full.device)[
            (mgr.row_perm[:, None] * block_dim
             + torch.arange(block_dim, device=full.device)[None, :]).reshape(-1)
            if block_dim > 1 else mgr.row_perm]
        v.v = full
    else:
        v.v = t.to(dev)
    return RC_OK


@_amgx_try
def AMGX_vector_set_zero(v: _VectorHandle, n=None, block_dim=None):
    if v.v is None:
        dev, prec = _vec_device(v)
        A = v.bound.A if v.bound is not None else None
        mgr = getattr(A, "manager", None) if A is not None else None
        if mgr is not None:
            v.v = mgr.new_ext_vec(prec)
            v.n = mgr.n_local
        else:
            assert n is not None
            v.n = int(n)
            v.block_dim = int(block_dim or 1)
            v.v = torch.zeros(v.n * v.block_dim, dtype=prec, device=dev)
    else:
        v.v.zero_()
    return RC_OK


@_amgx_try
def AMGX_vector_set_random(v: _VectorHandle, n):
    dev, prec = _vec_device(v)
    v.n = int(n)
    v.v = torch.rand(v.n * v.block_dim, dtype=prec, device=dev)
    return RC_OK


@_amgx_try
def AMGX_vector_download(v: _VectorHandle):
    A = v.bound.A if v.bound is not None else None
    mgr = getattr(A, "manager", None) if A is not None else None
    if mgr is not None:
        return RC_OK, mgr.permute_out(v.v).cpu().numpy()
    return RC_OK, v.v.detach().cpu().numpy()


@_amgx_try
def AMGX_vector_get_size(v: _VectorHandle):
    return RC_OK, v.n, v.block_dim


# ----------------------------------------------------------------- solver
@_amgx_try
def AMGX_solver_create(res: _ResourcesHandle, mode: str, cfg: _ConfigHandle):
    _parse_mode(mode)
    tree = cfg.cfg.tree
    solver_node = tree.get("solver")
    wants_print = tree.get("print_config") or (
        isinstance(solver_node, dict) and solver_node.get("print_config"))
    if wants_print:   # reference print_config param
        import json as _json
        print(_json.dumps(tree, indent=2, default=str))
    return RC_OK, _SolverHandle(res.res, mode, cfg.cfg)


@_amgx_try
def AMGX_solver_destroy(s: _SolverHandle):
    s.solver = None
    return RC_OK


@_amgx_try
def AMGX_solver_setup(s: _SolverHandle, m: _MatrixHandle):
    mem, _, _ = _parse_mode(s.mode)
    res = s.res if mem != "cpu" else Resources("cpu")
    s.solver = create_solver(s.cfg.root_scope(), resources=res)
    if getattr(s, "print_callback", None) is not None:
        s.solver.print_cb = s.print_callback   # registered pre-setup
    s.solver.setup(m.A)
    s.matrix = m
    return RC_OK


@_amgx_try
def AMGX_solver_resetup(s: _SolverHandle, m: _MatrixHandle):
    s.solver.resetup(m.A)
    return RC_OK


@_amgx_try
def AMGX_solver_solve(s: _SolverHandle, rhs: _VectorHandle,
                      sol: _VectorHandle, zero_initial_guess=False):
    if sol.v is None or sol.v.numel() != rhs.v.numel():
        sol.v = torch.zeros_like(rhs.v)
        sol.n = rhs.n
        sol.block_dim = rhs.block_dim
    s.status = s.solver.solve(rhs.v, sol.v,
                              zero_initial_guess=zero_initial_guess)
    return RC_OK


@_amgx_try
def AMGX_solver_solve_with_0_initial_guess(s, rhs, sol):
    return AMGX_solver_solve(s, rhs, sol, zero_initial_guess=True)


@_amgx_try
def AMGX_solver_get_status(s: _SolverHandle):
    # reference AMGX_SOLVE_SUCCESS=0, FAILED=1, DIVERGED=2
    st = s.status
    code = 0 if st.converged else (2 if st.status == st.DIVERGED else 1)
    return RC_OK, code


@_amgx_try
def AMGX_solver_get_iterations_number(s: _SolverHandle):
    return RC_OK, s.status.iterations


@_amgx_try
def AMGX_solver_get_iteration_residual(s: _SolverHandle, it: int = -1,
                                       idx: int = 0):
    res = s.status.residuals
    return RC_OK, float(res[it] if res else float("nan"))


# ----------------------------------------------------------------- IO
@_amgx_try
def AMGX_read_system(m: _MatrixHandle, rhs: _VectorHandle,
                     sol: _VectorHandle, path: str):
    from .io.binary import is_binary_file, read_system_any
    from .io.matrix_market import read_system
    mem, vecprec, matprec = _parse_mode(m.mode)
    dev = mem if mem == "cpu" else m.res.device
    reader = read_system_any if is_binary_file(path) else read_system
    A, b, x0 = reader(path, device=dev, dtype=matprec)
    # resources-config IO knobs (reference src/amgx_c.cu:5008-5019):
    # block_convert=b regroups a scalar system into bxb block-CSR;
    # rhs_from_a=1 generates b = A*[1..1]^T when the file has no RHS
    cfg_h = getattr(m, "res_cfg", None)
    scope = cfg_h.cfg.root_scope() if cfg_h is not None else None
    bconv = int(scope.get("block_convert") or 0) if scope is not None else 0
    if bconv > 1 and A.block_dim == 1:
        A = _block_convert(A, bconv)
    m.A = A
    n = A.n_rows * A.block_dim
    if b is None and scope is not None \
            and int(scope.get("rhs_from_a") or 0) == 1:
        from . import ops as _ops
        ones = torch.ones(n, dtype=A.dtype, device=A.values.device)
        b = _ops.spmv(A, ones)
    if rhs is not None:
        rhs.v = b.to(vecprec) if b is not None else \
            torch.ones(n, dtype=vecprec, device=dev)
        rhs.n = n
    if sol is not None:
        sol.v = x0.to(vecprec) if x0 is not None else \
            torch.zeros(n, dtype=vecprec, device=dev)
        sol.n = n
    return RC_OK


def _block_convert(A, b: int):
    """Scalar CSR -> bxb block-CSR by grouping consecutive rows/cols
    (reference Ahc.convert, src/amgx_c.cu:1303-1317; src/matrix_io).
    n must be divisible by b; absent entries zero-fill their block."""
    from .matrix import CSRMatrix
    if A.n_rows % b:
        raise ValueError(
            f"block_convert={b}: {A.n_rows} rows not divisible")
    bsr = A.to_scipy().tobsr((b, b))
    bsr.sort_indices()
    ro = torch.from_numpy(bsr.indptr.astype(np.int32))
    ci = torch.from_numpy(bsr.indices.astype(np.int32))
    va = torch.from_numpy(np.ascontiguousarray(bsr.data)).to(A.dtype)
    dev = A.values.device
    return CSRMatrix(ro.to(dev), ci.to(dev), va.to(dev),
                     n_cols=bsr.shape[1] // b, block_dim=b)


@_amgx_try
def AMGX_write_system(m: _MatrixHandle, rhs: _VectorHandle,
                      sol: _VectorHandle, path: str):
    from .io.matrix_market import write_system
    b = rhs.v if rhs is not None else None
    x = sol.v if sol is not None else None
    if path.endswith((".bin", ".amgxb")):     # binary writer (reference
        from .io.binary import write_system_binary   # matrix_writer=binary)
        write_system_binary(path, m.A, b, x)
        return RC_OK
    write_system(path, m.A, b, x)
    return RC_OK


@_amgx_try
def AMGX_generate_distributed_poisson_7pt(m: _MatrixHandle,
                                          rhs: _VectorHandle,
                                          sol: _VectorHandle,
                                          allocated_halo_depth, num_import_rings,
                                          nx, ny, nz, px=1, py=1, pz=None):
    """Built-in 3D Poisson generator (reference src/amgx_c.cu:4566-4731).
    Current partitioner: z-slabs (px=py=1)."""
    import torch.distributed as tdist
    from .problems import poisson_3d, poisson_3d_local
    from .distributed.manager import DistributedManager
    mem, vecprec, matprec = _parse_mode(m.mode)
    dev = mem if mem == "cpu" else m.res.device
    world = tdist.get_world_size() if tdist.is_initialized() else 1
    if world == 1:
        m.A = poisson_3d(nx, ny, nz, device=dev, dtype=matprec)
        n = m.A.n_rows
        if rhs is not None:
            rhs.v = torch.ones(n, dtype=vecprec, device=dev)
            rhs.n = n
        if sol is not None:
            sol.v = torch.zeros(n, dtype=vecprec, device=dev)
            sol.n = n
        return RC_OK
    assert px == 1 and py == 1, "x/y process grids: z-slab partitioner only"
    rank = tdist.get_rank()
    ro, cols, vals, rs = poisson_3d_local(nx, ny, nz, rank, world)
    m.A = DistributedManager.upload_global_csr(
        ro, cols, vals, nx * ny * nz, rs, nx * ny * nz * world,
        device=dev, dtype=matprec)
    mgr = m.A.manager
    if rhs is not None:
        rhs.v = mgr.new_ext_vec(vecprec)
        rhs.v[:mgr.owned_size] = 1.0
        rhs.n = mgr.n_local
        rhs.bound = m
    if sol is not None:
        sol.v = mgr.new_ext_vec(vecprec)
        sol.n = mgr.n_local
        sol.bound = m
    return RC_OK


@_amgx_try
def AMGX_pin_memory(ptr, bytes_=0):
    return RC_OK


@_amgx_try
def AMGX_unpin_memory(ptr):
    return RC_OK


# solve-status enum (reference AMGX_SOLVE_STATUS, include/amgx_c.h)
AMGX_SOLVE_SUCCESS = 0
AMGX_SOLVE_FAILED = 1
AMGX_SOLVE_DIVERGED = 2


# -------------------------------------------------- remaining API surface
_ERROR_STRINGS = {
    RC_OK: "No error",
    RC_BAD_PARAMETERS: "Incorrect parameters",
    RC_UNKNOWN: "Unknown error",
    RC_NOT_SUPPORTED_TARGET: "Unsupported target",
    RC_NOT_SUPPORTED_BLOCKSIZE: "Unsupported block size",
    RC_CUDA_FAILURE: "HIP/GPU failure",
    RC_IO_ERROR: "I/O error",
    RC_BAD_MODE: "Invalid mode",
    RC_CORE: "Core library error",
    RC_PLUGIN: "Plugin error",
    RC_BAD_CONFIGURATION: "Invalid configuration",
    RC_NOT_IMPLEMENTED: "Not implemented",
    RC_LICENSE_NOT_FOUND: "License not found",
    RC_INTERNAL: "Internal error",
}


def _c_abi_world_size():
    """Helper for the C ABI shim (csrc_capi/amgx_c_shim.cpp): world size of
    the live torch.distributed group, 1 when single-process."""
    import torch.distributed as tdist
    return tdist.get_world_size() if tdist.is_initialized() else 1


def AMGX_get_error_string(rc: int):
    """Reference AMGX_get_error_string (include/amgx_c.h)."""
    return _ERROR_STRINGS.get(rc, "Unknown error code")


@_amgx_try
def AMGX_config_create_from_file_and_string(path: str, options: str):
    rc, h = AMGX_config_create_from_file(path)
    assert rc == RC_OK
    rc = AMGX_config_add_parameters(h, options)
    assert rc == RC_OK
    return RC_OK, h


@_amgx_try
def AMGX_distribution_set_32bit_colindices(dist: _DistributionHandle,
                                           use32: bool):
    dist.colindices_32bit = bool(use32)
    return RC_OK


@_amgx_try
def AMGX_matrix_get_nnz(m: _MatrixHandle):
    return RC_OK, m.A.nnz


@_amgx_try
def AMGX_matrix_attach_geometry(m: _MatrixHandle, geox, geoy=None, geoz=None,
                                n=None, dimension=None):
    """Reference AMGX_matrix_attach_geometry: coordinates for the GEO
    selector. Accepts either one (n,dim) array or per-axis arrays."""
    if geoy is None and geoz is None:
        coords = np.asarray(geox, dtype=np.float64)
    else:
        axes = [np.asarray(a, dtype=np.float64)
                for a in (geox, geoy, geoz) if a is not None]
        coords = np.stack(axes, axis=-1)
    m.A._cache["geometry"] = coords
    return RC_OK


@_amgx_try
def AMGX_matrix_attach_coloring(m: _MatrixHandle, row_coloring, num_rows=None,
                                num_colors=None):
    """Reference AMGX_matrix_attach_coloring: user-provided row coloring used
    by multicolor smoothers instead of a computed one."""
    from .amg.coloring import MatrixColoring
    colors = torch.as_tensor(np.asarray(row_coloring), dtype=torch.int32)
    num = int(num_colors) if num_colors is not None \
        else int(colors.max()) + 1
    m.A._cache["coloring"] = MatrixColoring(
        colors.to(m.A.row_offsets.device), num)
    return RC_OK


@_amgx_try
def AMGX_matrix_set_boundary_separation(m: _MatrixHandle, flag: int):
    m.boundary_separation = int(flag)
    return RC_OK


@_amgx_try
def AMGX_matrix_upload_all_global_32(m, n_global, n, nnz, block_dimx,
                                     block_dimy, row_ptrs,
                                     col_indices_global, data,
                                     diag_data=None, allocated_halo_depth=1,
                                     num_import_rings=1,
                                     partition_vector=None):
    """32-bit global column variant (reference amgx_c.h
    AMGX_matrix_upload_all_global_32); indices are widened internally."""
    cols = np.asarray(col_indices_global, dtype=np.int64)
    return AMGX_matrix_upload_all_global(
        m, n_global, n, nnz, block_dimx, block_dimy, row_ptrs, cols, data,
        diag_data, allocated_halo_depth, num_import_rings, partition_vector)


@_amgx_try
def AMGX_matrix_vector_multiply(m: _MatrixHandle, x: _VectorHandle,
                                y: _VectorHandle):
    """y = A x (reference AMGX_matrix_vector_multiply)."""
    if y.v is None or y.v.numel() != x.v.numel():
        y.v = torch.zeros_like(x.v)
        y.n = x.n
        y.block_dim = x.block_dim
    ops.spmv(m.A, x.v, y.v)
    return RC_OK


@_amgx_try
def AMGX_matrix_comm_from_maps_one_ring(m: _MatrixHandle, allocated_halo_depth,
                                        num_neighbors, neighbors,
                                        send_sizes, send_maps,
                                        recv_sizes, recv_maps):
    """Build the distributed halo structure from user-provided B2L maps
    (reference AMGX_matrix_comm_from_maps_one_ring, src/amgx_c.cu ->
    cacheMaps + updateMapsReorder). The local matrix must already index halo
    columns as n_local + position, where positions follow the concatenated
    recv maps."""
    import torch.distributed as tdist
    from .distributed.manager import DistributedManager
    A = m.A
    assert A is not None, "upload the local matrix first"
    dev = A.row_offsets.device
    mgr = DistributedManager(dev, A.block_dim)
    mgr.n_local = A.n_rows
    n_halo = int(sum(int(s) for s in recv_sizes))
    mgr.n_halo = n_halo
    if tdist.is_initialized():
        counts = [None] * mgr.world
        tdist.all_gather_object(counts, A.n_rows)
        mgr.row_start = int(sum(counts[:mgr.rank]))
        mgr.n_global = int(sum(counts))
    else:
        mgr.row_start, mgr.n_global = 0, A.n_rows
    pos = 0
    for k in range(int(num_neighbors)):
        nb = int(neighbors[k])
        mgr.neighbors.append(nb)
        smap = np.asarray(send_maps[k], dtype=np.int64)
        mgr.b2l.append(torch.from_numpy(smap.astype(np.int32)).to(dev))
        sz = int(recv_sizes[k])
        rmap = np.asarray(recv_maps[k], dtype=np.int64)
        # our layout needs per-neighbor contiguous halo slices in map order
        assert (rmap == np.arange(pos, pos + sz) + mgr.n_local).all() or \
               (rmap == np.arange(pos, pos + sz)).all(), \
            "recv maps must be consecutive halo positions"
        mgr.halo_slices.append((pos, pos + sz))
        pos += sz
    mgr.halo_global = np.arange(n_halo)   # opaque (maps-based upload)
    # rows are NOT renumbered interior-first here, so no row is provably
    # halo-free: treat every row as boundary (computed after the exchange)
    mgr.boundary_start = 0
    mgr.row_perm = torch.arange(A.n_rows, dtype=torch.int64, device=dev)
    mgr.row_iperm = mgr.row_perm.clone()
    mgr._alloc_send_bufs(A.dtype)
    A.manager = mgr
    A.n_cols = A.n_rows + n_halo
    return RC_OK


@_amgx_try
def AMGX_matrix_comm_from_maps(m, allocated_halo_depth, num_import_rings,
                               num_neighbors, neighbors, send_sizes,
                               send_maps, recv_sizes, recv_maps):
    assert int(num_import_rings) == 1, "only one-ring maps supported"
    return AMGX_matrix_comm_from_maps_one_ring(
        m, allocated_halo_depth, num_neighbors, neighbors, send_sizes,
        send_maps, recv_sizes, recv_maps)


@_amgx_try
def AMGX_read_system_maps_one_ring(m: _MatrixHandle):
    """Return the halo maps of a distributed matrix (reference
    AMGX_read_system_maps_one_ring out-params)."""
    mgr = getattr(m.A, "manager", None)
    assert mgr is not None, "matrix is not distributed"
    neighbors = list(mgr.neighbors)
    send_maps = [b.cpu().numpy().copy() for b in mgr.b2l]
    send_sizes = [int(b.numel()) for b in mgr.b2l]
    recv_maps = [np.arange(lo, hi) + mgr.n_local
                 for (lo, hi) in mgr.halo_slices]
    recv_sizes = [int(hi - lo) for (lo, hi) in mgr.halo_slices]
    return (RC_OK, len(neighbors), neighbors, send_sizes, send_maps,
            recv_sizes, recv_maps)


@_amgx_try
def AMGX_free_system_maps_one_ring(*args):
    return RC_OK   # Python GC owns the arrays


@_amgx_try
def AMGX_solver_calculate_residual_norm(s: _SolverHandle, m: _MatrixHandle,
                                        rhs: _VectorHandle,
                                        sol: _VectorHandle):
    r = ops.residual(m.A, sol.v, rhs.v)
    mgr = getattr(m.A, "manager", None)
    if mgr is not None:
        nrm = mgr.global_norm(float(torch.linalg.vector_norm(
            r.reshape(-1)[:mgr.owned_size])), "L2")
    else:
        nrm = float(torch.linalg.vector_norm(r))
    return RC_OK, nrm


@_amgx_try
def AMGX_solver_register_print_callback(s: _SolverHandle, cb):
    # per-solver output redirection (reference solver-level callback):
    # consumed by Solver._out in solvers/base.py
    s.print_callback = cb
    if s.solver is not None:
        s.solver.print_cb = cb
    return RC_OK


@_amgx_try
def AMGX_read_system_distributed(m: _MatrixHandle, rhs: _VectorHandle,
                                 sol: _VectorHandle, path: str,
                                 allocated_halo_depth=1, num_partitions=None,
                                 partition_sizes=None, partition_vector=None):
    """Rank-partitioned read (reference src/distributed/distributed_io.cu
    DistributedRead): every rank reads the file and keeps its contiguous row
    slab; halo structure is rebuilt from global column ids."""
    import torch.distributed as tdist
    from .distributed.manager import DistributedManager
    from .io.matrix_market import read_system
    from .io.binary import is_binary_file, read_system_any
    mem, vecprec, matprec = _parse_mode(m.mode)
    dev = mem if mem == "cpu" else m.res.device
    reader = read_system_any if is_binary_file(path) else read_system
    A, b, x0 = reader(path, device="cpu", dtype=matprec)
    world = tdist.get_world_size() if tdist.is_initialized() else 1
    rank = tdist.get_rank() if tdist.is_initialized() else 0
    n = A.n_rows
    if partition_vector is not None:
        pv = np.asarray(partition_vector)
        sizes = [int((pv == r).sum()) for r in range(world)]
    elif partition_sizes is not None:
        sizes = [int(s) for s in partition_sizes]
    else:
        base = n // world
        sizes = [base + (1 if r < n % world else 0) for r in range(world)]
    offs = np.zeros(world + 1, dtype=np.int64)
    offs[1:] = np.cumsum(sizes)
    lo, hi = int(offs[rank]), int(offs[rank + 1])
    ro = A.row_offsets.cpu().numpy().astype(np.int64)
    ci = A.col_indices.cpu().numpy().astype(np.int64)
    va = A.values.cpu().numpy()
    s0, s1 = ro[lo], ro[hi]
    local_ro = ro[lo:hi + 1] - s0
    m.A = DistributedManager.upload_global_csr(
        local_ro, ci[s0:s1], va[s0:s1], hi - lo, lo, n,
        device=dev, block_dim=A.block_dim, dtype=matprec)
    mgr = m.A.manager
    if rhs is not None:
        rhs.v = mgr.new_ext_vec(vecprec)
        src = b if b is not None else torch.ones(n, dtype=vecprec)
        rhs.v[:mgr.owned_size] = mgr.permute_in(
            src[lo * A.block_dim:hi * A.block_dim].to(vecprec).to(dev)
        )[:mgr.owned_size]
        rhs.n = mgr.n_local
        rhs.bound = m
    if sol is not None:
        sol.v = mgr.new_ext_vec(vecprec)
        if x0 is not None:
            sol.v[:mgr.owned_size] = mgr.permute_in(
                x0[lo * A.block_dim:hi * A.block_dim].to(vecprec).to(dev)
            )[:mgr.owned_size]
        sol.n = mgr.n_local
        sol.bound = m
    return RC_OK


@_amgx_try
def AMGX_read_system_global(m: _MatrixHandle, rhs: _VectorHandle,
                            sol: _VectorHandle, path: str):
    return AMGX_read_system_distributed(m, rhs, sol, path)


@_amgx_try
def AMGX_write_system_distributed(m: _MatrixHandle, rhs: _VectorHandle,
                                  sol: _VectorHandle, path: str,
                                  allocated_halo_depth=1, num_partitions=None,
                                  partition_sizes=None,
                                  partition_vector=None):
    """Gather the distributed system to rank 0 and write one file (reference
    AMGX_write_system_distributed consolidates partitions)."""
    import torch.distributed as tdist
    from .io.matrix_market import write_system
    from .io.binary import write_system_binary
    mgr = getattr(m.A, "manager", None)
    if mgr is None:
        return AMGX_write_system(m, rhs, sol, path)
    A = m.A
    perm = mgr.row_perm.cpu().numpy()          # internal new -> old local
    ro = A.row_offsets.cpu().numpy().astype(np.int64)
    ci = A.col_indices.cpu().numpy().astype(np.int64)
    va = A.values.cpu().numpy()
    # back to user (old-local) row order with GLOBAL columns
    iold = np.argsort(perm, kind="stable")      # old -> new position
    halo_map = mgr.halo_global
    rows_out, cols_out, vals_out = [], [], []
    for old in range(mgr.n_local):
        new = int(iold[old])
        s, e = ro[new], ro[new + 1]
        cols = ci[s:e].copy()
        own = cols < mgr.n_local
        gcols = np.empty_like(cols)
        gcols[own] = perm[cols[own]] + mgr.row_start
        gcols[~own] = halo_map[cols[~own] - mgr.n_local]
        order = np.argsort(gcols)
        rows_out.append(np.full(e - s, old + mgr.row_start, dtype=np.int64))
        cols_out.append(gcols[order])
        vals_out.append(va[s:e][order])
    payload = (np.concatenate(rows_out) if rows_out else np.zeros(0),
               np.concatenate(cols_out) if cols_out else np.zeros(0),
               np.concatenate(vals_out) if vals_out else np.zeros(0),
               mgr.permute_out(rhs.v).cpu().numpy() if rhs is not None
               and rhs.v is not None else None,
               mgr.permute_out(sol.v).cpu().numpy() if sol is not None
               and sol.v is not None else None)
    gathered = [None] * mgr.world
    tdist.all_gather_object(gathered, payload)
    if mgr.rank == 0:
        import scipy.sparse as sp
        rows = np.concatenate([g[0] for g in gathered])
        cols = np.concatenate([g[1] for g in gathered])
        vals = np.concatenate([g[2] for g in gathered], axis=0)
        n = mgr.n_global
        if A.block_dim == 1:
            sm = sp.csr_matrix((vals, (rows, cols)), shape=(n, n))
            Afull = CSRMatrix.from_scipy(sm, dtype=A.dtype)
        else:
            order = np.argsort(rows * n + cols, kind="stable")
            ro_f = np.zeros(n + 1, dtype=np.int64)
            np.add.at(ro_f[1:], rows, 1)
            np.cumsum(ro_f, out=ro_f)
            Afull = CSRMatrix(
                torch.from_numpy(ro_f.astype(np.int32)),
                torch.from_numpy(cols[order].astype(np.int32)),
                torch.from_numpy(vals[order]).to(A.dtype),
                n_cols=n, block_dim=A.block_dim)
        bfull = (np.concatenate([g[3] for g in gathered])
                 if gathered[0][3] is not None else None)
        xfull = (np.concatenate([g[4] for g in gathered])
                 if gathered[0][4] is not None else None)
        bt = torch.from_numpy(bfull) if bfull is not None else None
        xt = torch.from_numpy(xfull) if xfull is not None else None
        if path.endswith((".bin", ".amgxb")):
            write_system_binary(path, Afull, bt, xt)
        else:
            write_system(path, Afull, bt, xt)
    if tdist.is_initialized():
        tdist.barrier()
    return RC_OK


# ------------------------------------------------------------- eigensolvers
class _EigenSolverHandle:
    def __init__(self, res: Resources, mode: str, cfg: AMGConfig):
        self.res = res
        self.mode = mode
        self.cfg = cfg
        self.solver = None
        self.status = None


@_amgx_try
def AMGX_eigensolver_create(res: _ResourcesHandle, mode: str,
                            cfg: _ConfigHandle):
    """Reference AMGX_eigensolver_create (include/amgx_eig_c.h:16)."""
    _parse_mode(mode)
    return RC_OK, _EigenSolverHandle(res.res, mode, cfg.cfg)


@_amgx_try
def AMGX_eigensolver_setup(s: _EigenSolverHandle, m: _MatrixHandle):
    from .eigensolvers import create_eigensolver
    s.solver = create_eigensolver(s.cfg.root_scope(), resources=s.res)
    s.solver.setup(m.A)
    return RC_OK


@_amgx_try
def AMGX_eigensolver_pagerank_setup(s: _EigenSolverHandle, m: _MatrixHandle):
    from .eigensolvers import create_eigensolver
    s.solver = create_eigensolver(s.cfg.root_scope(), resources=s.res)
    s.solver.pagerank_setup(m.A)
    return RC_OK


@_amgx_try
def AMGX_eigensolver_solve(s: _EigenSolverHandle, x0: _VectorHandle = None):
    st = s.solver.solve(x0.v if x0 is not None else None)
    s.status = st
    if x0 is not None and st.eigenvector is not None:
        v = st.eigenvector
        x0.v = v[:, 0] if v.dim() > 1 else v
        x0.n = int(x0.v.numel())
    return RC_OK


@_amgx_try
def AMGX_eigensolver_destroy(s: _EigenSolverHandle):
    s.solver = None
    return RC_OK
