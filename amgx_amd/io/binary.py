"""Binary system format (reference ReadNVAMGBinary, src/readers.cu:1676 and
the binary writer in src/matrix_io.cu): a packed little-endian dump of the
CSR system plus optional diagonal/rhs/solution, for fast reload of large
systems. Layout (this framework's own format, role-equivalent to
NVAMGBinary):

    magic   8s   b"AMGXAMDB"
    version u32  (=1)
    flags   u32  bit0 has_diag, bit1 has_rhs, bit2 has_sol
    n       u64  rows
    nnz     u64
    bx, by  u32  block dims
    row_offsets i64[n+1]
    col_indices i64[nnz]
    values      f64[nnz*bx*by]
    [diag  f64[n*bx*by]]  [rhs  f64[n*bx]]  [sol  f64[n*bx]]
"""

from __future__ import annotations

import struct

import numpy as np
import torch

MAGIC = b"AMGXAMDB"


def write_system_binary(path: str, A, b=None, x=None):
    bd = A.block_dim
    n, nnz = A.n_rows, A.nnz
    flags = ((1 if A.diag is not None else 0)
             | (2 if b is not None else 0)
             | (4 if x is not None else 0))
    with open(path, "wb") as f:
        f.write(MAGIC)
        f.write(struct.pack("<IIQQII", 1, flags, n, nnz, bd, bd))
        f.write(A.row_offsets.cpu().numpy().astype("<i8").tobytes())
        f.write(A.col_indices.cpu().numpy().astype("<i8").tobytes())
        f.write(A.values.cpu().numpy().astype("<f8").tobytes())
        if A.diag is not None:
            f.write(A.diag.cpu().numpy().astype("<f8").tobytes())
        if b is not None:
            f.write(np.asarray(b.cpu() if torch.is_tensor(b) else b)
                    .astype("<f8").tobytes())
        if x is not None:
            f.write(np.asarray(x.cpu() if torch.is_tensor(x) else x)
                    .astype("<f8").tobytes())


def read_system_binary(path: str, device="cpu", dtype=torch.float64):
    from ..matrix import CSRMatrix
    with open(path, "rb") as f:
        magic = f.read(8)
        if magic != MAGIC:
            raise ValueError(f"{path}: not an AMGXAMDB binary file")
        version, flags, n, nnz, bx, by = struct.unpack("<IIQQII", f.read(32))
        if version != 1:
            raise ValueError(f"unsupported binary version {version}")
        ro = np.frombuffer(f.read(8 * (n + 1)), dtype="<i8")
        ci = np.frombuffer(f.read(8 * nnz), dtype="<i8")
        vals = np.frombuffer(f.read(8 * nnz * bx * by), dtype="<f8")
        diag = rhs = sol = None
        if flags & 1:
            diag = np.frombuffer(f.read(8 * n * bx * by), dtype="<f8")
        if flags & 2:
            rhs = np.frombuffer(f.read(8 * n * bx), dtype="<f8")
        if flags & 4:
            sol = np.frombuffer(f.read(8 * n * bx), dtype="<f8")
    dev = torch.device(device)
    va = torch.from_numpy(vals.copy()).to(dtype)
    if bx > 1:
        va = va.reshape(nnz, bx, by)
    dg = None
    if diag is not None:
        dg = torch.from_numpy(diag.copy()).to(dtype)
        dg = dg.reshape(n, bx, by) if bx > 1 else dg
        dg = dg.to(dev)
    A = CSRMatrix(torch.from_numpy(ro.astype(np.int32)).to(dev),
                  torch.from_numpy(ci.astype(np.int32)).to(dev),
                  va.to(dev), n_cols=int(n), block_dim=int(bx), diag=dg)
    b = torch.from_numpy(rhs.copy()).to(dtype).to(dev) if rhs is not None \
        else None
    x = torch.from_numpy(sol.copy()).to(dtype).to(dev) if sol is not None \
        else None
    return A, b, x


def is_binary_file(path: str) -> bool:
    try:
        with open(path, "rb") as f:
            head = f.read(14)
            return head[:8] == MAGIC or head == NVAMG_MAGIC
    except OSError:
        return False


# --------------------------------------------------------------- NVAMGBinary
# Interop with the reference's binary system format (header "%%NVAMGBinary\n"
# + 9 uint32 flags; reference reader src/readers.cu:1676-1900, writer
# src/matrix_io.cu:290-380): files written by upstream AmgX load here and
# vice versa. Layout:
#   "%%NVAMGBinary\n" (14 bytes)
#   u32[9]: is_mtx, is_rhs, is_soln, matrix_format (0=CSR,1=COO,+16 complex),
#           has_diag, block_dimx, block_dimy, num_rows, num_nz
#   i32[num_rows+1] row_offsets;  i32[num_nz] col_indices
#   f64[bx*by*(num_nz + has_diag*num_rows)] values (+ external diag tail)
#   [f64[num_rows*by] rhs]  [f64[num_rows*by] soln]
NVAMG_MAGIC = b"%%NVAMGBinary\n"
_NVAMG_COMPLEX = 16


def read_system_nvamg(path: str, device="cpu", dtype=torch.float64):
    from ..matrix import CSRMatrix
    with open(path, "rb") as f:
        if f.read(14) != NVAMG_MAGIC:
            raise ValueError(f"{path}: not an NVAMGBinary file")
        (is_mtx, is_rhs, is_soln, mat_fmt, has_diag, bx, by, n,
         nnz) = struct.unpack("<9I", f.read(36))
        if mat_fmt & _NVAMG_COMPLEX:
            raise NotImplementedError(
                "complex NVAMGBinary systems: load via the host path")
        if mat_fmt & 1:
            raise NotImplementedError(
                "COO NVAMGBinary files are unsupported (reference "
                "ReadNVAMGBinary rejects them too)")
        bb = bx * by
        ro = np.frombuffer(f.read(4 * (n + 1)), dtype="<i4")
        ci = np.frombuffer(f.read(4 * nnz), dtype="<i4")
        vals = np.frombuffer(f.read(8 * nnz * bb), dtype="<f8")
        diag = None
        if has_diag:
            diag = np.frombuffer(f.read(8 * n * bb), dtype="<f8")
        rhs = sol = None
        if is_rhs:
            rhs = np.frombuffer(f.read(8 * n * by), dtype="<f8")
        if is_soln:
            sol = np.frombuffer(f.read(8 * n * by), dtype="<f8")
    dev = torch.device(device)
    va = torch.from_numpy(vals.copy()).to(dtype)
    if bx > 1:
        va = va.reshape(nnz, bx, by)
    dg = None
    if diag is not None:
        dg = torch.from_numpy(diag.copy()).to(dtype)
        dg = dg.reshape(n, bx, by) if bx > 1 else dg
        dg = dg.to(dev)
    A = CSRMatrix(torch.from_numpy(ro.astype(np.int32)).to(dev),
                  torch.from_numpy(ci.astype(np.int32)).to(dev),
                  va.to(dev), n_cols=int(n), block_dim=int(bx), diag=dg)
    b = torch.from_numpy(rhs.copy()).to(dtype).to(dev) if rhs is not None \
        else None
    x = torch.from_numpy(sol.copy()).to(dtype).to(dev) if sol is not None \
        else None
    return A, b, x


def write_system_nvamg(path: str, A, b=None, x=None):
    bd = A.block_dim
    n, nnz = A.n_rows, A.nnz
    has_diag = 1 if A.diag is not None else 0
    with open(path, "wb") as f:
        f.write(NVAMG_MAGIC)
        f.write(struct.pack("<9I", 1, 1 if b is not None else 0,
                            1 if x is not None else 0, 0, has_diag,
                            bd, bd, n, nnz))
        f.write(A.row_offsets.cpu().numpy().astype("<i4").tobytes())
        f.write(A.col_indices.cpu().numpy().astype("<i4").tobytes())
        f.write(A.values.cpu().numpy().astype("<f8").tobytes())
        if A.diag is not None:
            f.write(A.diag.cpu().numpy().astype("<f8").tobytes())
        if b is not None:
            f.write(np.asarray(b.cpu() if torch.is_tensor(b) else b)
                    .astype("<f8").tobytes())
        if x is not None:
            f.write(np.asarray(x.cpu() if torch.is_tensor(x) else x)
                    .astype("<f8").tobytes())


def read_system_any(path: str, device="cpu", dtype=torch.float64):
    """Dispatch on magic: native AMGXAMDB or reference NVAMGBinary."""
    with open(path, "rb") as f:
        head = f.read(14)
    if head[:8] == MAGIC:
        return read_system_binary(path, device, dtype)
    if head == NVAMG_MAGIC:
        return read_system_nvamg(path, device, dtype)
    raise ValueError(f"{path}: unknown binary system format")
