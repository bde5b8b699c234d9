#!/usr/bin/env python3
"""Matrix-file converter (reference examples/convert.c): MatrixMarket <->
binary system files, auto-detecting the input format. The binary writer
emits the reference-compatible %%NVAMGBinary layout by default so files
interoperate with upstream AmgX; --native selects the AMGXAMDB layout.

Usage:
    python convert.py input.mtx output.bin          # mtx -> NVAMGBinary
    python convert.py input.bin output.mtx          # binary -> MatrixMarket
    python convert.py --native input.mtx output.bin # mtx -> AMGXAMDB
"""
import argparse
import os
import sys

sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)),
                                os.pardir))

from amgx_amd.io.binary import (is_binary_file, read_system_any,  # noqa: E402
                                write_system_binary, write_system_nvamg)
from amgx_amd.io.matrix_market import read_system, write_system  # noqa: E402


def main():
    ap = argparse.ArgumentParser(description=__doc__)
    ap.add_argument("input")
    ap.add_argument("output")
    ap.add_argument("--native", action="store_true",
                    help="write the AMGXAMDB binary layout instead of the "
                         "reference NVAMGBinary layout")
    args = ap.parse_args()

    if is_binary_file(args.input):
        A, b, x = read_system_any(args.input)
    else:
        A, b, x = read_system(args.input)
    print(f"read {args.input}: {A.n_rows} rows, {A.nnz} nnz, "
          f"block_dim {A.block_dim}")

    if args.output.endswith(".mtx"):
        write_system(args.output, A, b, x)
        kind = "MatrixMarket"
    elif args.native:
        write_system_binary(args.output, A, b, x)
        kind = "AMGXAMDB binary"
    else:
        write_system_nvamg(args.output, A, b, x)
        kind = "NVAMGBinary"
    print(f"wrote {args.output} ({kind})")


if __name__ == "__main__":
    main()
