#!/usr/bin/env python3
"""Empirically decode the v_mfma_f64_4x4x4f64 fragment layout on gfx950.

Full 64x64 one-hot probe: for every (A-lane i, B-lane j) pair record which
C lanes light up.  The first pass (16x16, block-0 only) showed the operands
are NOT laid out as 4 independent 16-lane blocks, so this version scans the
whole wave and prints the raw match structure for offline decoding, plus a
compressed per-lane role table.
"""
import json
import sys

import torch

sys.path.insert(0, ".")
from amgx_amd import _core  # noqa: E402


def main():
    matches = []       # (i, j, c_lane)
    for i in range(64):
        a = torch.zeros(64, dtype=torch.float64)
        a[i] = 1.0
        ac = a.cuda()
        # batch the 64 B probes as columns? keep simple: one probe per j
        for j in range(64):
            b = torch.zeros(64, dtype=torch.float64)
            b[j] = 1.0
            c = _core.mfma4_probe(ac, b.cuda()).cpu()
            nz = (c.abs() > 0.5).nonzero().flatten().tolist()
            for l in nz:
                matches.append((i, j, l))
    print("n_matches:", len(matches))
    # per-A-lane: set of (j, c) pairs
    from collections import defaultdict
    a_map = defaultdict(list)
    for i, j, l in matches:
        a_map[i].append((j, l))
    for i in range(64):
        if a_map[i]:
            print(f"A[{i}]:", a_map[i])
    print("RAW", json.dumps(matches))


if __name__ == "__main__":
    main()
