#!/usr/bin/env python3
"""Empirically decode the v_mfma_f64_4x4x4_4b_f64 fragment layout on gfx950.

Probes all 16x16 one-hot (A-slot, B-slot) pairs of block 0 through
_core.mfma4_probe and solves for the lane->(m,k) / (k,n) / (m,n) mappings:
an A-slot holding A[m][k] pairs with exactly the 4 B-slots holding B[k][*],
and the product lands in the C slot of C[m][n].  Prints the mapping tables
plus a consistency check, so kernels_mfma.hip's layout constants can be
pinned from one run.
"""
import sys

import torch

sys.path.insert(0, ".")
from amgx_amd import _core  # noqa: E402


def main():
    match = {}        # (i, j) -> c-slot
    for i in range(16):
        for j in range(16):
            a = torch.zeros(64, dtype=torch.float64)
            b = torch.zeros(64, dtype=torch.float64)
            a[i] = 1.0
            b[j] = 1.0
            c = _core.mfma4_probe(a.cuda(), b.cuda()).cpu()
            nz = (c[:16].abs() > 0.5).nonzero().flatten().tolist()
            assert len(nz) <= 1, f"({i},{j}) -> {nz}"
            if nz:
                match[(i, j)] = nz[0]

    # group A slots by their set of matching B slots (same k)
    a_sets = {i: frozenset(j for (ii, j) in match if ii == i)
              for i in range(16)}
    k_classes = []
    for i in range(16):
        if a_sets[i] not in [s for s, _ in k_classes]:
            k_classes.append((a_sets[i], len(k_classes)))
    k_of_a = {i: dict(k_classes)[a_sets[i]] for i in range(16)}
    # m label: order within the k class
    m_of_a = {}
    for kc, _ in k_classes:
        members = sorted(i for i in range(16) if a_sets[i] == kc)
        for m, i in enumerate(members):
            m_of_a[i] = m
    # B slots: k from which class's set they belong to; n = order within
    k_of_b, n_of_b = {}, {}
    for kset, klab in k_classes:
        members = sorted(kset)
        for n, j in enumerate(members):
            k_of_b[j] = klab
            n_of_b[j] = n
    # C slots: lane of C[m][n]
    c_of_mn = {}
    for (i, j), l in match.items():
        c_of_mn[(m_of_a[i], n_of_b[j])] = l

    print("A slot -> (m, k):", [(i, m_of_a[i], k_of_a[i])
                                for i in range(16)])
    print("B slot -> (k, n):", [(j, k_of_b[j], n_of_b[j])
                                for j in range(16)])
    print("C (m, n) -> slot:", sorted(c_of_mn.items()))

    # verify with a random product under the decoded mapping
    g = torch.Generator().manual_seed(3)
    Am = torch.rand(4, 4, generator=g, dtype=torch.float64)
    Bm = torch.rand(4, 4, generator=g, dtype=torch.float64)
    a = torch.zeros(64, dtype=torch.float64)
    b = torch.zeros(64, dtype=torch.float64)
    for i in range(16):
        a[i] = Am[m_of_a[i]][k_of_a[i]]
    for j in range(16):
        b[j] = Bm[k_of_b[j]][n_of_b[j]]
    c = _core.mfma4_probe(a.cuda(), b.cuda()).cpu()
    C = Am @ Bm
    ok = all(abs(float(c[c_of_mn[(m, n)]]) - float(C[m][n])) < 1e-12
             for m in range(4) for n in range(4))
    print("CONSISTENT" if ok else "INCONSISTENT")
    # also check the other three blocks share the layout
    for blk in range(1, 4):
        a2 = torch.zeros(64, dtype=torch.float64)
        b2 = torch.zeros(64, dtype=torch.float64)
        for i in range(16):
            a2[blk * 16 + i] = Am[m_of_a[i]][k_of_a[i]]
            b2[blk * 16 + i] = Bm[k_of_b[i]][n_of_b[i]]
        c2 = _core.mfma4_probe(a2.cuda(), b2.cuda()).cpu()
        ok2 = all(abs(float(c2[blk * 16 + c_of_mn[(m, n)]])
                      - float(C[m][n])) < 1e-12
                  for m in range(4) for n in range(4))
        print(f"block {blk}: {'same layout' if ok2 else 'DIFFERENT'}")


if __name__ == "__main__":
    main()
