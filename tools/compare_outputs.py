#!/usr/bin/env python3
"""Compare two G2Vec output triples (biomarkers / lgroups / vectors).

For users migrating from the reference: run both implementations on the
same inputs and diff the artifacts —

    python tools/compare_outputs.py REF_PREFIX OURS_PREFIX [--vec-atol 1e-4]

Exit 0 when biomarker lists match, L-group assignments match, and vector
values agree within tolerance; prints a per-file report either way.
(Note: run-to-run identity additionally needs fixed seeds on our side and
`--compat-lgroup-bug` for the shipped reference's L-group behavior —
docs/PARITY.md.)
"""
from __future__ import annotations

import argparse
import sys


def load_biomarkers(prefix):
    with open(prefix + "_biomarkers.txt") as f:
        lines = f.read().splitlines()
    assert lines[0] == "GeneSymbol", f"bad header in {prefix}_biomarkers.txt"
    return lines[1:]


def load_lgroups(prefix):
    with open(prefix + "_lgroups.txt") as f:
        lines = f.read().splitlines()
    out = {}
    for ln in lines[1:]:
        g, v = ln.split("\t")
        out[g] = int(v)
    return out


def load_vectors(prefix):
    with open(prefix + "_vectors.txt") as f:
        lines = f.read().splitlines()
    out = {}
    for ln in lines[1:]:
        parts = ln.split("\t")
        out[parts[0]] = [float(x) for x in parts[1:]]
    return out


def main() -> int:
    ap = argparse.ArgumentParser()
    ap.add_argument("ref_prefix")
    ap.add_argument("ours_prefix")
    ap.add_argument("--vec-atol", type=float, default=1e-4)
    args = ap.parse_args()
    ok = True

    a = load_biomarkers(args.ref_prefix)
    b = load_biomarkers(args.ours_prefix)
    inter = len(set(a) & set(b))
    same = a == b
    ok &= same
    print(f"biomarkers: {'MATCH' if same else 'DIFFER'} "
          f"({len(a)} vs {len(b)} entries, {inter} shared)")

    la, lb = load_lgroups(args.ref_prefix), load_lgroups(args.ours_prefix)
    genes_same = set(la) == set(lb)
    n_diff = sum(1 for g in la if g in lb and la[g] != lb[g])
    same = genes_same and n_diff == 0
    ok &= same
    print(f"lgroups:    {'MATCH' if same else 'DIFFER'} "
          f"({len(la)} genes; {n_diff} assignments differ"
          f"{'' if genes_same else '; gene sets differ'})")

    va, vb = load_vectors(args.ref_prefix), load_vectors(args.ours_prefix)
    genes_same = set(va) == set(vb)
    worst = 0.0
    if genes_same:
        for g in va:
            for x, y in zip(va[g], vb[g]):
                worst = max(worst, abs(x - y))
    same = genes_same and worst <= args.vec_atol
    ok &= same
    print(f"vectors:    {'MATCH' if same else 'DIFFER'} "
          f"(max |diff| {worst:.3e}, atol {args.vec_atol:g}"
          f"{'' if genes_same else '; gene sets differ'})")
    return 0 if ok else 1


if __name__ == "__main__":
    sys.exit(main())
