#!/usr/bin/env python3
"""Difficulty calibration for the bench dataset.

Sweeps (shared_frac, off_frac) of the synthetic ex_*-shaped dataset and
prints the seeded val-ACC trajectory of the reference training config, so
the bench's convergence headline (wall-clock to val-ACC >= 0.88) can be
tuned to a REAL multi-epoch climb like the published transcript
(0.6336 -> 0.8837 over 27 epochs, reference README.md:35-41) instead of a
degenerate epoch-0 crossing.

Runs on CPU; use --reps 3 (default) to keep the Python walk oracle under
~30 s per variant. The chosen knobs go into bench.py's build_dataset.
"""
from __future__ import annotations

import argparse
import sys
import time

import torch

import os
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from bench import build_dataset                      # noqa: E402
from g2vec_amd.config import G2VecConfig             # noqa: E402
from g2vec_amd.graph import build_group_graph        # noqa: E402
from g2vec_amd.models.cbow import CbowTrainer        # noqa: E402
from g2vec_amd.paths import integrate_pathsets       # noqa: E402
from g2vec_amd.walks import generate_walks           # noqa: E402


def run_variant(shared_frac: float, off_frac: float, reps: int,
                epochs: int, seed: int = 0, train_seed: int = 0):
    expr, labels, edge_idx, n_genes = build_dataset(
        seed, shared_frac=shared_frac, off_frac=off_frac)
    expr_t = torch.from_numpy(expr)
    labels_t = torch.from_numpy(labels)
    edge_t = torch.from_numpy(edge_idx)
    t0 = time.perf_counter()
    walksets = []
    nnz = []
    for group in (0, 1):
        g = build_group_graph(expr_t, labels_t, group, edge_t, n_genes)
        nnz.append(int(g.col_idx.numel()))
        walksets.append(generate_walks(g, 80, reps, seed, group))
    n_walks = sum(int(w.nodes.shape[0]) for w in walksets)
    ps, _freq, n_gip = integrate_pathsets(walksets[0], walksets[1], n_genes)
    walk_s = time.perf_counter() - t0
    cfg = G2VecConfig(hidden=128, epochs=epochs, early_stop=False,
                      seed=train_seed, device="cpu", dtype="fp32")
    tr = CbowTrainer(cfg, n_genes, torch.device("cpu"),
                     log=lambda *a, **k: None)
    st = tr.setup(ps)
    hist = []
    for _ in range(epochs):
        _a_tr, a_val = tr.run_epoch(st)
        hist.append(a_val)
    cross = next((i for i, a in enumerate(hist) if a >= 0.88), None)
    return {"nnz": nnz, "n_walks": n_walks, "n_paths": ps.n_paths,
            "n_gip": n_gip, "walk_s": walk_s, "hist": hist, "cross": cross}


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--reps", type=int, default=3)
    ap.add_argument("--epochs", type=int, default=40)
    ap.add_argument("--shared", type=float, nargs="+", default=[0.3])
    ap.add_argument("--off", type=float, nargs="+", default=[0.55])
    ap.add_argument("--seed", type=int, default=0)
    ap.add_argument("--train-seeds", type=int, default=1)
    args = ap.parse_args()

    for sf in args.shared:
        for of in args.off:
            for ts in range(args.train_seeds):
                r = run_variant(sf, of, args.reps, args.epochs, args.seed,
                                train_seed=ts)
                h = r["hist"]
                marks = [0, 1, 2, 5, 10, 15, 20, 25, 30, 35, len(h) - 1]
                traj = " ".join(f"{i}:{h[i]:.3f}" for i in sorted(set(
                    m for m in marks if 0 <= m < len(h))))
                print(f"shared={sf} off={of} ts={ts} nnz={r['nnz']} "
                      f"paths={r['n_paths']}/{r['n_walks']} gip={r['n_gip']} "
                      f"cross@{r['cross']}  {traj}", flush=True)


if __name__ == "__main__":
    main()
