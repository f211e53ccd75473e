#!/usr/bin/env python3
"""Scale-config end-to-end exercise: runs ALL pipeline stages (graphs,
walks, integration, training, L-groups, scoring, writers) at synthetic
scale, in-memory (no TSV round-trip — the file formats are covered by the
ex_*-scale tests). Validates that steps 5-7 hold up at 200k-1M genes
(torch k-means backend, vectorized scoring/writers)."""
import argparse
import sys
import tempfile
import time

import numpy as np
import torch

sys.path.insert(0, __file__.rsplit("/", 2)[0])

from bench import build_dataset  # noqa: E402
from g2vec_amd.cluster import find_lgroups  # noqa: E402
from g2vec_amd.config import G2VecConfig  # noqa: E402
from g2vec_amd.io import writers  # noqa: E402
from g2vec_amd.models.cbow import CbowTrainer  # noqa: E402
from g2vec_amd.parallel.dist import single  # noqa: E402
from g2vec_amd.pipeline import generate_paths  # noqa: E402
from g2vec_amd.scoring import select_biomarkers  # noqa: E402


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--n-genes", type=int, default=200000)
    ap.add_argument("--n-edges", type=int, default=20000000)
    ap.add_argument("--n-modules", type=int, default=64)
    ap.add_argument("--hidden", type=int, default=256)
    ap.add_argument("--epochs", type=int, default=30)
    args = ap.parse_args()

    dev = torch.device("cuda" if torch.cuda.is_available() else "cpu")
    t = {}
    t0 = time.perf_counter()
    expr, labels, edge_idx, n_genes = build_dataset(
        0, args.n_genes, args.n_edges, n_modules=args.n_modules)
    t["synth"] = time.perf_counter() - t0

    cfg = G2VecConfig(hidden=args.hidden, epochs=args.epochs, seed=0,
                      device=str(dev.type))
    expr_t = torch.from_numpy(expr).to(dev)
    labels_t = torch.from_numpy(labels).to(dev)
    edge_t = torch.from_numpy(edge_idx).to(dev)
    ctx = single(dev)

    t0 = time.perf_counter()
    ps, freq, n_in_paths, _stats = generate_paths(
        cfg, expr_t, labels_t, edge_t, n_genes, ctx, log=print)
    if dev.type == "cuda":
        torch.cuda.synchronize()
    t["paths"] = time.perf_counter() - t0
    print(f"paths: {ps.n_paths}, genes-in-paths {n_in_paths}", file=sys.stderr)

    t0 = time.perf_counter()
    res = CbowTrainer(cfg, n_genes, dev, ctx, log=lambda *a, **k: None).train(ps)
    t["train"] = time.perf_counter() - t0

    W = res.W_ih.float().cpu().numpy()
    t0 = time.perf_counter()
    lg = find_lgroups(W, freq.cpu().numpy(), device=dev)
    t["lgroups"] = time.perf_counter() - t0

    t0 = time.perf_counter()
    genes = [f"G{i:07d}" for i in range(n_genes)]
    bio = select_biomarkers(W, lg, expr, labels, genes, 50)
    t["scoring"] = time.perf_counter() - t0

    t0 = time.perf_counter()
    out = tempfile.mkdtemp() + "/scale"
    writers.write_biomarkers(out, bio)
    writers.write_lgroups(out, lg, genes)
    writers.write_vectors(out, W, genes)
    t["write"] = time.perf_counter() - t0

    print({"n_genes": n_genes, "n_paths": ps.n_paths,
           "acc_val": round(res.acc_val, 4), "n_biomarkers": len(bio),
           "lg_counts": np.bincount(lg, minlength=3).tolist(),
           "timers_s": {k: round(v, 2) for k, v in t.items()}})


if __name__ == "__main__":
    main()
