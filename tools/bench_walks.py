#!/usr/bin/env python3
"""Walk-kernel microbench: builds the ex_*-shaped per-group graphs and times
walk_kernel dispatches alone (torch events), separate from the pipeline."""
import argparse
import sys
import time

import numpy as np
import torch

sys.path.insert(0, __file__.rsplit("/", 2)[0])

from bench import build_dataset  # noqa: E402
from g2vec_amd.graph import build_group_graph  # noqa: E402
from g2vec_amd.walks import generate_walks  # noqa: E402


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--n-genes", type=int, default=7523)
    ap.add_argument("--n-edges", type=int, default=298799)
    ap.add_argument("--reps", type=int, default=10)
    ap.add_argument("--len-path", type=int, default=80)
    ap.add_argument("--iters", type=int, default=10)
    args = ap.parse_args()

    dev = torch.device("cuda" if torch.cuda.is_available() else "cpu")
    expr, labels, edge_idx, n_genes = build_dataset(0, args.n_genes,
                                                    args.n_edges)
    expr_t = torch.from_numpy(expr).to(dev)
    labels_t = torch.from_numpy(labels).to(dev)
    edge_t = torch.from_numpy(edge_idx).to(dev)
    graphs = [build_group_graph(expr_t, labels_t, g, edge_t, n_genes)
              for g in (0, 1)]
    for g in (0, 1):
        print(f"group {g}: nnz {graphs[g].col_idx.numel()}", file=sys.stderr)

    # warmup
    for g in (0, 1):
        generate_walks(graphs[g], args.len_path, args.reps, 0, g)
    if dev.type == "cuda":
        torch.cuda.synchronize()
    t0 = time.perf_counter()
    n_walks = 0
    lens = []
    for it in range(args.iters):
        for g in (0, 1):
            ws = generate_walks(graphs[g], args.len_path, args.reps, it, g)
            n_walks += ws.nodes.shape[0]
            lens.append(ws.lengths.float().mean().item())
    if dev.type == "cuda":
        torch.cuda.synchronize()
    dt = time.perf_counter() - t0
    print(f"{n_walks} walks in {dt:.4f}s = {n_walks/dt/1e6:.2f} M walks/s; "
          f"mean len {np.mean(lens):.1f}; "
          f"{n_walks*np.mean(lens)/dt/1e9:.2f} G steps/s")


if __name__ == "__main__":
    main()
