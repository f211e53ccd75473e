#!/usr/bin/env python3
"""Step-3 end-to-end breakdown (round-1 verdict item 4).

Times the WARMED input pipeline — per-group graph construction, walk
kernel, dedup/integration — at ex_* scale with sub-phase granularity,
so the gap between walk-kernel time (~1.5 ms) and step-3 wall is
attributed, optimized, and reported honestly in profiles/.

Usage: python tools/bench_step3.py [--iters 5] [--real-data] [--json OUT]
"""
from __future__ import annotations

import argparse
import json
import sys
import time

import numpy as np
import torch

import os
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from bench import build_dataset                      # noqa: E402
from g2vec_amd.graph import (build_group_graph, edge_pcc_weights,  # noqa: E402
                             zscore_group)
from g2vec_amd.paths import integrate_pathsets       # noqa: E402
from g2vec_amd.walks import generate_walks           # noqa: E402


class T:
    """Sync-bracketed phase timer; reentrant (nested phases nest on a
    stack, so integrate_pathsets' sub-timers can share the instance)."""

    def __init__(self, dev):
        self.dev = dev
        self.acc = {}
        self.stack = []

    def __call__(self, name):
        self.pending = name
        return self

    def __enter__(self):
        if self.dev.type == "cuda":
            torch.cuda.synchronize()
        self.stack.append((self.pending, time.perf_counter()))

    def __exit__(self, *a):
        if self.dev.type == "cuda":
            torch.cuda.synchronize()
        name, t0 = self.stack.pop()
        self.acc[name] = self.acc.get(name, 0.0) + (time.perf_counter() - t0)


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--iters", type=int, default=5)
    ap.add_argument("--reps", type=int, default=10)
    ap.add_argument("--len-path", type=int, default=80)
    ap.add_argument("--real-data", action="store_true")
    ap.add_argument("--json", default="")
    args = ap.parse_args()

    dev = torch.device("cuda" if torch.cuda.is_available() else "cpu")
    if args.real_data:
        from g2vec_amd.utils import refdata
        ds = refdata.make_real_dataset(seed=0)
        g2i = {g: i for i, g in enumerate(ds["net_genes"])}
        keep = np.array([g2i[g] for g in ds["expr_genes"]])
        idx_of = np.full(len(ds["net_genes"]), -1, np.int64)
        idx_of[keep] = np.arange(len(keep))
        e = ds["edge_idx"]
        m = (idx_of[e[:, 0]] >= 0) & (idx_of[e[:, 1]] >= 0)
        edge_idx = np.stack([idx_of[e[m, 0]], idx_of[e[m, 1]]],
                            1).astype(np.int32)
        expr, labels, n_genes = ds["expr"], np.asarray(ds["labels"]), 7523
    else:
        expr, labels, edge_idx, n_genes = build_dataset(0)
    expr_t = torch.from_numpy(expr).to(dev)
    labels_t = torch.from_numpy(labels).to(dev)
    edge_t = torch.from_numpy(edge_idx).to(dev)

    # warmup round: one FULL iteration including integrate (allocator
    # arenas for the full-size tensors, kernel load, torch op caches) —
    # its wall is reported separately as step3_cold_ms
    if dev.type == "cuda":
        torch.cuda.synchronize()
    tc = T(dev)
    c0 = time.perf_counter()
    wsets = []
    for group in (0, 1):
        with tc("cold.graph"):
            g = build_group_graph(expr_t, labels_t, group, edge_t, n_genes)
        with tc("cold.walks"):
            wsets.append(generate_walks(g, args.len_path, args.reps, 999,
                                        group))
    with tc("cold.integrate"):
        integrate_pathsets(wsets[0], wsets[1], n_genes)
    if dev.type == "cuda":
        torch.cuda.synchronize()
    cold_s = time.perf_counter() - c0
    del wsets

    t = T(dev)
    n_walks = n_paths = 0
    wall = 0.0
    for it in range(args.iters):
        if dev.type == "cuda":
            torch.cuda.synchronize()
        w0 = time.perf_counter()
        walksets = []
        for group in (0, 1):
            with t("graph.zscore"):
                zt = zscore_group(expr_t, labels_t, group)
            with t("graph.edge_dedup"):
                key = edge_t[:, 0].long() * n_genes + edge_t[:, 1].long()
                key = torch.unique(key)
                src = (key // n_genes).int()
                dst = (key % n_genes).int()
                pairs = torch.stack([src, dst], dim=1)
            with t("graph.pcc"):
                w = edge_pcc_weights(zt, pairs, "auto")
            with t("graph.threshold_csr"):
                keep = w > 0.5
                src2, dst2, w2 = src[keep], dst[keep], w[keep]
                order = torch.argsort(src2.long() * n_genes + dst2.long())
                src2, dst2, w2 = src2[order], dst2[order], w2[order]
                counts = torch.bincount(src2.long(), minlength=n_genes)
                row_ptr = torch.zeros(n_genes + 1, dtype=torch.int64,
                                      device=dev)
                torch.cumsum(counts, 0, out=row_ptr[1:])
                from g2vec_amd.graph import CsrGraph
                g = CsrGraph(row_ptr.int().contiguous(), dst2.contiguous(),
                             w2.float().contiguous(), n_genes)
            with t("walks.kernel"):
                ws = generate_walks(g, args.len_path, args.reps, it, group)
            walksets.append(ws)
        with t("integrate"):
            ps, _freq, _nip = integrate_pathsets(walksets[0], walksets[1],
                                                 n_genes, timers=t)
        if dev.type == "cuda":
            torch.cuda.synchronize()
        wall += time.perf_counter() - w0
        n_walks = sum(int(w.nodes.shape[0]) for w in walksets)
        n_paths = ps.n_paths

    out = {
        "device": str(dev), "iters": args.iters,
        "n_genes": n_genes, "n_walks": n_walks, "n_paths": n_paths,
        "real_data": bool(args.real_data),
        "step3_cold_ms": round(cold_s * 1e3, 3),
        "cold_phases_ms": {k: round(v * 1e3, 3)
                           for k, v in sorted(tc.acc.items())},
        "step3_wall_ms": round(wall / args.iters * 1e3, 3),
        "walks_per_sec": round(n_walks * args.iters / wall, 1),
        "phases_ms": {k: round(v / args.iters * 1e3, 3)
                      for k, v in sorted(t.acc.items())},
    }
    print(json.dumps(out, indent=1))
    if args.json:
        with open(args.json, "w") as f:
            json.dump(out, f, indent=1)


if __name__ == "__main__":
    main()
