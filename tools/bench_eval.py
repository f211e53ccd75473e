#!/usr/bin/env python3
"""A/B microbench of the fused eval kernels (subwave vs scan vs scan+LDS)."""
import os
import sys
import time

import numpy as np
import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from g2vec_amd import ops  # noqa: E402


def main():
    dev = torch.device("cuda")
    rng = np.random.default_rng(0)
    G = int(sys.argv[1]) if len(sys.argv) > 1 else 7523
    P = int(sys.argv[2]) if len(sys.argv) > 2 else 75000
    lo = int(os.environ.get('LMIN', 2)); hi = int(os.environ.get('LMAX', 40))
    lens = rng.integers(lo, hi, size=P)
    genes = np.concatenate([rng.integers(0, G, size=n) for n in lens])
    offs = np.concatenate([[0], np.cumsum(lens)])
    labels = rng.integers(0, 2, size=P).astype(np.float32)
    s = torch.randn(G, device=dev) * 0.2
    g_t = torch.from_numpy(genes.astype(np.int32)).to(dev)
    o_t = torch.from_numpy(offs.astype(np.int32)).to(dev)
    l_t = torch.from_numpy(labels).to(dev)
    p_split = P * 4 // 5
    inv_b = 1.0 / p_split
    lens_t = (o_t[1:] - o_t[:-1]).long()
    pathid = torch.repeat_interleave(
        torch.arange(P, dtype=torch.int32, device=dev), lens_t)
    cap = (hi - 1) // 64 + 2
    piece = torch.empty(P * cap, dtype=torch.float32, device=dev)
    counts = torch.zeros(2, device=dev)
    dO = torch.zeros(p_split, device=dev)

    def t(name, fn, iters=300):
        fn()
        torch.cuda.synchronize()
        t0 = time.perf_counter()
        for _ in range(iters):
            fn()
        torch.cuda.synchronize()
        print(f"{name:24s} {(time.perf_counter()-t0)/iters*1e6:9.2f} us "
              f"counts={counts.tolist()}", flush=True)

    def run_counts():
        counts.zero_()
        ops.native().cbow_eval_counts_(s, g_t, o_t, l_t, p_split, counts,
                                       dO=dO, inv_b=inv_b)

    os.environ.pop("G2VEC_EVAL_LDS", None)      # default: LDS variant off
    t("subwave", run_counts)
    base_counts = counts.tolist()
    base_dO = dO.clone()
    t("scan", lambda: (counts.zero_(), ops.native().cbow_eval_scan_(
        s, g_t, pathid, o_t, l_t, p_split, cap, piece, counts,
        dO=dO, inv_b=inv_b)))
    for grid in (512, 768, 1024, 1536, 2048):
        os.environ["G2VEC_EVAL_LDS"] = str(256 * 1024)
        os.environ["G2VEC_EVAL_LDS_GRID"] = str(grid)
        dO.zero_()
        t(f"subwave+lds g{grid}", run_counts)
        assert counts.tolist() == base_counts, (counts.tolist(), base_counts)
        assert torch.equal(dO, base_dO), "LDS variant must be bitwise-equal"
    os.environ.pop("G2VEC_EVAL_LDS", None)


if __name__ == "__main__":
    main()
