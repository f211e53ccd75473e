"""Microbench: sensitivity of the fast-path gather kernels to gene-id
locality at the 1M-gene scale.

Each path's genes are drawn either uniformly over [0, G) ("uniform" — the
worst case) or from a W-wide window at a path-dependent offset
("win<W>" — what a co-occurrence-based gene relabeling would produce).
Times cbow_fwd_scalar (s-gather), cbow_eval_counts (same gather + counts)
and scatter_dO (dO gathered by PATH id — expected insensitive to gene
locality, it is the control)."""
import sys
import time

import torch

sys.path.insert(0, ".")
from g2vec_amd import ops  # noqa: E402

G, P, L = 1_000_000, 4_400_000, 21
dev = torch.device("cuda")
ITERS = 30


def make_ps(mode):
    g = torch.Generator(device="cpu").manual_seed(0)
    if mode == "uniform":
        genes = torch.randint(0, G, (P * L,), generator=g, dtype=torch.int32)
    else:
        W = int(mode[3:])
        starts = torch.randint(0, G - W, (P,), generator=g)
        off = torch.randint(0, W, (P, L), generator=g)
        genes = (starts[:, None] + off).reshape(-1).to(torch.int32)
    offsets = torch.arange(0, (P + 1) * L, L, dtype=torch.int32)
    labels = (torch.arange(P) % 2).float()
    return genes.to(dev), offsets.to(dev), labels.to(dev)


def t(fn):
    for _ in range(3):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(ITERS):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / ITERS * 1e3


s = torch.randn(G, device=dev)
counts = torch.zeros(2, device=dev)
for mode in ("uniform", "win65536", "win4096", "win256"):
    genes, offsets, labels = make_ps(mode)
    plan = ops.build_scatter_plan(genes, offsets, G)
    _l, _c, dO = ops.cbow_fwd_scalar(s, genes, offsets, labels, 1.0 / P, True)
    ms_fwd = t(lambda: ops.cbow_fwd_scalar(s, genes, offsets, labels,
                                           1.0 / P, True))
    ms_ev = t(lambda: ops.cbow_eval_counts_(s, genes, offsets, labels,
                                            P // 2, counts))
    ms_sc = t(lambda: ops.scatter_dO(genes, offsets, dO, G, plan=plan))
    print(f"{mode:>9}: fwd {ms_fwd:7.3f} ms  eval {ms_ev:7.3f} ms  "
          f"scatter {ms_sc:7.3f} ms", flush=True)
