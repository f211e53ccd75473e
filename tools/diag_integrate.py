#!/usr/bin/env python3
"""Micro-diagnosis of the slow integrate ops (bincount, masked index)."""
import time

import torch


def bench(name, fn, iters=20):
    fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    print(f"{name:40s} {(time.perf_counter()-t0)/iters*1e3:8.3f} ms",
          flush=True)


def main():
    dev = torch.device("cuda")
    G = 7523
    nnz = 1_500_000
    genes = torch.randint(0, G, (nnz,), dtype=torch.int32, device=dev)
    genes_l = genes.long()
    is_poor = torch.arange(nnz, device=dev) >= nnz // 2
    h = torch.randint(-2**62, 2**62, (75000,), dtype=torch.int64, device=dev)
    hs, _ = torch.sort(h)
    kept = torch.arange(75000, device=dev)

    bench("bincount int64 minlength", lambda: torch.bincount(genes_l, minlength=G))
    ones = torch.ones(nnz, dtype=torch.float32, device=dev)
    bench("scatter_add f32", lambda: torch.zeros(G, device=dev).scatter_add_(0, genes_l, ones))
    bench("index_add f32", lambda: torch.zeros(G, device=dev).index_add_(0, genes_l, ones))
    onesi = torch.ones(nnz, dtype=torch.int32, device=dev)
    bench("index_add i32", lambda: torch.zeros(G, dtype=torch.int32, device=dev).index_add_(0, genes_l, onesi))
    bench("histc f32", lambda: torch.histc(genes.float(), bins=G, min=0, max=G - 1))
    bench("masked bincount(2 of them)", lambda: (torch.bincount(genes_l[~is_poor], minlength=G),
                                                 torch.bincount(genes_l[is_poor], minlength=G)))

    def f_search():
        pos = torch.searchsorted(hs, hs).clamp_(max=hs.numel() - 1)
        return hs[pos] == hs

    bench("searchsorted 75k", f_search)
    m = f_search()
    bench("bool index kept[m]", lambda: kept[m])
    bench("nonzero(m)", lambda: torch.nonzero(m))
    bench("masked_select", lambda: torch.masked_select(kept, m))
    bench("sort 150k i64", lambda: torch.sort(torch.cat([h, h]), stable=True))
    P = 75000
    lens = torch.randint(1, 40, (P,), device=dev)
    bench("repeat_interleave 1.5M", lambda: torch.repeat_interleave(
        torch.arange(P, device=dev), lens))
    bench("item() sync", lambda: int(lens[-1].item()))


def skew():
    """Hub-skewed gene histogram: atomic contention check."""
    dev = torch.device("cuda")
    G = 7523
    nnz = 1_500_000
    g = (torch.randn(nnz, device=dev).abs() * (G / 20)).long().clamp_(max=G - 1)
    bench("bincount SKEWED int64", lambda: torch.bincount(g, minlength=G))
    onesi = torch.ones(nnz, dtype=torch.int32, device=dev)
    bench("index_add SKEWED i32", lambda: torch.zeros(
        G, dtype=torch.int32, device=dev).index_add_(0, g, onesi))


if __name__ == "__main__":
    main()
    skew()
