"""L-group discovery (reference step 5, G2Vec.py:167-200).

KMeans (k=3, random_state=0 — the only seeded RNG in the reference) over
the gene embeddings; the largest cluster becomes "other" (2); the two
remaining clusters are labelled good (0) / poor (1).

The reference's good/poor disambiguation is DEAD CODE as shipped
(`freqIdx` is a Python list, so `freqIdx==0` is scalar False
and the frequency counts are always 0 — the `else` branch at
G2Vec.py:192-194 always runs, picking good = the larger remaining cluster
index). This module implements the INTENDED semantics by default and
reproduces the shipped behaviour under compat_lgroup_bug=True.
"""
from __future__ import annotations

import numpy as np


def kmeans_torch(X, k: int, seed: int = 0, iters: int = 300,
                 tol: float = 1e-6) -> np.ndarray:
    """Seeded Lloyd k-means with k-means++ init on torch tensors (runs on
    GPU when X is a CUDA tensor). The scalable backend for L-group
    clustering at 200k-1M genes, where sklearn's host KMeans would take
    hours; sklearn (random_state=0, reference parity) stays the default at
    bundled-dataset scale."""
    import torch

    n = X.shape[0]
    gen = torch.Generator().manual_seed(seed)
    idx = [int(torch.randint(n, (1,), generator=gen))]
    for _ in range(k - 1):
        d2 = torch.cdist(X, X[idx]).min(dim=1).values.clamp_(min=0) ** 2
        tot = float(d2.sum())
        probs = (d2 / tot).cpu() if tot > 0 else torch.full((n,), 1.0 / n)
        idx.append(int(torch.multinomial(probs, 1, generator=gen)))
    C = X[idx].clone()
    assign = torch.zeros(n, dtype=torch.long, device=X.device)
    for _ in range(iters):
        assign = torch.cdist(X, C).argmin(dim=1)
        newC = C.clone()
        for j in range(k):
            sel = assign == j
            if bool(sel.any()):
                newC[j] = X[sel].mean(dim=0)
        shift = float((newC - C).norm())
        C = newC
        if shift < tol:
            break
    return assign.cpu().numpy()


def find_lgroups(embeddings: np.ndarray, gene_freq: np.ndarray,
                 compat_lgroup_bug: bool = False, backend: str = "auto",
                 device=None) -> np.ndarray:
    """embeddings: f32 [G, h]; gene_freq: int [G] in {0 good,1 poor,2 other}.
    Returns int32 [G] with 0 good / 1 poor / 2 other.
    backend: 'sklearn' (reference parity: KMeans random_state=0) |
    'torch' (scalable, GPU-capable) | 'auto' (sklearn up to 50k genes)."""
    if backend == "auto":
        backend = "sklearn" if embeddings.shape[0] <= 50_000 else "torch"
    if backend == "torch":
        import torch
        X = torch.from_numpy(np.ascontiguousarray(embeddings))
        if device is not None:
            X = X.to(device)
        km_idx = kmeans_torch(X, 3, seed=0)
    else:
        from sklearn.cluster import KMeans
        km = KMeans(n_clusters=3, random_state=0, n_init=10).fit(embeddings)
        km_idx = km.labels_

    # largest cluster -> 2 (strict > keeps the lowest index on ties,
    # G2Vec.py:174-180)
    sizes = [int(np.count_nonzero(km_idx == i)) for i in range(3)]
    largest = 0
    for i in (1, 2):
        if sizes[i] > sizes[largest]:
            largest = i
    remaining = [i for i in range(3) if i != largest]

    if compat_lgroup_bug:
        # shipped behaviour: gpDiff stays all-zero -> else branch ->
        # good = larger remaining index (G2Vec.py:192-194)
        good, poor = remaining[1], remaining[0]
    else:
        gp = np.zeros(3, dtype=np.float32)
        for i in remaining:
            n_good = int(np.count_nonzero((km_idx == i) & (gene_freq == 0)))
            n_poor = int(np.count_nonzero((km_idx == i) & (gene_freq == 1)))
            gp[i] = n_good - n_poor
        if gp[remaining[0]] > gp[remaining[1]]:
            good, poor = remaining[0], remaining[1]
        else:
            good, poor = remaining[1], remaining[0]

    out = np.zeros(embeddings.shape[0], dtype=np.int32)
    out[km_idx == good] = 0
    out[km_idx == poor] = 1
    out[km_idx == largest] = 2
    return out
