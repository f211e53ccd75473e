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


def find_lgroups(embeddings: np.ndarray, gene_freq: np.ndarray,
                 compat_lgroup_bug: bool = False) -> np.ndarray:
    """embeddings: f32 [G, h]; gene_freq: int [G] in {0 good,1 poor,2 other}.
    Returns int32 [G] with 0 good / 1 poor / 2 other."""
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
