"""Literal-semantics port of the reference's step-3/5 algorithms
(SURVEY §4.1): the independent oracle the framework is A/B-tested
against. Unlike g2vec_amd.ops.cpu_ref (the framework's own CSR/seeded
algorithm), this module mirrors the REFERENCE's computation shape —
dense adjacency rows, np.random.choice sampling, Python-set dedup — so
tests prove reference equivalence, not merely self-consistency.

Semantics transcribed from (behavior, not code):
  compute_PCC / construct_adjMat   G2Vec.py:354-368 / :370-391
  generate_randomPath/generate_pathSet  G2Vec.py:324-352
  integrate_pathSet                G2Vec.py:310-322
  count_geneFreq                   G2Vec.py:288-308
  find_lgroups                     G2Vec.py:167-200 (incl. the shipped
      freqIdx==0 scalar-False bug, reproducible via compat_bug=True —
      SURVEY §2.9)
"""
from __future__ import annotations

from typing import Dict, List, Sequence, Set, Tuple

import numpy as np

Path = Tuple[int, ...]


# ---------------------------------------------------------------- step 2b/3
def lp_pcc(x: np.ndarray, y: np.ndarray) -> float:
    """Pearson correlation, 0 when either std is 0 (population std,
    matching np.ndarray.std() in the reference)."""
    sx, sy = x.std(), y.std()
    if not (sx > 0.0 and sy > 0.0):
        return 0.0
    zx = (x - x.mean()) / sx
    zy = (y - y.mean()) / sy
    return float((zx * zy).mean())


def lp_adj_matrix(edge_idx: np.ndarray, expr: np.ndarray,
                  labels: np.ndarray, group: int) -> np.ndarray:
    """Dense f32 [G,G] adjacency for one prognosis group: cell (src,dst)
    holds |PCC| over the group's samples iff > 0.5, else 0. Directed;
    repeated file edges overwrite the same cell."""
    G = expr.shape[1]
    sub = expr[labels == group]
    adj = np.zeros((G, G), dtype=np.float32)
    for s, d in edge_idx:
        w = abs(lp_pcc(sub[:, int(s)], sub[:, int(d)]))
        if w > 0.5:
            adj[int(s), int(d)] = w
    return adj


def lp_random_path(src: int, adj: np.ndarray, max_len: int,
                   rng: np.random.RandomState) -> Path:
    """One non-revisiting weighted walk over dense rows. Per step: copy
    the current row, zero every visited node (incl. current), renormalize,
    np.random.choice over ALL genes with that probability vector; dead
    end (all-zero row) stops. Returns the SORTED tuple of visited nodes —
    order and multiplicity discarded, a path is a gene set."""
    n = adj.shape[0]
    visited: List[int] = []
    node = src
    for _ in range(max_len):
        visited.append(node)
        row = adj[node].copy()
        row[visited] = 0.0
        tot = row.sum()
        if not tot > 0.0:
            break
        node = int(rng.choice(n, size=1, p=row / tot)[0])
    return tuple(sorted(visited))


def lp_pathset(adj: np.ndarray, max_len: int, iterations: int,
               rng: np.random.RandomState) -> Set[Path]:
    """iterations x every source gene; Python-set dedup of sorted tuples."""
    out: Set[Path] = set()
    n = adj.shape[0]
    for _ in range(iterations):
        for src in range(n):
            out.add(lp_random_path(src, adj, max_len, rng))
    return out


# ---------------------------------------------------------------- integrate
def lp_integrate(good: Set[Path], poor: Set[Path], n_genes: int
                 ) -> np.ndarray:
    """i32 [P, G+1] multi-hot rows + trailing label. Paths present in
    BOTH groups are dropped from both."""
    both = good & poor
    rows = []
    for label, pset in enumerate((good, poor)):
        for path in pset - both:
            r = np.zeros(n_genes + 1, dtype=np.int32)
            r[list(path)] = 1
            r[-1] = label
            rows.append(r)
    return (np.array(rows, dtype=np.int32) if rows
            else np.zeros((0, n_genes + 1), dtype=np.int32))


def lp_gene_freq(path_rows: np.ndarray,
                 gene_list: Sequence[str]) -> Dict[str, int]:
    """gene -> 0 (in more good paths) / 1 (more poor) / 2 (tie). Only
    genes that appear in at least one kept path are keyed."""
    gene_arr = np.asarray(gene_list)
    per_label: List[Dict[str, int]] = [dict(), dict()]
    seen: Set[str] = set()
    for row in path_rows:
        label = int(row[-1])
        for g in gene_arr[row[:-1] == 1]:
            per_label[label][g] = per_label[label].get(g, 0) + 1
            seen.add(g)
    out: Dict[str, int] = {}
    for g in seen:
        ng, np_ = per_label[0].get(g, 0), per_label[1].get(g, 0)
        out[g] = 0 if ng > np_ else (1 if ng < np_ else 2)
    return out


# ---------------------------------------------------------------- step 5
def lp_find_lgroups(mat: np.ndarray, gene_list: Sequence[str],
                    gene_freq: Dict[str, int],
                    compat_bug: bool = False) -> np.ndarray:
    """K-Means(k=3, random_state=0) over all embedding rows; largest
    cluster -> 2 ("other", first index wins ties); the two remaining
    clusters -> good(0)/poor(1) by their good-vs-poor path-frequency
    counts. compat_bug=True reproduces the shipped behavior where the
    frequency comparison is dead code (freqIdx==0 on a Python list is
    scalar False), so the counts are always 0-0 and the else branch
    assigns good = the LARGER remaining cluster index."""
    from sklearn.cluster import KMeans

    # n_init=10 pins the default the reference ran under (sklearn >= 0.18
    # in 2018); modern sklearn's n_init="auto" would run a single init
    km = KMeans(n_clusters=3, random_state=0, n_init=10).fit(mat)
    cl = km.labels_
    freq_per_gene = [gene_freq.get(g, 2) for g in gene_list]
    if not compat_bug:
        freq_per_gene = np.asarray(freq_per_gene)
    # largest cluster (first index wins ties)
    largest = 0
    size_largest = int(np.count_nonzero(cl == 0))
    for i in (1, 2):
        sz = int(np.count_nonzero(cl == i))
        if sz > size_largest:
            largest, size_largest = i, sz
    rest = [i for i in range(3) if i != largest]
    diff = np.zeros(3, dtype=np.float32)
    for i in rest:
        n_good = np.count_nonzero(np.logical_and(cl == i, freq_per_gene == 0))
        n_poor = np.count_nonzero(np.logical_and(cl == i, freq_per_gene == 1))
        diff[i] = n_good - n_poor
    if diff[rest[0]] > diff[rest[1]]:
        good_cl, poor_cl = rest[0], rest[1]
    else:
        good_cl, poor_cl = rest[1], rest[0]
    out = np.zeros(mat.shape[0], dtype=np.int32)
    out[cl == good_cl] = 0
    out[cl == poor_cl] = 1
    out[cl == largest] = 2
    return out


# ---------------------------------------------------------------- step 6
def lp_minmax(scores: np.ndarray, lo: float = 0.0,
              hi: float = 1.0) -> np.ndarray:
    mn, mx = scores.min(), scores.max()
    return (hi - lo) / (mx - mn) * (scores - mn) + lo


def lp_tstat(x: np.ndarray, y: np.ndarray) -> float:
    """Pooled-variance two-sample t (ddof=1), 0 on zero denominators."""
    import math
    sx, sy = x.std(ddof=1), y.std(ddof=1)
    nx, ny = len(x), len(y)
    d1 = math.sqrt(((nx - 1.0) * sx * sx + (ny - 1.0) * sy * sy)
                   / float(nx + ny - 2))
    d2 = math.sqrt(1.0 / nx + 1.0 / ny)
    if d1 > 0.0 and d2 > 0.0:
        return (x.mean() - y.mean()) / d1 / d2
    return 0.0


def lp_select_biomarkers(mat: np.ndarray, lgroups: np.ndarray,
                         expr: np.ndarray, labels: np.ndarray,
                         gene_list: Sequence[str],
                         num_biomarker: int) -> List[str]:
    """Main-inline step 6: per L-group (0 good then 1 poor), gene score =
    0.5*(minmax row-norm + minmax |pooled-t|); Python stable sort
    DESCENDING by score, top-N genes sorted by symbol; the two lists
    concatenate and the union sorts again."""
    from operator import itemgetter

    gene_arr = np.asarray(gene_list)
    out: List[str] = []
    for grp in (0, 1):
        sub_mat = mat[lgroups == grp]
        sub_genes = gene_arr[lgroups == grp]
        sub_expr = expr[:, lgroups == grp]
        d = lp_minmax(np.linalg.norm(sub_mat, axis=1), 0.0, 1.0)
        t = np.zeros(sub_expr.shape[1], dtype=np.float32)
        for i in range(sub_expr.shape[1]):
            t[i] = abs(lp_tstat(sub_expr[labels == 0, i],
                                sub_expr[labels == 1, i]))
        score = 0.5 * (d + lp_minmax(t, 0.0, 1.0))
        pairs = sorted(zip(sub_genes.tolist(), score.tolist()),
                       key=itemgetter(1), reverse=True)
        out += sorted(g for g, _s in pairs[:num_biomarker])
    return sorted(out)
