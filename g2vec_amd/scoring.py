"""Biomarker scoring (reference step 6, G2Vec.py:85-157). Host-side NumPy
by design (steps 5-6 stay on host — [G]-sized vectors)."""
from __future__ import annotations

from typing import List, Sequence

import numpy as np


def transform_minmax(scores: np.ndarray, lo: float = 0.0, hi: float = 1.0
                     ) -> np.ndarray:
    """Min-max rescale (G2Vec.py:133-136). Guard: constant input -> zeros
    (the reference divides by zero there)."""
    old_min, old_max = float(scores.min()), float(scores.max())
    if old_max <= old_min:
        return np.zeros_like(scores)
    return (hi - lo) / (old_max - old_min) * (scores - old_min) + lo


def t_statistic(x: np.ndarray, y: np.ndarray) -> float:
    """Pooled-variance two-sample t (ddof=1 stds; 0 on zero denominators),
    G2Vec.py:138-149."""
    nx, ny = len(x), len(y)
    if nx < 2 or ny < 2:
        return 0.0
    sx, sy = x.std(ddof=1), y.std(ddof=1)
    d1 = np.sqrt(((nx - 1.0) * sx * sx + (ny - 1.0) * sy * sy) / (nx + ny - 2.0))
    d2 = np.sqrt(1.0 / nx + 1.0 / ny)
    if d1 > 0.0 and d2 > 0.0:
        return float((x.mean() - y.mean()) / d1 / d2)
    return 0.0


def t_scores(expr: np.ndarray, labels: np.ndarray) -> np.ndarray:
    """|t| per gene column, good (label 0) vs poor (label 1) samples
    (G2Vec.py:151-157). Vectorized over genes; `t_statistic` is the scalar
    oracle it is tested against."""
    good = expr[labels == 0].astype(np.float64)
    poor = expr[labels == 1].astype(np.float64)
    nx, ny = good.shape[0], poor.shape[0]
    if nx < 2 or ny < 2:
        return np.zeros(expr.shape[1], dtype=np.float32)
    sx = good.std(axis=0, ddof=1)
    sy = poor.std(axis=0, ddof=1)
    d1 = np.sqrt(((nx - 1.0) * sx * sx + (ny - 1.0) * sy * sy) / (nx + ny - 2.0))
    d2 = np.sqrt(1.0 / nx + 1.0 / ny)
    diff = good.mean(axis=0) - poor.mean(axis=0)
    with np.errstate(divide="ignore", invalid="ignore"):
        t = np.where(d1 > 0.0, diff / np.where(d1 > 0, d1, 1.0) / d2, 0.0)
    return np.abs(t).astype(np.float32)


def select_biomarkers(embeddings: np.ndarray, lgroup_idx: np.ndarray,
                      expr: np.ndarray, labels: np.ndarray,
                      genes: Sequence[str], num_biomarker: int) -> List[str]:
    """Top-N per L-group by gene score = 0.5*(minmax d-score + minmax t-score),
    union sorted (G2Vec.py:85-109)."""
    genes = np.asarray(genes)
    result: List[str] = []
    for grp in (0, 1):      # 0 good, 1 poor
        sel = lgroup_idx == grp
        if not sel.any():
            continue
        mat = embeddings[sel]
        sub_genes = genes[sel]
        sub_expr = expr[:, sel]
        d = transform_minmax(np.linalg.norm(mat, axis=1))
        t = transform_minmax(t_scores(sub_expr, labels))
        score = 0.5 * (d + t)
        # stable sort desc by score (reference: Python sorted, stable)
        order = np.argsort(-score, kind="stable")[:num_biomarker]
        result += sorted(sub_genes[order].tolist())
    return sorted(result)
