"""Preprocessing: label matching, gene-set intersection, restriction.

Reference semantics (all host-side):
  match_labels        G2Vec.py:428-434 (raises on any missing sample)
  find_common_genes   G2Vec.py:420-426 (sorted A->Z — defines the global gene index)
  restrict_network    G2Vec.py:393-402
  restrict_data       G2Vec.py:404-412
"""
from __future__ import annotations

from typing import Dict, List

import numpy as np


class SampleMismatchError(RuntimeError):
    pass


def match_labels(clinical: Dict[str, int], samples: np.ndarray) -> np.ndarray:
    try:
        labels = np.array([clinical[s] for s in samples], dtype=np.int64)
    except KeyError as e:
        raise SampleMismatchError(
            f"sample {e.args[0]!r} present in expression data but missing from "
            f"clinical data (reference exits(1) here, G2Vec.py:432-433)") from e
    bad = np.flatnonzero((labels != 0) & (labels != 1))
    if bad.size:
        # the reference would silently mis-group any other value (its
        # label==0 / label==1 masks both miss it); fail loud instead
        raise ValueError(
            f"clinical labels must be 0 (good) or 1 (poor): sample "
            f"{samples[bad[0]]!r} has label {int(labels[bad[0]])}"
            + (f" (+{bad.size - 1} more)" if bad.size > 1 else ""))
    return labels


def find_common_genes(network_genes, expr_genes) -> List[str]:
    """Sorted intersection — the sort (G2Vec.py:425) fixes gene indexing for
    every downstream stage and for output-file row order."""
    return sorted(set(network_genes) & set(expr_genes))


def restrict_network(network: dict, common_genes: List[str]) -> dict:
    common = set(common_genes)
    edges = [(s, d) for (s, d) in network["edge"] if s in common and d in common]
    return {"edge": edges, "gene": common}


def restrict_data(data: dict, common_genes: List[str]) -> dict:
    gene2idx = {g: i for i, g in enumerate(data["gene"])}
    idx = np.array([gene2idx[g] for g in common_genes], dtype=np.int64)
    return {
        "sample": data["sample"].copy(),
        "label": data["label"].copy(),
        "expr": np.ascontiguousarray(data["expr"][:, idx]),
        "gene": np.array(common_genes),
    }


def edges_to_indices(edges, common_genes: List[str]) -> np.ndarray:
    """Edge list -> int32 [E, 2] index pairs over the sorted common gene list."""
    gene2idx = {g: i for i, g in enumerate(common_genes)}
    if not edges:
        return np.zeros((0, 2), dtype=np.int32)
    out = np.empty((len(edges), 2), dtype=np.int32)
    for k, (s, d) in enumerate(edges):
        out[k, 0] = gene2idx[s]
        out[k, 1] = gene2idx[d]
    return out
