"""Random-path generation driver (reference step 3, G2Vec.py:324-352 —
"the most time consuming step").

On GPU the walks run in the CSR biased-random-walk HIP kernel (one
64-lane wavefront per walk, LDS visited list, counter-based device RNG);
on CPU the `ops.cpu_ref` oracle runs the same algorithm. Sources can be
sharded for data-parallel generation (each rank walks a slice of the
source genes for all repetitions).
"""
from __future__ import annotations

from typing import NamedTuple, Tuple

import torch

from . import ops
from .graph import CsrGraph


class WalkSet(NamedTuple):
    nodes: torch.Tensor     # i32 [n_walks, len_path], -1 padded
    lengths: torch.Tensor   # i32 [n_walks]
    hashes: torch.Tensor    # i64 [n_walks] order-independent path-set hash


def generate_walks(graph: CsrGraph, len_path: int, num_repetition: int,
                   seed: int, group: int,
                   src_range: Tuple[int, int] = None) -> WalkSet:
    """Walks from every source gene in [src_lo, src_hi) x num_repetition.
    seed is mixed with the group so the two groups draw independent streams."""
    lo, hi = src_range if src_range is not None else (0, graph.n_nodes)
    device = graph.row_ptr.device
    sources = torch.arange(lo, hi, dtype=torch.int32, device=device)
    mixed_seed = (seed * 0x9E3779B1 + (group + 1) * 0x85EBCA77) & 0x7FFFFFFFFFFFFFFF
    nodes, lengths, hashes = ops.random_walks(
        graph.row_ptr, graph.col_idx, graph.weights, sources,
        num_repetition, len_path, mixed_seed)
    return WalkSet(nodes, lengths, hashes)
