"""Path-set integration: dedup, cross-group common-path removal,
gene frequencies, and the sparse path tensor for training.

Reference semantics:
  - a path is an unordered gene SET (tuple(sorted(path)), G2Vec.py:345);
    duplicates within a group collapse (set, G2Vec.py:348-352)
  - paths present in BOTH groups are dropped entirely (G2Vec.py:313-315)
  - per-gene frequency label: 0 if the gene appears in more good paths,
    1 if more poor, 2 on ties (G2Vec.py:288-308)

Dedup here is by a 64-bit order-independent hash of the gene set
(sum of per-gene splitmix64 mixes) — collision probability ~1e-9 at the
150k-walk scale. The path tensor stays in sparse index form
(<= len_path i32 per path) instead of the reference's dense multi-hot
rows, removing the O(P*G) memory axis entirely.
"""
from __future__ import annotations

from typing import NamedTuple, Tuple

import torch

from .walks import WalkSet


def _check_i32_nnz(nnz: int) -> None:
    """The HIP kernels index flat gene instances with int32: a PathSet
    whose total nnz reaches 2^31 would silently wrap in the .int() casts
    below (e.g. 1M genes x lenPath 512 x 10 reps x 2 groups). Fail loudly
    instead."""
    if nnz >= 2 ** 31:
        raise OverflowError(
            f"path set has {nnz} gene instances — the int32 kernel index "
            f"space holds < 2^31. Reduce len_path/num_repetition or shard "
            f"walk generation across more ranks.")


class PathSet(NamedTuple):
    genes: torch.Tensor     # i32 [nnz]   flat gene indices
    offsets: torch.Tensor   # i32 [P+1]
    labels: torch.Tensor    # f32 [P]     0 good / 1 poor
    n_genes: int

    @property
    def n_paths(self) -> int:
        return int(self.labels.shape[0])


def _unique_first(hashes: torch.Tensor) -> torch.Tensor:
    """Indices of one representative per distinct hash (deterministic:
    the representative with the smallest original index)."""
    sh, perm = torch.sort(hashes, stable=True)
    first = torch.ones_like(sh, dtype=torch.bool)
    if sh.numel() > 1:
        first[1:] = sh[1:] != sh[:-1]
    # within equal hashes stable sort keeps original order -> first is min-index
    return perm[first]


def integrate_pathsets(good: WalkSet, poor: WalkSet, n_genes: int
                       ) -> Tuple[PathSet, torch.Tensor, int]:
    """Returns (pathset, gene_freq i64 [G] with values {0,1,2}, n_genes_in_paths)."""
    device = good.nodes.device
    kept = []
    kept_hashes = []
    for ws in (good, poor):
        idx = _unique_first(ws.hashes)
        kept.append(idx)
        kept_hashes.append(ws.hashes[idx])
    common_g = torch.isin(kept_hashes[0], kept_hashes[1])
    common_p = torch.isin(kept_hashes[1], kept_hashes[0])
    keep_g = kept[0][~common_g]
    keep_p = kept[1][~common_p]

    parts_nodes, parts_len, parts_lab = [], [], []
    for ws, keep, lab in ((good, keep_g, 0.0), (poor, keep_p, 1.0)):
        if keep.numel() == 0:
            continue
        parts_nodes.append(ws.nodes[keep])
        parts_len.append(ws.lengths[keep])
        parts_lab.append(torch.full((keep.numel(),), lab, dtype=torch.float32,
                                    device=device))
    if not parts_nodes:
        empty = torch.zeros(0, dtype=torch.int32, device=device)
        return (PathSet(empty, torch.zeros(1, dtype=torch.int32, device=device),
                        torch.zeros(0, dtype=torch.float32, device=device), n_genes),
                torch.full((n_genes,), 2, dtype=torch.int64, device=device), 0)

    nodes = torch.cat(parts_nodes)           # [P, L]
    lengths = torch.cat(parts_len).long()    # [P]
    labels = torch.cat(parts_lab)

    mask = nodes >= 0
    genes = nodes[mask].int()
    _check_i32_nnz(int(genes.numel()))
    offsets = torch.zeros(len(lengths) + 1, dtype=torch.int64, device=device)
    torch.cumsum(lengths, 0, out=offsets[1:])

    # gene frequencies (paths are sets: non-revisiting walks never repeat a gene)
    seg = torch.repeat_interleave(torch.arange(len(lengths), device=device), lengths)
    is_poor = labels[seg] > 0.5
    cnt_g = torch.bincount(genes[~is_poor].long(), minlength=n_genes)
    cnt_p = torch.bincount(genes[is_poor].long(), minlength=n_genes)
    freq = torch.full((n_genes,), 2, dtype=torch.int64, device=device)
    freq[cnt_g > cnt_p] = 0
    freq[cnt_p > cnt_g] = 1
    n_in_paths = int(((cnt_g + cnt_p) > 0).sum().item())

    ps = PathSet(genes.contiguous(), offsets.int().contiguous(),
                 labels.contiguous(), n_genes)
    return ps, freq, n_in_paths


def subset(ps: PathSet, idx: torch.Tensor) -> PathSet:
    """Re-index a PathSet by path indices (device-side CSR slicing)."""
    offs = ps.offsets.long()
    lens = (offs[1:] - offs[:-1])[idx]
    new_off = torch.zeros(len(idx) + 1, dtype=torch.int64, device=idx.device)
    torch.cumsum(lens, 0, out=new_off[1:])
    _check_i32_nnz(int(new_off[-1].item()))
    starts = offs[idx]
    # gather flat gene ranges
    seg = torch.repeat_interleave(torch.arange(len(idx), device=idx.device), lens)
    pos = torch.arange(seg.numel(), device=idx.device) - new_off[seg]
    src = starts[seg] + pos
    return PathSet(ps.genes[src].contiguous(), new_off.int().contiguous(),
                   ps.labels[idx].contiguous(), ps.n_genes)
