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


def _unique_first(hashes: torch.Tensor) -> Tuple[torch.Tensor, torch.Tensor]:
    """(indices of one representative per distinct hash, their SORTED hash
    values). Deterministic: the representative with the smallest original
    index (stable sort keeps original order within equal hashes)."""
    sh, perm = torch.sort(hashes, stable=True)
    first = torch.ones_like(sh, dtype=torch.bool)
    if sh.numel() > 1:
        first[1:] = sh[1:] != sh[:-1]
    return perm[first], sh[first]


def _isin_sorted(a_sorted: torch.Tensor, b_sorted: torch.Tensor) -> torch.Tensor:
    """bool mask: a_sorted[i] present in b_sorted. Both inputs sorted —
    one binary search instead of torch.isin's internal sort."""
    if b_sorted.numel() == 0:
        return torch.zeros_like(a_sorted, dtype=torch.bool)
    pos = torch.searchsorted(b_sorted, a_sorted)
    pos = pos.clamp_(max=b_sorted.numel() - 1)
    return b_sorted[pos] == a_sorted


class _NullTimer:
    def __call__(self, name):
        return self

    def __enter__(self):
        return self

    def __exit__(self, *a):
        return False


def integrate_pathsets(good: WalkSet, poor: WalkSet, n_genes: int,
                       timers=None) -> Tuple[PathSet, torch.Tensor, int]:
    """Returns (pathset, gene_freq i64 [G] with values {0,1,2}, n_genes_in_paths).

    Device-friendly schedule (step-3 hot path — this function was 23 ms
    of the 28 ms warmed step-3 wall in round 1): per-group dedup keeps
    its hashes sorted so cross-group common-path removal is a binary
    search (no second sort), and the kept walks' gene lists are packed
    into CSR with a flat gather over exactly the kept instances instead
    of a boolean mask over the whole padded [n_walks, len_path] buffer
    (12M elements at ex_* scale)."""
    t = timers or _NullTimer()
    device = good.nodes.device
    kept, kept_sorted_h = [], []
    with t("integ.dedup_sort"):
        for ws in (good, poor):
            idx, sh = _unique_first(ws.hashes)
            kept.append(idx)
            kept_sorted_h.append(sh)
    with t("integ.common_search"):
        keep_g = kept[0][~_isin_sorted(kept_sorted_h[0], kept_sorted_h[1])]
        keep_p = kept[1][~_isin_sorted(kept_sorted_h[1], kept_sorted_h[0])]
        n_g, n_p = int(keep_g.numel()), int(keep_p.numel())
    P = n_g + n_p
    if P == 0:
        empty = torch.zeros(0, dtype=torch.int32, device=device)
        return (PathSet(empty, torch.zeros(1, dtype=torch.int32, device=device),
                        torch.zeros(0, dtype=torch.float32, device=device), n_genes),
                torch.full((n_genes,), 2, dtype=torch.int64, device=device), 0)

    if good.nodes.shape[1] != poor.nodes.shape[1]:
        raise ValueError("good/poor walk sets must share len_path")
    L = int(good.nodes.shape[1])
    with t("integ.offsets"):
        lengths = torch.cat([good.lengths[keep_g], poor.lengths[keep_p]]).long()
        labels = torch.zeros(P, dtype=torch.float32, device=device)
        labels[n_g:] = 1.0
        offsets = torch.zeros(P + 1, dtype=torch.int64, device=device)
        torch.cumsum(lengths, 0, out=offsets[1:])
        nnz = int(offsets[-1].item())
        _check_i32_nnz(nnz)

    # flat gather of exactly the kept instances: row starts in the padded
    # buffers, one arange re-based per path
    with t("integ.seg"):
        seg = torch.repeat_interleave(torch.arange(P, device=device), lengths)
        pos = torch.arange(nnz, device=device) - offsets[seg]
    with t("integ.gather"):
        row_start = torch.cat([keep_g, keep_p + good.nodes.shape[0]]) * L
        flat = torch.cat([good.nodes.reshape(-1), poor.nodes.reshape(-1)])
        genes = flat[row_start[seg] + pos].int()

    # gene frequencies (paths are sets: non-revisiting walks never repeat
    # a gene, so per-path counts are 0/1)
    with t("integ.freq"):
        is_poor = seg >= n_g
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
