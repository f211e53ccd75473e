"""Device-op dispatch layer.

CUDA/HIP tensors -> the in-tree native extension `g2vec_amd._C`
(hand-written gfx950 HIP kernels; FAILS LOUDLY if the extension is not
built — there is deliberately no silent eager fallback on a GPU box).
CPU tensors -> the torch/numpy oracles in `cpu_ref`.
"""
from __future__ import annotations

from typing import NamedTuple, Optional

import torch

from . import cpu_ref

_NATIVE = None
_NATIVE_ERR: Optional[BaseException] = None
try:  # pragma: no cover - exercised only when the extension is built
    from g2vec_amd import _C as _NATIVE  # type: ignore
except Exception as e:  # noqa: BLE001
    _NATIVE_ERR = e


def native_available() -> bool:
    return _NATIVE is not None


def native():
    if _NATIVE is None:
        raise RuntimeError(
            "g2vec_amd._C HIP extension is not importable on this host — the GPU "
            "compute path refuses to fall back to eager PyTorch. Build it in-tree "
            "with `python setup.py build_ext --inplace` (PYTORCH_ROCM_ARCH=gfx950). "
            f"Import error was: {_NATIVE_ERR!r}")
    return _NATIVE


# ------------------------------------------------------------------ walks
def random_walks(row_ptr, col_idx, weights, sources, num_repetition: int,
                 len_path: int, seed: int):
    if row_ptr.is_cuda:
        return native().random_walks(row_ptr, col_idx, weights, sources,
                                     num_repetition, len_path, seed)
    return cpu_ref.random_walks(row_ptr, col_idx, weights, sources,
                                num_repetition, len_path, seed)


# ------------------------------------------------------------------ CBOW fast (scalar) path
def cbow_fwd_scalar(s, genes, offsets, labels, inv_b: float, want_grad: bool):
    if s.is_cuda:
        loss, correct, dO = native().cbow_fwd_scalar(
            s, genes, offsets, labels, float(inv_b), bool(want_grad))
        return loss, correct, (dO if want_grad else None)
    return cpu_ref.cbow_fwd_scalar(s, genes, offsets, labels, inv_b, want_grad)


def cbow_eval_counts_(s, genes, offsets, labels, p_split: int,
                      counts: torch.Tensor, dO: Optional[torch.Tensor] = None,
                      inv_b: float = 1.0, scan=None) -> None:
    """Accumulate the concatenated train+val correct counts into counts[2]
    (caller zeroes it). One fused kernel on GPU; oracle math on CPU.

    dO given: also emit the train split's (p < p_split) dlogit into dO with
    scale inv_b — the fused next-epoch forward (this eval's s is the next
    epoch's pre-update s), numerically identical to a separate
    cbow_fwd_scalar over the train paths (fp32 summation order differs
    between the kernel variants).

    scan = (pathid i32[nnz], piece f32[>=P*cap], cap): persistent buffers
    enabling the instance-parallel segmented-scan variant — the hot
    steady-state eval (see eval_scan_kernel)."""
    if s.is_cuda:
        if scan is not None:
            pathid, piece, cap = scan
            native().cbow_eval_scan_(s, genes, pathid, offsets, labels,
                                     int(p_split), int(cap), piece, counts,
                                     dO=dO, inv_b=float(inv_b))
            return
        native().cbow_eval_counts_(s, genes, offsets, labels, int(p_split),
                                   counts, dO=dO, inv_b=float(inv_b))
        return
    _l, corr, d = cpu_ref.cbow_fwd_scalar(s, genes, offsets, labels, inv_b,
                                          dO is not None)
    counts[0] += corr[:p_split].sum()
    counts[1] += corr[p_split:].sum()
    if dO is not None:
        dO.copy_(d[:p_split])


class ScatterPlan(NamedTuple):
    """Precomputed gene-sorted instance layout for the deterministic
    c = X^T dO reduction (built once per path set; genes never change
    across epochs).

    When the dO table outgrows one XCD's 4 MB L2 (large path counts), the
    instances are sorted by (path-slab, gene) instead — slab_seg_ptr marks
    each ~3 MB slab's segment range and scatter_dO launches the reduce one
    slab at a time, so every XCD gathers from an L2-resident dO slice.
    Accumulation order (ascending slab) is fixed: still deterministic."""
    inst_path: torch.Tensor   # i32 [nnz]  path id of each instance, gene-sorted
    seg_start: torch.Tensor   # i32 [n_seg+1]
    seg_gene: torch.Tensor    # i32 [n_seg]
    slab_seg_ptr: Optional[list] = None   # [n_slabs+1] segment-range bounds


SLAB_BYTES = 3 << 20     # dO slice per launch; < the 4 MB per-XCD L2


def build_scatter_plan(genes: torch.Tensor, offsets: torch.Tensor,
                       n_genes: int, _force_slabs: bool = False
                       ) -> ScatterPlan:
    # _force_slabs: CPU tests exercise the slab geometry (the CPU compute
    # path itself never consumes a slabbed plan)
    counts = (offsets[1:] - offsets[:-1]).long()
    P = len(counts)
    path_of = torch.repeat_interleave(
        torch.arange(P, device=genes.device), counts)
    slab_paths = SLAB_BYTES // 4
    n_slabs = ((P + slab_paths - 1) // slab_paths
               if (genes.is_cuda or _force_slabs) else 1)
    if n_slabs <= 1:
        sorted_key, perm = torch.sort(genes.long(), stable=True)
    else:
        key = (path_of // slab_paths) * n_genes + genes.long()
        sorted_key, perm = torch.sort(key, stable=True)
    inst_path = path_of[perm].int().contiguous()
    seg_key, seg_counts = torch.unique_consecutive(sorted_key,
                                                   return_counts=True)
    seg_start = torch.zeros(len(seg_key) + 1, dtype=torch.int64,
                            device=genes.device)
    torch.cumsum(seg_counts, 0, out=seg_start[1:])
    if n_slabs <= 1:
        return ScatterPlan(inst_path, seg_start.int().contiguous(),
                           seg_key.int().contiguous())
    seg_slab = seg_key // n_genes
    seg_gene = (seg_key - seg_slab * n_genes).int().contiguous()
    ptr = torch.searchsorted(
        seg_slab, torch.arange(n_slabs + 1, device=genes.device)).cpu()
    return ScatterPlan(inst_path, seg_start.int().contiguous(), seg_gene,
                       [int(x) for x in ptr])


def scatter_dO(genes, offsets, dO, n_genes: int,
               plan: Optional[ScatterPlan] = None):
    if dO.is_cuda:
        if plan is None:
            plan = build_scatter_plan(genes, offsets, n_genes)
        if plan.slab_seg_ptr is None:
            return native().scatter_dO_det(plan.inst_path, plan.seg_start,
                                           plan.seg_gene, dO, int(n_genes))
        c = torch.zeros(n_genes, dtype=torch.float32, device=dO.device)
        ptr = plan.slab_seg_ptr
        for a, b in zip(ptr[:-1], ptr[1:]):     # ascending slabs: one
            if b > a:                           # L2-resident slice at a time
                native().scatter_dO_det_(plan.inst_path,
                                         plan.seg_start[a:b + 1],
                                         plan.seg_gene[a:b], dO, c)
        return c
    return cpu_ref.scatter_dO(genes, offsets, dO, n_genes)


def trunc_normal_(out: torch.Tensor, std: float, seed: int) -> None:
    """Seeded +-2sigma truncated-normal fill (K9, reference init
    semantics tf.truncated_normal G2Vec.py:234-235). Device kernel on
    GPU — no host-side multi-GB rejection loop at the 1M x 512 config."""
    if out.is_cuda:
        native().trunc_normal_(out, float(std), int(seed))
        return
    cpu_ref.trunc_normal_(out, std, seed)


def tf1_lr_t(lr: float, b1: float, b2: float, t: int) -> float:
    """TF1 AdamOptimizer effective step size (G2Vec.py:245-246 semantics)."""
    return lr * (1.0 - b2 ** t) ** 0.5 / (1.0 - b1 ** t)


def adam_rank1(W, m, v, c, who, t: int, lr: float, b1: float, b2: float,
               eps: float, lrt_buf: Optional[torch.Tensor] = None) -> None:
    """lrt_buf: optional persistent f32[1] device buffer holding lr_t —
    required under hipGraph capture (the per-step value must be read from
    device memory, not baked into the recorded launch)."""
    if W.is_cuda:
        if lrt_buf is None:
            lrt_buf = torch.tensor([tf1_lr_t(lr, b1, b2, t)],
                                   dtype=torch.float32, device=W.device)
        native().adam_rank1(W, m, v, c, who, lrt_buf, float(b1), float(b2),
                            float(eps))
        return
    cpu_ref.adam_rank1(W, m, v, c, who, t, lr, b1, b2, eps)


def adam_rank1_fused(W, m, v, c, who, mO, vO, t: int, lr: float,
                     b1: float, b2: float, eps: float,
                     lrt_buf: Optional[torch.Tensor] = None) -> None:
    """Fused epoch tail: rank-1 Adam on W_ih AND the dW_ho = W_pre^T c
    reduction AND the W_ho Adam update — one streaming pass over the
    pre-update W rows plus a single fold launch (replaces gemv_cols +
    adam_rank1 + adam_dense: three launches and an extra full W read).
    Deterministic (fixed per-thread/block fold order)."""
    if W.is_cuda:
        if lrt_buf is None:
            lrt_buf = torch.tensor([tf1_lr_t(lr, b1, b2, t)],
                                   dtype=torch.float32, device=W.device)
        native().adam_rank1(W, m, v, c, who, lrt_buf, float(b1), float(b2),
                            float(eps), mO=mO, vO=vO)
        return
    grad_who = torch.mv(W.t(), c)          # pre-update W
    cpu_ref.adam_rank1(W, m, v, c, who, t, lr, b1, b2, eps)
    cpu_ref.adam_dense(who, mO, vO, grad_who, t, lr, b1, b2, eps)


def adam_dense(W, m, v, grad, t: int, lr: float, b1: float, b2: float,
               eps: float, lrt_buf: Optional[torch.Tensor] = None) -> None:
    if W.is_cuda:
        if lrt_buf is None:
            lrt_buf = torch.tensor([tf1_lr_t(lr, b1, b2, t)],
                                   dtype=torch.float32, device=W.device)
        native().adam_dense(W, m, v, grad, lrt_buf, float(b1), float(b2),
                            float(eps))
        return
    cpu_ref.adam_dense(W, m, v, grad, t, lr, b1, b2, eps)


# ------------------------------------------------------------------ CBOW general (kernel-chain) path
def cbow_fwd(W, who, genes, offsets, labels, inv_b: float, want_grad: bool,
             act: int = 0):
    """act: 0 linear (reference semantics); 1 ReLU on the hidden vector
    (opt-in non-linear successor; returned H is the pre-activation)."""
    if W.is_cuda:
        loss, correct, dO, H = native().cbow_fwd(
            W, who, genes, offsets, labels, float(inv_b), bool(want_grad),
            act=int(act))
        if not want_grad:
            return loss, correct, None, None
        return loss, correct, dO, H
    return cpu_ref.cbow_fwd(W, who, genes, offsets, labels, inv_b, want_grad,
                            act=act)


def cbow_bwd_rows(who, genes, offsets, dO, n_genes: int,
                  plan: Optional[ScatterPlan] = None,
                  H_pre: Optional[torch.Tensor] = None):
    """General-path dW_ih backward. On GPU this is the deterministic
    per-gene-segment kernel (no atomics) driven by the same ScatterPlan as
    the fast path's c-reduction. H_pre given = ReLU backward mask source
    (the forward's pre-activation H)."""
    if dO.is_cuda:
        if plan is None:
            plan = build_scatter_plan(genes, offsets, n_genes)
        return native().cbow_bwd_rows(who, plan.inst_path, plan.seg_start,
                                      plan.seg_gene, dO, int(n_genes),
                                      Hpre=H_pre)
    return cpu_ref.cbow_bwd_rows(who, genes, offsets, dO, n_genes,
                                 H_pre=H_pre)


def gemv_rows(W, x, out) -> None:
    """out = W @ x (tall-skinny GEMV). Own wave-per-row kernel on GPU:
    rocBLAS gemvn ran at ~2% of HBM bandwidth on the [1M, 512] shape."""
    if W.is_cuda:
        native().gemv_rows_(W, x, out)
        return
    torch.mv(W, x, out=out)


def gemv_cols(W, c, out) -> None:
    """out = W^T @ c (column reduction over G rows); atomic-free
    partials + fold on GPU."""
    if W.is_cuda:
        native().gemv_cols_(W, c, out)
        return
    torch.mv(W.t(), c, out=out)


# ------------------------------------------------------------------ graph / PCC
def pcc_edges(zt, edge_idx, n_group: int):
    if zt.is_cuda:
        return native().pcc_edges(zt, edge_idx, int(n_group))
    return cpu_ref.pcc_edges(zt, edge_idx, n_group)


def corr_gemm(zt, n_group: int):
    if zt.is_cuda:
        return native().corr_gemm(zt, int(n_group))
    return cpu_ref.corr_gemm(zt, n_group)
