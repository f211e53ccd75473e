"""CPU reference implementations of every device op.

These are the numerics oracles for the HIP kernels (fp32, torch/numpy) and
the fallback execution path on hosts without a GPU. Semantics mirror the
reference pipeline:
  random walk      G2Vec.py:328-346 (non-revisiting weighted walk)
  CBOW fwd/loss    G2Vec.py:238-251 (linear net + sigmoid-CE, sum-reduce
                   because X is 0/1 multi-hot)
  Adam             G2Vec.py:245-246 (TF1 AdamOptimizer update rule)
  PCC              G2Vec.py:354-368
"""
from __future__ import annotations

from typing import Tuple

import numpy as np
import torch

# ---------------------------------------------------------------- RNG helpers
_SM64_GAMMA = np.uint64(0x9E3779B97F4A7C15)
_SM64_M1 = np.uint64(0xBF58476D1CE4E5B9)
_SM64_M2 = np.uint64(0x94D049BB133111EB)


def splitmix64(state: np.uint64) -> Tuple[np.uint64, np.uint64]:
    """One splitmix64 draw; returns (new_state, random_u64).
    Must stay bit-identical to the device implementation in g2vec_kernels.hip."""
    with np.errstate(over="ignore"):
        state = np.uint64(state + _SM64_GAMMA)
        z = state
        z = np.uint64((z ^ (z >> np.uint64(30))) * _SM64_M1)
        z = np.uint64((z ^ (z >> np.uint64(27))) * _SM64_M2)
        z = np.uint64(z ^ (z >> np.uint64(31)))
    return state, z


def _u01(r: np.uint64) -> float:
    return float(r >> np.uint64(11)) * (1.0 / 9007199254740992.0)


def gene_hash(g: int) -> np.uint64:
    """Order-independent per-gene mix used for path-set hashing (summed over
    the path's genes; commutative so no sort is needed)."""
    _, z = splitmix64(np.uint64(g) * _SM64_M1 + _SM64_GAMMA)
    return z


def path_hash(genes) -> np.int64:
    h = np.uint64(0)
    with np.errstate(over="ignore"):
        for g in genes:
            h = np.uint64(h + gene_hash(int(g)))
    return np.int64(h.astype(np.int64))


# ------------------------------------------------------------------ walks
def random_walks(row_ptr: torch.Tensor, col_idx: torch.Tensor, weights: torch.Tensor,
                 sources: torch.Tensor, num_repetition: int, len_path: int,
                 seed: int) -> Tuple[torch.Tensor, torch.Tensor, torch.Tensor]:
    """Non-revisiting weighted random walks on a CSR graph.

    One walk per (repetition, source), walk_id = rep * n_src + src_pos.
    Returns (nodes i32 [n_walks, len_path] padded -1, lengths i32, hashes i64).
    Matches the reference walk semantics (G2Vec.py:328-346): the current node
    is appended, all visited nodes (incl. current) are masked to weight 0,
    step chosen proportionally to remaining weights, dead-end breaks.
    """
    rp = row_ptr.cpu().numpy()
    ci = col_idx.cpu().numpy()
    w = weights.cpu().numpy().astype(np.float32)
    src = sources.cpu().numpy()
    n_src = len(src)
    n_walks = n_src * num_repetition
    nodes = np.full((n_walks, len_path), -1, dtype=np.int32)
    lengths = np.zeros(n_walks, dtype=np.int32)
    hashes = np.zeros(n_walks, dtype=np.int64)

    for wid in range(n_walks):
        rep = wid // n_src
        source = int(src[wid % n_src])
        # RNG keyed on the GLOBAL (source, repetition) so DP-sharded
        # generation is bitwise-identical to single-process
        gid = np.uint64(source) * np.uint64(num_repetition) + np.uint64(rep)
        with np.errstate(over="ignore"):
            state = np.uint64(np.uint64(seed) ^ np.uint64(gid * _SM64_M2 + np.uint64(1)))
        # warm the stream like the device kernel does
        state, _ = splitmix64(state)
        cur = source
        visited = []
        vset = set()
        for _step in range(len_path):
            visited.append(cur)
            vset.add(cur)
            s, e = int(rp[cur]), int(rp[cur + 1])
            if e <= s:
                break
            nb = ci[s:e]
            wt = w[s:e].copy()
            for k in range(len(nb)):
                if int(nb[k]) in vset:
                    wt[k] = 0.0
            tot = float(np.sum(wt, dtype=np.float32))
            state, r = splitmix64(state)
            if tot <= 0.0:
                break
            target = _u01(r) * tot
            c = np.cumsum(wt, dtype=np.float32)
            j = int(np.searchsorted(c, target, side="right"))
            if j >= len(nb) or wt[j] <= 0.0:
                # rounding tail: tot (pairwise np.sum) can exceed the
                # sequential cumsum total in f32, pushing target past
                # c[-1]. Fall back to the LAST POSITIVE-WEIGHT (unvisited)
                # neighbor, exactly like the device kernel's ballot
                # fallback (g2vec_kernels.hip walk_kernel) — never a
                # visited/zero-weight one (the non-revisit invariant that
                # paths.py's "paths are sets" assumption relies on).
                j = int(np.flatnonzero(wt > 0.0)[-1])
            cur = int(nb[j])
        plen = len(visited)
        nodes[wid, :plen] = visited
        lengths[wid] = plen
        hashes[wid] = path_hash(visited)
    return (torch.from_numpy(nodes), torch.from_numpy(lengths), torch.from_numpy(hashes))


# ------------------------------------------------------------------ K9 init
def trunc_normal_(out: torch.Tensor, std: float, seed: int) -> None:
    """Counter-based +-2sigma truncated normal, mirroring
    trunc_normal_kernel (g2vec_kernels.hip): element i draws Box-Muller
    normals from its own splitmix64 stream and resamples beyond 2sigma.
    Same streams as the device kernel; the float cos/log tails can differ
    by ulps across libm implementations, so parity tests compare with a
    small tolerance (plus bounds/moments), not bitwise."""
    import math

    flat = out.view(-1)
    n = flat.numel()
    vals = np.empty(n, dtype=np.float32)
    with np.errstate(over="ignore"):
        for i in range(n):
            state = np.uint64(seed) ^ np.uint64(
                np.uint64(i) * _SM64_M2 + np.uint64(1))
            state, _ = splitmix64(state)      # warm draw (kernel parity)
            while True:
                state, r1 = splitmix64(state)
                state, r2 = splitmix64(state)
                u1 = max(_u01(r1), 1e-300)
                x = np.float32(math.sqrt(-2.0 * math.log(u1)) *
                               math.cos(2.0 * math.pi * _u01(r2)))
                if abs(x) <= np.float32(2.0):
                    break
            vals[i] = x
    flat.copy_(torch.from_numpy(vals * np.float32(std)))


# ------------------------------------------------------------------ CBOW (fast scalar path)
def cbow_fwd_scalar(s: torch.Tensor, genes: torch.Tensor, offsets: torch.Tensor,
                    labels: torch.Tensor, inv_b: float, want_grad: bool):
    """Forward of the linear CBOW in collapsed scalar form.

    o_p = sum_{g in path p} s_g  where s = W_ih @ W_ho  (linearity of the
    reference net, G2Vec.py:238-240). Returns (loss[P], correct[P], dO[P]|None).
    """
    counts = (offsets[1:] - offsets[:-1]).long()
    seg = torch.repeat_interleave(torch.arange(len(counts)), counts)
    o = torch.zeros(len(counts), dtype=torch.float32)
    o.index_add_(0, seg, s[genes.long()])
    y = labels.float()
    loss = torch.clamp(o, min=0) - o * y + torch.log1p(torch.exp(-o.abs()))
    correct = ((o > 0).float() == y).float()
    dO = (torch.sigmoid(o) - y) * inv_b if want_grad else None
    return loss, correct, dO


def scatter_dO(genes: torch.Tensor, offsets: torch.Tensor, dO: torch.Tensor,
               n_genes: int) -> torch.Tensor:
    """c = X^T dO: for each gene, sum of dO over the paths containing it."""
    counts = (offsets[1:] - offsets[:-1]).long()
    seg = torch.repeat_interleave(torch.arange(len(counts)), counts)
    c = torch.zeros(n_genes, dtype=torch.float32)
    c.index_add_(0, genes.long(), dO[seg])
    return c


def adam_rank1(W: torch.Tensor, m: torch.Tensor, v: torch.Tensor,
               c: torch.Tensor, who: torch.Tensor, t: int,
               lr: float, b1: float, b2: float, eps: float) -> None:
    """Dense TF1-Adam step with the rank-1 gradient grad = c (outer) who.
    Moments stay dense (reference Adam decays untouched rows too)."""
    grad = torch.outer(c, who)
    adam_dense(W, m, v, grad, t, lr, b1, b2, eps)


def adam_dense(W: torch.Tensor, m: torch.Tensor, v: torch.Tensor,
               grad: torch.Tensor, t: int, lr: float, b1: float, b2: float,
               eps: float) -> None:
    """TF1 AdamOptimizer: theta -= lr * sqrt(1-b2^t)/(1-b1^t) * m/(sqrt(v)+eps)."""
    lr_t = lr * (1.0 - b2 ** t) ** 0.5 / (1.0 - b1 ** t)
    m.mul_(b1).add_(grad, alpha=1 - b1)
    v.mul_(b2).addcmul_(grad, grad, value=1 - b2)
    W.addcdiv_(m, v.sqrt() + eps, value=-lr_t)


# ------------------------------------------------------------------ CBOW (general kernel-chain path)
def cbow_fwd(W: torch.Tensor, who: torch.Tensor, genes: torch.Tensor,
             offsets: torch.Tensor, labels: torch.Tensor, inv_b: float,
             want_grad: bool, act: int = 0):
    """Row-gather forward: H_p = sum of W rows of the path's genes (sum, not
    mean — X is 0/1, G2Vec.py:239), o = act(H) @ who. act: 0 linear
    (reference), 1 ReLU (opt-in non-linear successor). Returns
    (loss, correct, dO|None, H|None) — H is the PRE-activation."""
    Wf = W.float()
    counts = (offsets[1:] - offsets[:-1]).long()
    P = len(counts)
    seg = torch.repeat_interleave(torch.arange(P), counts)
    H = torch.zeros(P, Wf.shape[1], dtype=torch.float32)
    H.index_add_(0, seg, Wf[genes.long()])
    o = (torch.relu(H) if act else H) @ who
    y = labels.float()
    loss = torch.clamp(o, min=0) - o * y + torch.log1p(torch.exp(-o.abs()))
    correct = ((o > 0).float() == y).float()
    if not want_grad:
        return loss, correct, None, None
    dO = (torch.sigmoid(o) - y) * inv_b
    return loss, correct, dO, H


def cbow_bwd_rows(who: torch.Tensor, genes: torch.Tensor, offsets: torch.Tensor,
                  dO: torch.Tensor, n_genes: int,
                  H_pre: torch.Tensor = None) -> torch.Tensor:
    """dW_ih = X^T (dO who^T [(.) relu_mask]): scatter-add dH rows into the
    touched gene rows. H_pre given = ReLU backward (mask = H_pre > 0)."""
    counts = (offsets[1:] - offsets[:-1]).long()
    seg = torch.repeat_interleave(torch.arange(len(counts)), counts)
    dH = dO[:, None] * who[None, :]
    if H_pre is not None:
        dH = dH * (H_pre > 0).float()
    dW = torch.zeros(n_genes, who.shape[0], dtype=torch.float32)
    dW.index_add_(0, genes.long(), dH[seg])
    return dW


# ------------------------------------------------------------------ graph / PCC
def pcc_edges(zt: torch.Tensor, edge_idx: torch.Tensor, n_group: int) -> torch.Tensor:
    """|PCC| per directed edge. zt: f32 [G, S_group] z-scored expression rows
    (zero rows where std==0, reproducing the pcc=0 rule of G2Vec.py:356-367)."""
    src = zt[edge_idx[:, 0].long()]
    dst = zt[edge_idx[:, 1].long()]
    return ((src * dst).sum(dim=1) / n_group).abs()


def corr_gemm(zt: torch.Tensor, n_group: int) -> torch.Tensor:
    """Full correlation matrix C = Zt Zt^T / S (the MFMA GEMM path on GPU)."""
    return (zt @ zt.t()) / n_group
