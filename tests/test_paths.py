"""Path-set integration semantics (dedup, common removal, gene freq —
G2Vec.py:288-322)."""
import torch

from g2vec_amd.ops.cpu_ref import path_hash
from g2vec_amd.paths import PathSet, integrate_pathsets, subset
from g2vec_amd.walks import WalkSet


def _ws(paths, len_path=6):
    n = len(paths)
    nodes = torch.full((n, len_path), -1, dtype=torch.int32)
    lengths = torch.zeros(n, dtype=torch.int32)
    hashes = torch.zeros(n, dtype=torch.int64)
    for i, p in enumerate(paths):
        nodes[i, :len(p)] = torch.tensor(p, dtype=torch.int32)
        lengths[i] = len(p)
        hashes[i] = int(path_hash(p))
    return WalkSet(nodes, lengths, hashes)


def test_integrate_dedup_and_common_removal():
    good = _ws([[0, 1, 2], [2, 1, 0], [3, 4], [5]])      # first two identical sets
    poor = _ws([[3, 4], [5, 6], [7]])                    # {3,4} common -> dropped
    ps, freq, n_in = integrate_pathsets(good, poor, 10)
    # good keeps {0,1,2} and {5}; poor keeps {5,6} and {7}
    assert ps.n_paths == 4
    labels = ps.labels.tolist()
    assert labels.count(0.0) == 2 and labels.count(1.0) == 2
    got = set()
    offs = ps.offsets.tolist()
    for i in range(ps.n_paths):
        got.add((tuple(sorted(ps.genes[offs[i]:offs[i + 1]].tolist())),
                 labels[i]))
    assert got == {((0, 1, 2), 0.0), ((5,), 0.0), ((5, 6), 1.0), ((7,), 1.0)}

    # gene freq: 0 good-more, 1 poor-more, 2 tie/absent (G2Vec.py:299-307)
    f = freq.tolist()
    assert f[0] == 0 and f[1] == 0 and f[2] == 0
    assert f[6] == 1 and f[7] == 1
    assert f[5] == 2          # one good + one poor -> tie
    assert f[9] == 2          # absent
    assert n_in == 6          # genes 0,1,2,5,6,7


def test_subset():
    genes = torch.tensor([0, 1, 2, 3, 4, 5], dtype=torch.int32)
    offsets = torch.tensor([0, 2, 3, 6], dtype=torch.int32)
    labels = torch.tensor([0., 1., 0.])
    ps = PathSet(genes, offsets, labels, 10)
    sub = subset(ps, torch.tensor([2, 0]))
    assert sub.offsets.tolist() == [0, 3, 5]
    assert sub.genes.tolist() == [3, 4, 5, 0, 1]
    assert sub.labels.tolist() == [0., 0.]


def test_scatter_plan_slab_geometry():
    """Slab-blocked plan (ops.SLAB_BYTES-sized path slabs): segments are
    (slab, gene)-sorted, slab_seg_ptr partitions them, every instance is
    covered exactly once, and each slab's instances stay in its path
    range."""
    import numpy as np

    import g2vec_amd.ops as ops
    rng = np.random.default_rng(3)
    P, G = 64, 37
    lens = rng.integers(1, 9, size=P)
    genes = torch.tensor(
        np.concatenate([rng.choice(G, size=n, replace=False) for n in lens]),
        dtype=torch.int32)
    offs = torch.tensor(np.concatenate([[0], np.cumsum(lens)]),
                        dtype=torch.int32)
    orig = ops.SLAB_BYTES
    try:
        ops.SLAB_BYTES = 4 * 16      # 16-path slabs -> 4 slabs
        plan = ops.build_scatter_plan(genes, offs, G, _force_slabs=True)
    finally:
        ops.SLAB_BYTES = orig
    ptr = plan.slab_seg_ptr
    assert ptr is not None and ptr[0] == 0 and ptr[-1] == plan.seg_gene.numel()
    assert int(plan.seg_start[-1]) == genes.numel()
    seen = 0
    for slab, (a, b) in enumerate(zip(ptr[:-1], ptr[1:])):
        gs = plan.seg_gene[a:b].tolist()
        assert gs == sorted(gs)                      # gene-sorted per slab
        for sidx in range(a, b):
            lo, hi = int(plan.seg_start[sidx]), int(plan.seg_start[sidx + 1])
            assert hi > lo
            for i in range(lo, hi):
                p = int(plan.inst_path[i])
                assert p // 16 == slab               # instance in its slab
                assert int(plan.seg_gene[sidx]) in genes[
                    int(offs[p]):int(offs[p + 1])].tolist()
            seen += hi - lo
    assert seen == genes.numel()
