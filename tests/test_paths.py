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
