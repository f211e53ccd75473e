"""Walk oracle: structural invariants + sampling distribution
(reference walk semantics, G2Vec.py:328-346)."""
import numpy as np
import torch

from g2vec_amd.graph import CsrGraph
from g2vec_amd.ops import cpu_ref
from g2vec_amd.walks import generate_walks


def _csr(edges, weights, G):
    order = np.lexsort((edges[:, 1], edges[:, 0]))
    e = edges[order]
    w = np.asarray(weights, dtype=np.float32)[order]
    counts = np.bincount(e[:, 0], minlength=G)
    rp = np.zeros(G + 1, dtype=np.int32)
    rp[1:] = np.cumsum(counts)
    return CsrGraph(torch.from_numpy(rp), torch.from_numpy(e[:, 1].astype(np.int32)),
                    torch.from_numpy(w), G)


def test_walk_invariants():
    rng = np.random.default_rng(0)
    G = 30
    edges = np.unique(rng.integers(0, G, size=(200, 2)), axis=0)
    edges = edges[edges[:, 0] != edges[:, 1]]
    w = rng.uniform(0.5, 1.0, size=len(edges))
    g = _csr(edges, w, G)
    adj = {(int(a), int(b)) for a, b in edges}

    nodes, lengths, hashes = cpu_ref.random_walks(
        g.row_ptr, g.col_idx, g.weights,
        torch.arange(G, dtype=torch.int32), 3, 12, seed=7)
    nodes, lengths = nodes.numpy(), lengths.numpy()
    assert nodes.shape == (90, 12)
    for i in range(90):
        L = lengths[i]
        path = nodes[i, :L]
        assert nodes[i, 0] == i % G                 # starts at source
        assert len(set(path.tolist())) == L         # non-revisiting
        assert np.all(nodes[i, L:] == -1)
        for k in range(L - 1):
            assert (int(path[k]), int(path[k + 1])) in adj  # edge-following
        # dead end / max-length: either L == len_path or last node's
        # unvisited out-weight is ~0
        if L < 12:
            s, e = int(g.row_ptr[path[-1]]), int(g.row_ptr[path[-1] + 1])
            rest = [int(c) for c in g.col_idx[s:e] if int(c) not in set(path.tolist())]
            assert len(rest) == 0 or all(
                g.weights[s + j] == 0 for j in range(e - s)
                if int(g.col_idx[s + j]) in rest)


def test_walk_sampling_distribution():
    # star: node 0 -> {1,2,3} with weights 0.6/0.3/0.1; one-step walks
    edges = np.array([[0, 1], [0, 2], [0, 3]])
    g = _csr(edges, [0.6, 0.3, 0.1], 4)
    n = 6000
    nodes, lengths, _ = cpu_ref.random_walks(
        g.row_ptr, g.col_idx, g.weights,
        torch.zeros(1, dtype=torch.int32), n, 2, seed=3)
    first = nodes.numpy()[:, 1]
    freq = np.bincount(first, minlength=4)[1:4] / n
    assert np.allclose(freq, [0.6, 0.3, 0.1], atol=0.03)


def test_hash_order_independent():
    h1 = cpu_ref.path_hash([3, 1, 2])
    h2 = cpu_ref.path_hash([2, 3, 1])
    h3 = cpu_ref.path_hash([1, 2, 4])
    assert h1 == h2 and h1 != h3


def test_generate_walks_sharding():
    edges = np.array([[0, 1], [1, 2], [2, 3], [3, 0]])
    g = _csr(edges, [1, 1, 1, 1], 4)
    full = generate_walks(g, 4, 2, seed=1, group=0)
    assert full.nodes.shape[0] == 8
    lo = generate_walks(g, 4, 2, seed=1, group=0, src_range=(0, 2))
    assert lo.nodes.shape[0] == 4
    assert set(lo.nodes[:, 0].tolist()) == {0, 1}


def test_self_loops_never_revisited():
    """A self-loop edge (absent from the reference data but legal in user
    files) is a visited candidate from step one — the non-revisit mask
    must keep the walker off it."""
    import torch

    from g2vec_amd import ops
    G = 6
    # each node: self-loop + edge to next
    row_ptr = torch.arange(0, 2 * G + 1, 2, dtype=torch.int32)
    col, w = [], []
    for i in range(G):
        col += [i, (i + 1) % G]
        w += [5.0, 1.0]
    nodes, lengths, _ = ops.random_walks(
        torch.tensor(row_ptr.tolist(), dtype=torch.int32),
        torch.tensor(col, dtype=torch.int32),
        torch.tensor(w), torch.arange(G, dtype=torch.int32), 3, 8, seed=5)
    for k in range(nodes.shape[0]):
        path = nodes[k, :int(lengths[k])].tolist()
        assert len(path) == len(set(path))
