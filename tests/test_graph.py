"""Graph construction vs a literal dense-adjacency oracle of the reference
semantics (construct_adjMat, G2Vec.py:370-391)."""
import numpy as np
import torch

from g2vec_amd.graph import build_group_graph, zscore_group


def _dense_adj_reference(expr, labels, group, edges, G, thresh=0.5):
    """Direct reimplementation of the reference semantics for the oracle:
    dense [G,G], |PCC| over the group's samples, kept iff > thresh."""
    X = expr[labels == group]
    adj = np.zeros((G, G), dtype=np.float32)
    for (s, d) in edges:
        x, y = X[:, s], X[:, d]
        sx, sy = x.std(), y.std()
        if sx > 0 and sy > 0:
            pcc = float(np.mean((x - x.mean()) / sx * (y - y.mean()) / sy))
        else:
            pcc = 0.0
        if abs(pcc) > thresh:
            adj[s, d] = abs(pcc)
    return adj


def test_zscore_group():
    expr = torch.tensor([[1., 2., 5.], [3., 2., 1.], [5., 2., 3.]])
    labels = torch.tensor([0, 0, 0])
    zt = zscore_group(expr, labels, 0)
    assert zt.shape == (3, 3)
    # constant gene -> zero row
    assert torch.all(zt[1] == 0)
    assert abs(float(zt[0].mean())) < 1e-6
    assert abs(float((zt[0] ** 2).mean()) - 1.0) < 1e-5


def test_csr_matches_dense_reference():
    rng = np.random.default_rng(0)
    G, S, E = 40, 30, 300
    expr = rng.standard_normal((S, G)).astype(np.float32)
    # plant a few strong correlations
    expr[:, 1] = expr[:, 0] * 1.0 + 0.1 * rng.standard_normal(S)
    expr[:, 3] = -expr[:, 2] + 0.1 * rng.standard_normal(S)
    labels = np.array([0] * 15 + [1] * 15)
    edges = [(int(a), int(b)) for a, b in
             rng.integers(0, G, size=(E, 2)) if a != b]
    edges += [(0, 1), (2, 3)]

    for group in (0, 1):
        dense = _dense_adj_reference(expr, labels, group, edges, G)
        g = build_group_graph(torch.from_numpy(expr), torch.from_numpy(labels),
                              group, torch.tensor(edges, dtype=torch.int32),
                              G, mode="edge")
        rebuilt = np.zeros((G, G), dtype=np.float32)
        rp = g.row_ptr.numpy()
        for i in range(G):
            for k in range(rp[i], rp[i + 1]):
                rebuilt[i, g.col_idx[k]] = g.weights[k]
        assert np.allclose(rebuilt, dense, atol=1e-5), f"group {group}"
        assert (rebuilt > 0).sum() > 0  # planted edges survive in some group


def test_duplicate_edges_dedup():
    expr = np.random.default_rng(1).standard_normal((10, 4)).astype(np.float32)
    expr[:, 1] = expr[:, 0]
    labels = np.zeros(10, dtype=np.int64)
    edges = torch.tensor([[0, 1], [0, 1], [0, 1]], dtype=torch.int32)
    g = build_group_graph(torch.from_numpy(expr), torch.from_numpy(labels), 0,
                          edges, 4, mode="edge")
    assert g.col_idx.numel() == 1  # dense adjMat overwrites the same cell
    assert abs(float(g.weights[0]) - 1.0) < 1e-5
