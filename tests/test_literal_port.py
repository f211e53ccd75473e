"""A/B tests of the framework against the literal reference port
(tests/literal_port.py) — reference EQUIVALENCE, not self-consistency
(round-1 verdict item 5 / SURVEY §4.1).

Exact tests (deterministic semantics): adjacency/PCC thresholding,
path integration + common-path removal, gene frequencies, L-group
assignment (both bug-compat modes), and walks on a deterministic graph.
Distributional tests (the walk RNGs differ by design): path-length and
gene-coverage histograms of the port's dense np.random.choice walker vs
the framework's CSR walker on the same fixed graph.
"""
import numpy as np
import pytest
import torch

from g2vec_amd import ops
from g2vec_amd.cluster import find_lgroups
from g2vec_amd.graph import build_group_graph
from g2vec_amd.paths import integrate_pathsets
from g2vec_amd.utils import synth
from g2vec_amd.walks import WalkSet

from literal_port import (lp_adj_matrix, lp_find_lgroups, lp_gene_freq,
                          lp_integrate, lp_random_path)


@pytest.fixture(scope="module")
def small_ds():
    ds = synth.synth_dataset(220, 3000, 80, n_modules=6, seed=3,
                             dead_frac=0.3, shared_frac=0.2)
    return ds


@pytest.fixture(scope="module")
def graphs(small_ds):
    ds = small_ds
    expr_t = torch.from_numpy(ds["expr"])
    lab_t = torch.from_numpy(ds["labels"])
    edge_t = torch.from_numpy(ds["edge_idx"])
    return [build_group_graph(expr_t, lab_t, g, edge_t, 220)
            for g in (0, 1)]


def _csr_to_dense(g):
    adj = np.zeros((g.n_nodes, g.n_nodes), dtype=np.float32)
    rp = g.row_ptr.numpy()
    for i in range(g.n_nodes):
        adj[i, g.col_idx[rp[i]:rp[i + 1]].numpy()] = \
            g.weights[rp[i]:rp[i + 1]].numpy()
    return adj


def test_adjacency_matches_port(small_ds, graphs):
    """CSR group graph == the port's dense construct_adjMat semantics
    (|PCC| over group samples, strict > 0.5, directed, overwrite dups)."""
    ds = small_ds
    for grp in (0, 1):
        want = lp_adj_matrix(ds["edge_idx"], ds["expr"], ds["labels"], grp)
        got = _csr_to_dense(graphs[grp])
        assert (got > 0).sum() == (want > 0).sum()
        assert np.array_equal(got > 0, want > 0)
        np.testing.assert_allclose(got, want, atol=2e-5)


def _walkset_to_paths(ws: WalkSet):
    out = set()
    for i in range(ws.nodes.shape[0]):
        out.add(tuple(sorted(ws.nodes[i, :int(ws.lengths[i])].tolist())))
    return out


def test_integrate_and_freq_match_port(graphs):
    """Framework integrate (hash dedup + common removal + bincount freq)
    == the port's set-based integrate_pathSet + count_geneFreq on the
    SAME walk multisets."""
    G = graphs[0].n_nodes
    gene_names = [f"g{i:04d}" for i in range(G)]
    walksets = []
    for grp, g in enumerate(graphs):
        srcs = torch.arange(G, dtype=torch.int32)
        nodes, lengths, hashes = ops.random_walks(
            g.row_ptr, g.col_idx, g.weights, srcs, 4, 24, seed=11 + grp)
        walksets.append(WalkSet(nodes, lengths, hashes))
    ps, freq, n_gip = integrate_pathsets(walksets[0], walksets[1], G)

    sets = [_walkset_to_paths(w) for w in walksets]
    rows = lp_integrate(sets[0], sets[1], G)
    port_freq = lp_gene_freq(rows, gene_names)

    # kept paths: same (gene-set, label) collection
    port_paths = set()
    for r in rows:
        port_paths.add((frozenset(np.flatnonzero(r[:-1] == 1).tolist()),
                        int(r[-1])))
    fw_paths = set()
    offs = ps.offsets.numpy()
    for p in range(ps.n_paths):
        fw_paths.add((frozenset(ps.genes[offs[p]:offs[p + 1]].tolist()),
                      int(ps.labels[p])))
    assert fw_paths == port_paths

    # gene frequency labels: port keys only genes seen in paths; the
    # framework array defaults unseen genes to 2, same as the port's
    # consumer (geneFreq.get(gene, 2), G2Vec.py:172)
    f = freq.numpy()
    for i, name in enumerate(gene_names):
        assert f[i] == port_freq.get(name, 2), name
    assert n_gip == len(port_freq)


def test_find_lgroups_matches_port_both_modes():
    rng = np.random.default_rng(5)
    W = rng.standard_normal((400, 16)).astype(np.float32)
    W[:150] += 2.0          # three separable blobs
    W[150:250] -= 2.0
    freq = rng.integers(0, 3, size=400)
    gene_names = [f"g{i}" for i in range(400)]
    freq_dict = {g: int(v) for g, v in zip(gene_names, freq) if v != 2}
    for compat in (False, True):
        want = lp_find_lgroups(W, gene_names, freq_dict, compat_bug=compat)
        got = find_lgroups(W, freq, compat_lgroup_bug=compat,
                           backend="sklearn")
        assert np.array_equal(got, want), f"compat={compat}"


def test_walks_deterministic_graph_match_port():
    """On a weighted chain (single out-neighbor per node) no randomness
    survives: the port and the framework walker must emit IDENTICAL
    paths, including the dead-end stop."""
    G = 12
    adj = np.zeros((G, G), dtype=np.float32)
    for i in range(G - 1):
        adj[i, i + 1] = 0.9
    adj[G - 1, 0] = 0.9     # cycle closure: non-revisit must cut it
    rng = np.random.RandomState(0)
    row_ptr = torch.tensor([0] + list(np.cumsum((adj > 0).sum(1))),
                           dtype=torch.int32)
    col_idx = torch.tensor(np.flatnonzero(adj)[...] % G, dtype=torch.int32)
    weights = torch.tensor(adj[adj > 0], dtype=torch.float32)
    for src in range(G):
        want = lp_random_path(src, adj, 8, rng)
        nodes, lengths, _ = ops.random_walks(
            row_ptr, col_idx, weights, torch.tensor([src], dtype=torch.int32),
            1, 8, seed=99)
        got = tuple(sorted(nodes[0, :int(lengths[0])].tolist()))
        assert got == want, src


def test_walk_distribution_matches_port(graphs):
    """Distributional A/B on the same fixed graph: the two walkers use
    different RNG streams by design, so compare path-length histograms
    (total-variation distance) and per-gene coverage counts (correlation)
    over ~4.4k walks each."""
    g = graphs[0]
    G = g.n_nodes
    adj = _csr_to_dense(g)
    reps = 20
    rng = np.random.RandomState(7)
    port_lens = []
    port_cov = np.zeros(G, dtype=np.int64)
    for _ in range(reps):
        for src in range(G):
            p = lp_random_path(src, adj, 24, rng)
            port_lens.append(len(p))
            port_cov[list(p)] += 1
    srcs = torch.arange(G, dtype=torch.int32)
    nodes, lengths, _ = ops.random_walks(
        g.row_ptr, g.col_idx, g.weights, srcs, reps, 24, seed=17)
    fw_lens = lengths.numpy()
    fw_cov = np.zeros(G, dtype=np.int64)
    m = nodes.numpy() >= 0
    np.add.at(fw_cov, nodes.numpy()[m], 1)

    # length histograms: total variation distance
    bins = np.arange(1, 26)
    hp = np.bincount(port_lens, minlength=26)[1:26].astype(float)
    hf = np.bincount(fw_lens, minlength=26)[1:26].astype(float)
    hp /= hp.sum()
    hf /= hf.sum()
    tv = 0.5 * np.abs(hp - hf).sum()
    assert tv < 0.05, f"length-histogram TV distance {tv:.4f}"
    assert abs(np.mean(port_lens) - fw_lens.mean()) < 0.5

    # coverage: every gene visited comparably often
    r = np.corrcoef(port_cov, fw_cov)[0, 1]
    assert r > 0.98, f"coverage correlation {r:.4f}"


def test_biomarker_selection_matches_port():
    """Step 6 (d-score + t-score + top-N union) — framework
    select_biomarkers vs the literal port, exact list equality."""
    from g2vec_amd.scoring import select_biomarkers
    from literal_port import lp_select_biomarkers

    rng = np.random.default_rng(17)
    G, S, h = 300, 60, 16
    W = rng.standard_normal((G, h)).astype(np.float32)
    expr = rng.standard_normal((S, G)).astype(np.float32)
    labels = (rng.random(S) < 0.45).astype(np.int64)
    lg = rng.integers(0, 3, size=G)
    genes = [f"SYM{i:04d}" for i in range(G)]
    for n in (10, 50, 500):
        want = lp_select_biomarkers(W, lg, expr, labels, genes, n)
        got = select_biomarkers(W, lg, expr, labels, genes, n)
        assert got == want, n


def test_walk_distribution_matches_port_on_real_topology():
    """Distributional A/B on the REAL ex_NETWORK-derived group graph:
    the framework's CSR walker vs the port's dense np.random.choice
    walker, sampled over 200 shared sources x 3 repetitions."""
    from g2vec_amd.utils import refdata

    ds = refdata.make_real_dataset(seed=0)
    g2i = {g: i for i, g in enumerate(ds["net_genes"])}
    keep = np.array([g2i[g] for g in ds["expr_genes"]])
    idx_of = np.full(len(ds["net_genes"]), -1, np.int64)
    idx_of[keep] = np.arange(len(keep))
    e = ds["edge_idx"]
    m = (idx_of[e[:, 0]] >= 0) & (idx_of[e[:, 1]] >= 0)
    ei = torch.from_numpy(
        np.stack([idx_of[e[m, 0]], idx_of[e[m, 1]]], 1).astype(np.int32))
    g = build_group_graph(torch.from_numpy(ds["expr"]),
                          torch.from_numpy(ds["labels"]), 0, ei, 7523)

    # dense view of the same graph for the port walker
    adj = np.zeros((7523, 7523), dtype=np.float32)
    rp = g.row_ptr.numpy()
    ci = g.col_idx.numpy()
    wv = g.weights.numpy()
    col_rows = np.repeat(np.arange(7523), np.diff(rp))
    adj[col_rows, ci] = wv

    rng_pick = np.random.default_rng(3)
    deg = np.diff(rp)
    sources = rng_pick.choice(np.flatnonzero(deg > 0), size=200,
                              replace=False)
    reps = 5
    rng = np.random.RandomState(11)
    port_lens = []
    port_cov = np.zeros(7523, dtype=np.int64)
    for _ in range(reps):
        for src in sources:
            p = lp_random_path(int(src), adj, 80, rng)
            port_lens.append(len(p))
            port_cov[list(p)] += 1

    from g2vec_amd import ops
    nodes, lengths, _ = ops.random_walks(
        g.row_ptr, g.col_idx, g.weights,
        torch.tensor(sources, dtype=torch.int32), reps, 80, seed=29)
    fw_lens = lengths.numpy()
    fw_cov = np.zeros(7523, dtype=np.int64)
    msk = nodes.numpy() >= 0
    np.add.at(fw_cov, nodes.numpy()[msk], 1)

    hp = np.bincount(port_lens, minlength=82)[1:82].astype(float)
    hf = np.bincount(fw_lens, minlength=82)[1:82].astype(float)
    hp /= hp.sum()
    hf /= hf.sum()
    tv = 0.5 * np.abs(hp - hf).sum()
    # 1000 walks per side: statistical tolerance is looser than the
    # 4.4k-walk synthetic A/B (coverage counts are sparse at this sample
    # size — the synthetic test pins the tight bound)
    assert tv < 0.12, f"length-histogram TV {tv:.4f}"
    assert abs(np.mean(port_lens) - fw_lens.mean()) < 3.0
    touched = (port_cov + fw_cov) > 0
    r = np.corrcoef(port_cov[touched], fw_cov[touched])[0, 1]
    assert r > 0.88, f"coverage correlation {r:.4f}"
