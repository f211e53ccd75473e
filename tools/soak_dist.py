"""Distributional walk A/B soak: framework CSR walker vs literal-port
dense walker over 20 random graph shapes (TV distance of length
histograms + coverage correlation)."""
import sys
import os
_root = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, _root)
sys.path.insert(0, os.path.join(_root, 'tests'))
import numpy as np, torch
from g2vec_amd import ops
from g2vec_amd.graph import build_group_graph
from g2vec_amd.utils import synth
from literal_port import lp_random_path

def csr_to_dense(g):
    G = g.n_nodes
    adj = np.zeros((G, G), dtype=np.float32)
    rp = g.row_ptr.numpy(); ci = g.col_idx.numpy(); w = g.weights.numpy()
    for r in range(G):
        adj[r, ci[rp[r]:rp[r+1]]] = w[rp[r]:rp[r+1]]
    return adj

rng = np.random.default_rng(5)
N_ROUNDS = int(os.environ.get('SOAK_ROUNDS', '20'))
for rnd in range(N_ROUNDS):
    G = int(rng.integers(60, 220))
    lp = int(rng.integers(8, 30))
    ds = synth.synth_dataset(G, int(G * rng.integers(6, 15)), 60,
                             n_modules=int(rng.integers(3, 8)),
                             seed=int(rng.integers(0, 10**6)),
                             dead_frac=float(rng.uniform(0, 0.3)),
                             shared_frac=float(rng.uniform(0, 0.3)))
    g = build_group_graph(torch.from_numpy(ds["expr"]),
                          torch.from_numpy(ds["labels"]),
                          int(rng.integers(0, 2)),
                          torch.from_numpy(ds["edge_idx"]), G)
    adj = csr_to_dense(g)
    reps = 20
    prng = np.random.RandomState(int(rng.integers(0, 10**6)))
    port_lens = []
    port_cov = np.zeros(G, np.int64)
    for _ in range(reps):
        for src in range(G):
            p = lp_random_path(src, adj, lp, prng)
            port_lens.append(len(p))
            port_cov[list(p)] += 1
    srcs = torch.arange(G, dtype=torch.int32)
    nodes, lengths, _ = ops.random_walks(g.row_ptr, g.col_idx, g.weights,
                                         srcs, reps, lp,
                                         seed=int(rng.integers(0, 10**6)))
    fw_lens = lengths.numpy()
    fw_cov = np.zeros(G, np.int64)
    m = nodes.numpy() >= 0
    np.add.at(fw_cov, nodes.numpy()[m], 1)
    hp = np.bincount(port_lens, minlength=lp + 2)[1:lp + 2].astype(float)
    hf = np.bincount(fw_lens, minlength=lp + 2)[1:lp + 2].astype(float)
    hp /= hp.sum(); hf /= hf.sum()
    tv = 0.5 * np.abs(hp - hf).sum()
    r = np.corrcoef(port_cov, fw_cov)[0, 1]
    dm = abs(np.mean(port_lens) - fw_lens.mean())
    assert tv < 0.07, (rnd, tv)
    assert r > 0.97, (rnd, r)
    assert dm < 0.6, (rnd, dm)
    print(f"round {rnd}: G={G} lp={lp} tv={tv:.4f} cov_r={r:.4f} dmean={dm:.3f} OK", flush=True)
print(f"DIST SOAK PASS: {N_ROUNDS} rounds")
