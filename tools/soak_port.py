"""Ad-hoc soak: framework integrate/freq vs literal port over many random
walk multisets and graph shapes (50 rounds)."""
import sys
import os
_root = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, _root)
sys.path.insert(0, os.path.join(_root, 'tests'))
import numpy as np, torch
from g2vec_amd import ops
from g2vec_amd.graph import build_group_graph
from g2vec_amd.paths import integrate_pathsets
from g2vec_amd.utils import synth
from g2vec_amd.walks import WalkSet
from literal_port import lp_integrate, lp_gene_freq

def walkset_to_paths(w):
    out = []
    for i in range(w.nodes.shape[0]):
        n = int(w.lengths[i])
        out.append(tuple(sorted(w.nodes[i, :n].tolist())))
    return out

rng = np.random.default_rng(2)
N_ROUNDS = int(os.environ.get('SOAK_ROUNDS', '50'))
for rnd in range(N_ROUNDS):
    G = int(rng.integers(40, 300))
    ds = synth.synth_dataset(G, int(G * rng.integers(5, 20)), 60,
                             n_modules=int(rng.integers(3, 9)),
                             seed=int(rng.integers(0, 10**6)),
                             dead_frac=float(rng.uniform(0, 0.4)),
                             shared_frac=float(rng.uniform(0, 0.4)))
    expr_t = torch.from_numpy(ds["expr"]); lab_t = torch.from_numpy(ds["labels"])
    edge_t = torch.from_numpy(ds["edge_idx"])
    graphs = [build_group_graph(expr_t, lab_t, g, edge_t, G) for g in (0, 1)]
    walksets = []
    lp = int(rng.integers(4, 40))        # len_path: one global per run
    for grp, g in enumerate(graphs):
        srcs = torch.arange(G, dtype=torch.int32)
        nodes, lengths, hashes = ops.random_walks(
            g.row_ptr, g.col_idx, g.weights, srcs, int(rng.integers(1, 5)),
            lp, seed=int(rng.integers(0, 10**6)))
        walksets.append(WalkSet(nodes, lengths, hashes))
    ps, freq, n_gip = integrate_pathsets(walksets[0], walksets[1], G)
    sets = [set(walkset_to_paths(w)) for w in walksets]
    rows = lp_integrate(sets[0], sets[1], G)
    names = [f"g{i:05d}" for i in range(G)]
    port_freq = lp_gene_freq(rows, names)
    port_paths = set()
    for r in rows:
        port_paths.add((frozenset(np.flatnonzero(r[:-1] == 1).tolist()), int(r[-1])))
    fw_paths = set()
    offs = ps.offsets.numpy()
    for p in range(ps.n_paths):
        fw_paths.add((frozenset(ps.genes[offs[p]:offs[p+1]].tolist()), int(ps.labels[p])))
    assert fw_paths == port_paths, f"round {rnd}: path sets differ"
    f = freq.numpy()
    for i, nm in enumerate(names):
        assert f[i] == port_freq.get(nm, 2), (rnd, nm)
    assert n_gip == len(port_freq), rnd
    print(f"round {rnd}: G={G} paths={ps.n_paths} OK", flush=True)
print(f"SOAK PASS: {N_ROUNDS} rounds")
