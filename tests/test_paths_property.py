"""Property-based checks of path-set integration (hypothesis)."""
import os

import torch
from hypothesis import given, settings
from hypothesis import strategies as st

_SOAK = max(int(os.environ.get("G2VEC_SOAK", "1")), 1)  # soak runs scale the example budget

from g2vec_amd.ops.cpu_ref import path_hash
from g2vec_amd.paths import integrate_pathsets
from g2vec_amd.walks import WalkSet

paths_strategy = st.lists(
    st.lists(st.integers(0, 19), min_size=1, max_size=6, unique=True),
    min_size=1, max_size=25)


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


@settings(max_examples=60 * _SOAK, deadline=None)
@given(good=paths_strategy, poor=paths_strategy)
def test_integrate_properties(good, poor):
    ps, freq, n_in = integrate_pathsets(_ws(good), _ws(poor), 20)

    good_sets = {frozenset(p) for p in good}
    poor_sets = {frozenset(p) for p in poor}
    common = good_sets & poor_sets

    kept = []
    offs = ps.offsets.tolist()
    for i in range(ps.n_paths):
        genes = frozenset(ps.genes[offs[i]:offs[i + 1]].tolist())
        kept.append((genes, int(ps.labels[i].item())))

    # 1. exactly the per-group unique sets minus the cross-group common ones
    expect = {(s, 0) for s in good_sets - common} | \
             {(s, 1) for s in poor_sets - common}
    assert set(kept) == expect
    assert len(kept) == len(expect)          # no duplicates survive

    # 2. gene frequency semantics (G2Vec.py:299-307) vs brute force
    for g in range(20):
        fg = sum(1 for s, lab in kept if lab == 0 and g in s)
        fp = sum(1 for s, lab in kept if lab == 1 and g in s)
        want = 0 if fg > fp else (1 if fp > fg else 2)
        assert int(freq[g]) == want

    # 3. genes-in-paths count
    assert n_in == len({g for s, _ in kept for g in s})
