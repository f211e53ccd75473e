"""Real bundled-reference-data integration (round-1 verdict item 3).

The reference ships its real example network and clinical files
(`ex_NETWORK.txt`: 298,799 directed edges over 9,904 genes, header
`src\tdest`, ex_NETWORK.txt:1; `ex_CLINICAL.txt`: 135 samples, 77 good /
58 poor) but NOT `ex_EXPRESSION.txt` (absent from the mount,
.MISSING_LARGE_BLOBS:1). This module runs the pipeline on the REAL
topology: it parses the two real files and synthesizes a seeded
expression matrix over 7,523 of the 9,904 network genes — the published
run's common-gene count (reference README.md:27) — so the full pipeline
(restriction, per-group PCC graphs, walks over the real hub structure
with max out-degree 889, training, scoring) exercises real data.

Gene selection is the top `n_common` by total degree (deterministic,
name-tiebroken): it keeps the high-degree hubs whose thresholded rows
exercise the walk kernel's >256-neighbor fallback. Module structure for
the expression factor model comes from label propagation on the real
restricted topology, so within-community network edges are exactly the
ones whose |PCC| can clear the 0.5 threshold — the walk graph IS a
subgraph of the real network.

A committed binary cache (`ex_ref.npz`, a different representation — int
edge indices + name table, not the TSV) stands in when `/root/reference`
is not mounted (e.g. on a GPU box that only receives the repo snapshot).
"""
from __future__ import annotations

import os
from typing import Dict, List, Optional

import numpy as np

from . import synth

REF_DIR = "/root/reference"
CACHE_PATH = os.path.join(os.path.dirname(__file__), "..", "data",
                          "ex_ref.npz")
N_COMMON_PUBLISHED = 7523      # reference README.md:27


def _parse_reference(ref_dir: str) -> Dict:
    """Parse the real ex_NETWORK.txt / ex_CLINICAL.txt (formats:
    G2Vec.py:455-476 and :436-453; headers skipped)."""
    net_path = os.path.join(ref_dir, "ex_NETWORK.txt")
    cli_path = os.path.join(ref_dir, "ex_CLINICAL.txt")
    with open(net_path) as f:
        rows = [ln.rstrip("\n").split("\t") for ln in f.readlines()[1:] if ln.strip()]
    genes = sorted({g for r in rows for g in r[:2]})
    g2i = {g: i for i, g in enumerate(genes)}
    edge_idx = np.array([[g2i[a], g2i[b]] for a, b in rows], dtype=np.int32)
    with open(cli_path) as f:
        crows = [ln.rstrip("\n").split("\t") for ln in f.readlines()[1:] if ln.strip()]
    samples = [r[0] for r in crows]
    labels = np.array([int(r[1]) for r in crows], dtype=np.int64)
    return {"genes": genes, "edge_idx": edge_idx,
            "samples": samples, "labels": labels}


def build_cache(ref_dir: str = REF_DIR, cache_path: str = CACHE_PATH) -> str:
    """Snapshot the parsed reference files into the committed npz cache."""
    raw = _parse_reference(ref_dir)
    os.makedirs(os.path.dirname(cache_path), exist_ok=True)
    np.savez_compressed(
        cache_path,
        genes=np.array(raw["genes"]), edge_idx=raw["edge_idx"],
        samples=np.array(raw["samples"]), labels=raw["labels"])
    return cache_path


def load_ref_raw(ref_dir: Optional[str] = None) -> Dict:
    """Real files when mounted, committed cache otherwise."""
    ref_dir = ref_dir or REF_DIR
    if os.path.exists(os.path.join(ref_dir, "ex_NETWORK.txt")):
        return _parse_reference(ref_dir)
    z = np.load(CACHE_PATH, allow_pickle=False)
    return {"genes": [str(g) for g in z["genes"]],
            "edge_idx": z["edge_idx"].astype(np.int32),
            "samples": [str(s) for s in z["samples"]],
            "labels": z["labels"].astype(np.int64)}


def _label_propagation(n: int, u: np.ndarray, v: np.ndarray, k: int,
                       iters: int, seed: int) -> np.ndarray:
    """Community ids in [0,k) by synchronous majority label propagation
    over the undirected edge set (u,v)."""
    rng = np.random.default_rng(seed + 99)
    uu = np.concatenate([u, v])
    vv = np.concatenate([v, u])
    mod = rng.integers(0, k, size=n)
    for _ in range(iters):
        cnt = np.zeros((n, k), np.int32)
        np.add.at(cnt, (uu, mod[vv]), 1)
        best = cnt.argmax(1)
        has = cnt.max(1) > 0
        mod = np.where(has, best, mod)
    return mod


def make_real_dataset(seed: int = 0, n_common: int = N_COMMON_PUBLISHED,
                      n_modules: int = 32, lp_iters: int = 4,
                      shared_frac: float = 0.1, off_frac: float = 0.45,
                      min_module: int = 30, dead_frac: float = 0.0,
                      ref_dir: Optional[str] = None,
                      sample_seed: Optional[int] = None) -> Dict:
    """Real network + real clinical + synthesized expression.

    Difficulty note: with the default knobs (size-ordered community ids,
    shared=0.1, off=0.45 — calibrated in the real-topology sweep) the
    seeded val-ACC climbs ~0.58 -> ~0.88 at the full 10-repetition walk
    budget: the bench --real-data probe measured best-ACC 0.8825 with
    3/3 training seeds crossing 0.88 (profiles/bench_real_r2.json). At
    reduced walk budgets (reps <= 3) the ceiling sits lower (~0.83):
    off-class walks over a module's genes are label-ambiguous for the
    linear model until enough paths accumulate.

    Returns {'expr' f32 [S, n_common] (samples x chosen genes),
    'expr_genes', 'samples', 'labels', 'net_genes', 'edge_idx' (full real
    network, int32 into net_genes), 'module'} — feed the expr through
    write_expression_tsv (or in-memory) and the real network/clinical
    through the standard loaders."""
    raw = load_ref_raw(ref_dir)
    genes: List[str] = raw["genes"]
    e = raw["edge_idx"]
    G = len(genes)
    deg = np.bincount(e[:, 0], minlength=G) + np.bincount(e[:, 1], minlength=G)
    # top-degree, name-tiebroken (genes is sorted, argsort stable)
    order = np.argsort(-deg, kind="stable")
    chosen = np.sort(order[:n_common])               # ascending gene index
    keep = np.zeros(G, bool)
    keep[chosen] = True
    idx_of = np.full(G, -1, np.int64)
    idx_of[chosen] = np.arange(n_common)
    ke = e[keep[e[:, 0]] & keep[e[:, 1]]]
    u, v = idx_of[ke[:, 0]], idx_of[ke[:, 1]]
    module = _label_propagation(n_common, u, v, n_modules, lp_iters, seed)
    sizes = np.bincount(module, minlength=n_modules)
    module = np.where(sizes[module] >= min_module, module, -1)
    # renumber communities by ASCENDING size: module_activity marks the
    # first round(K*shared_frac) ids shared-in-both-classes and
    # alternates good/poor over the rest, so with ascending ids the
    # ambiguous (shared) modules are the SMALLEST communities and the
    # large ones split between the classes in size-interleaved order —
    # otherwise the giant label-prop community can land on a shared id
    # and erase a third of the label signal
    live_ids = np.unique(module[module >= 0])
    order = live_ids[np.argsort(sizes[live_ids], kind="stable")]
    remap = np.full(n_modules, -1, np.int64)
    remap[order] = np.arange(len(order))
    module = np.where(module >= 0, remap[np.clip(module, 0, None)], -1)
    if dead_frac > 0:
        # the published real run covers only 3,773 of 7,523 genes with
        # paths (reference README.md:32) — about half the real expression
        # genes never co-express above threshold. Kill a matching
        # fraction outright (pure noise in BOTH classes): their walks are
        # singletons in both group graphs and drop as common paths,
        # instead of becoming conflicting one-class singletons.
        rng = np.random.default_rng(seed + 13)
        dead = rng.random(n_common) < dead_frac
        module = np.where(dead, -1, module)
    expr_genes = [genes[i] for i in chosen]
    # sample_seed (default = seed) draws only the expression sampling:
    # DP weak scaling passes a per-rank sample_seed over a shared seed so
    # every rank sees the SAME module ground truth (labels are the real
    # clinical file, fixed) — see bench.build_dataset's cohort_seed note
    expr = synth.synth_expression(
        expr_genes, raw["labels"], module,
        seed if sample_seed is None else sample_seed,
        shared_frac=shared_frac, off_frac=off_frac)
    return {"expr": expr, "expr_genes": expr_genes,
            "samples": raw["samples"], "labels": raw["labels"],
            "net_genes": genes, "edge_idx": e, "module": module,
            "n_restricted_edges": int(len(ke))}


def write_dataset_files(outdir: str, seed: int = 0,
                        ref_dir: Optional[str] = None, **kw) -> Dict[str, str]:
    """Materialize the file triple for the standard CLI/pipeline path.
    Network/clinical are the REAL reference files when mounted (returned
    by path, parsed by the production loaders); otherwise they are
    reconstructed from the cache in the same TSV format. Expression is
    the synthesized matrix."""
    ref_dir = ref_dir or REF_DIR
    ds = make_real_dataset(seed=seed, ref_dir=ref_dir, **kw)
    os.makedirs(outdir, exist_ok=True)
    paths = {"expression": os.path.join(outdir, "real_EXPRESSION.txt")}
    synth.write_expression_tsv(paths["expression"], ds["expr_genes"],
                               ds["samples"], ds["expr"])
    if os.path.exists(os.path.join(ref_dir, "ex_NETWORK.txt")):
        paths["network"] = os.path.join(ref_dir, "ex_NETWORK.txt")
        paths["clinical"] = os.path.join(ref_dir, "ex_CLINICAL.txt")
    else:
        paths["network"] = os.path.join(outdir, "real_NETWORK.txt")
        paths["clinical"] = os.path.join(outdir, "real_CLINICAL.txt")
        synth.write_network_tsv(paths["network"], ds["net_genes"],
                                ds["edge_idx"])
        synth.write_clinical_tsv(paths["clinical"], ds["samples"],
                                 ds["labels"])
    return paths
