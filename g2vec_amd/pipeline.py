"""End-to-end 7-step pipeline driver (reference main(), G2Vec.py:11-119).

Console transcript mirrors the reference's phase-numbered prints (they ARE
the published baseline); structured JSONL metrics go beside
them when cfg.log_jsonl is set.
"""
from __future__ import annotations

import time
from typing import Dict, Optional

import numpy as np
import torch

from . import io as gio
from . import preprocess as pp
from .cluster import find_lgroups
from .config import G2VecConfig, resolve_device
from .graph import build_group_graph
from .models.cbow import CbowTrainer
from .parallel.dist import DistContext, single
from .paths import PathSet, integrate_pathsets
from .scoring import select_biomarkers
from .utils.obs import JsonlLogger, PhaseTimers
from .walks import WalkSet, generate_walks


def _gather_walks(ctx: DistContext, ws: WalkSet) -> WalkSet:
    """C5: all-gather per-rank walk shards into the global walk set."""
    if ctx.world == 1:
        return ws
    nodes = torch.cat(ctx.allgather_varlen(ws.nodes))
    lengths = torch.cat(ctx.allgather_varlen(ws.lengths.unsqueeze(1))).squeeze(1)
    hashes = torch.cat(ctx.allgather_varlen(ws.hashes.unsqueeze(1))).squeeze(1)
    return WalkSet(nodes, lengths, hashes)


def generate_paths(cfg: G2VecConfig, expr_t: torch.Tensor, labels_t: torch.Tensor,
                   edge_idx_t: torch.Tensor, n_genes: int, ctx: DistContext,
                   log=print, timers: Optional[PhaseTimers] = None):
    """Steps 2b-3: per-group graph construction + random walks + integration.
    Returns (pathset, gene_freq, n_genes_in_paths, stats dict)."""
    timers = timers or PhaseTimers()
    seed = cfg.seed if cfg.seed is not None else int(time.time_ns() & 0x7FFFFFFF)
    walksets = []
    stats: Dict[str, float] = {}
    lo, hi = ctx.shard_range(n_genes)
    from .graph import dedupe_edges
    edge_idx_t = dedupe_edges(edge_idx_t, n_genes)   # once for both groups
    if expr_t.is_cuda:
        # the two prognosis groups' graph-build + walk chains are
        # independent: run them on separate HIP streams so each group's
        # PCC/threshold/CSR torch chains and walk kernel overlap the
        # other's (the C5 all-gather stays on the default stream, after
        # both streams join). Bitwise identical output — only scheduling
        # changes.
        streams = [torch.cuda.Stream(), torch.cuda.Stream()]
        graphs = [None, None]
        raw = [None, None]
        with timers.phase("graphs_walks_overlapped"):
            for group in (0, 1):
                streams[group].wait_stream(torch.cuda.current_stream())
                with torch.cuda.stream(streams[group]):
                    g = build_group_graph(expr_t, labels_t, group,
                                          edge_idx_t, n_genes,
                                          threshold=cfg.pcc_threshold,
                                          mode=cfg.pcc_mode,
                                          edges_deduped=True)
                    graphs[group] = g
                    raw[group] = generate_walks(g, cfg.len_path,
                                                cfg.num_repetition, seed,
                                                group, (lo, hi))
            for st_ in streams:
                torch.cuda.current_stream().wait_stream(st_)
        for group in (0, 1):
            with timers.phase(f"walks_g{group}"):
                walksets.append(_gather_walks(ctx, raw[group]))
            stats[f"nnz_g{group}"] = int(graphs[group].col_idx.numel())
    else:
        for group in (0, 1):
            with timers.phase(f"graph_g{group}"):
                g = build_group_graph(expr_t, labels_t, group, edge_idx_t,
                                      n_genes, threshold=cfg.pcc_threshold,
                                      mode=cfg.pcc_mode, edges_deduped=True)
            with timers.phase(f"walks_g{group}"):
                ws = generate_walks(g, cfg.len_path, cfg.num_repetition, seed,
                                    group, (lo, hi))
                ws = _gather_walks(ctx, ws)
            walksets.append(ws)
            stats[f"nnz_g{group}"] = int(g.col_idx.numel())
    with timers.phase("integrate"):
        ps, freq, n_in_paths = integrate_pathsets(walksets[0], walksets[1],
                                                  n_genes)
    stats["n_walks"] = int(walksets[0].nodes.shape[0] + walksets[1].nodes.shape[0])
    return ps, freq, n_in_paths, stats


def run(cfg: G2VecConfig, ctx: Optional[DistContext] = None) -> Dict:
    cfg.validate()
    device = torch.device(resolve_device(cfg.device))
    ctx = ctx or single(device)
    log = print if ctx.is_primary else (lambda *a, **k: None)
    jsonl = JsonlLogger(cfg.log_jsonl if ctx.is_primary else "")
    timers = PhaseTimers(jsonl)

    log(">>> 0. Arguments")
    log("    " + str(cfg))

    log(">>> 1. Load data")
    with timers.phase("load"):
        data = gio.load_expression(cfg.expression_file)
        clinical = gio.load_clinical(cfg.clinical_file)
        network = gio.load_network(cfg.network_file)

    log(">>> 2. Preprocess data")
    with timers.phase("preprocess"):
        data["label"] = pp.match_labels(clinical, data["sample"])
        common = pp.find_common_genes(network["gene"], data["gene"])
        network = pp.restrict_network(network, common)
        data = pp.restrict_data(data, common)
        edge_idx = pp.edges_to_indices(network["edge"], common)
    n_samples, n_genes = data["expr"].shape
    n_edges = len(network["edge"])
    n_poor = int(np.sum(np.asarray(data["label"]) == 1))
    if min(n_poor, n_samples - n_poor) < 2:
        # per-group PCC and the pooled t-statistic need >= 2 samples per
        # prognosis group; the reference nan-propagates here — fail loud
        raise ValueError(
            f"each prognosis group needs >= 2 samples: clinical file has "
            f"{n_samples - n_poor} good / {n_poor} poor")
    log("    n_samples: %d" % n_samples)
    log("    n_genes  : %d\t(common genes in both EXPRESSION and NETWORK)" % n_genes)
    log("    n_edges  : %d\t(edges with the common genes)" % n_edges)
    jsonl.emit("counts", n_samples=n_samples, n_genes=n_genes, n_edges=n_edges)

    expr_t = torch.from_numpy(data["expr"]).to(device)
    labels_t = torch.from_numpy(np.asarray(data["label"])).to(device)
    edge_idx_t = torch.from_numpy(edge_idx).to(device)

    log(">>> 3. Generate random paths from each group")
    log("    *** most time consuming step ***")
    if cfg.load_paths:
        blob = torch.load(cfg.load_paths, map_location=device)
        ps = PathSet(blob["genes"], blob["offsets"], blob["labels"], n_genes)
        freq, n_in_paths = blob["freq"], int(blob["n_in_paths"])
        stats = {}
    else:
        ps, freq, n_in_paths, stats = generate_paths(
            cfg, expr_t, labels_t, edge_idx_t, n_genes, ctx, log, timers)
        if cfg.save_paths and ctx.is_primary:
            torch.save({"genes": ps.genes, "offsets": ps.offsets,
                        "labels": ps.labels, "freq": freq,
                        "n_in_paths": n_in_paths}, cfg.save_paths)
    log("    n_paths : %d" % ps.n_paths)
    log("    n_genes : %d\t(genes in good or poor random paths)" % n_in_paths)
    jsonl.emit("paths", n_paths=ps.n_paths, n_genes_in_paths=n_in_paths, **stats)

    log(">>> 4. Compute distributed representations using modified CBOW")
    if cfg.load_model:
        # resume from a --save-model checkpoint: skip training entirely
        # (steps 5-7 only need W_ih); shape metadata must match this run
        blob = torch.load(cfg.load_model, map_location="cpu")
        if int(blob["n_genes"]) != n_genes or int(blob["hidden"]) != cfg.hidden:
            raise ValueError(
                f"--load-model checkpoint was trained on n_genes="
                f"{blob['n_genes']}/hidden={blob['hidden']}, this run has "
                f"n_genes={n_genes}/hidden={cfg.hidden}")
        from .models.cbow import TrainResult
        res = TrainResult(W_ih=blob["W_ih"].float(),
                          stop_epoch=int(blob.get("stop_epoch", -1)),
                          acc_val=float(blob.get("acc_val", float("nan"))),
                          acc_tr=float(blob.get("acc_tr", float("nan"))),
                          epochs_run=0, acc_val_history=[],
                          epoch_times_s=[], wall_to_acc_s=None)
        log("    (loaded weights from %s; training skipped)" % cfg.load_model)
    else:
        trainer = CbowTrainer(cfg, n_genes, device, ctx, log=log)
        with timers.phase("train"):
            res = trainer.train(ps)
    jsonl.emit("train", acc_val=res.acc_val, acc_tr=res.acc_tr,
               stop_epoch=res.stop_epoch, epochs_run=res.epochs_run,
               wall_to_acc088_s=res.wall_to_acc_s)
    if cfg.save_model and ctx.is_primary:
        torch.save({"W_ih": res.W_ih.float().cpu(),
                    "acc_val": res.acc_val, "acc_tr": res.acc_tr,
                    "stop_epoch": res.stop_epoch,
                    "hidden": cfg.hidden, "n_genes": n_genes,
                    "gene_index": list(map(str, data["gene"]))},
                   cfg.save_model)

    W = res.W_ih.float().cpu().numpy()
    result: Dict = {
        "n_samples": n_samples, "n_genes": n_genes, "n_edges": n_edges,
        "n_paths": ps.n_paths, "n_genes_in_paths": n_in_paths,
        "acc_val": res.acc_val, "acc_tr": res.acc_tr,
        "stop_epoch": res.stop_epoch, "W_ih": W,
        "timers": timers.summary(), "genes": data["gene"],
    }
    if not ctx.is_primary:
        return result

    log(">>> 5. Find L-groups")
    with timers.phase("lgroups"):
        lg = find_lgroups(W, freq.cpu().numpy(), cfg.compat_lgroup_bug,
                          backend=cfg.kmeans_backend, device=device)

    log(">>> 6. Select biomarkers with gene scores")
    with timers.phase("scoring"):
        biomarkers = select_biomarkers(W, lg, data["expr"], data["label"],
                                       data["gene"], cfg.num_biomarker)

    log(">>> 7. Save results")
    with timers.phase("write"):
        f1 = gio.write_biomarkers(cfg.result_name, biomarkers)
        log("    %s" % f1)
        f2 = gio.write_lgroups(cfg.result_name, lg, data["gene"])
        log("    %s" % f2)
        f3 = gio.write_vectors(cfg.result_name, W, data["gene"])
        log("    %s" % f3)

    result.update({"lgroups": lg, "biomarkers": biomarkers,
                   "timers": timers.summary()})
    jsonl.emit("done", **{k: v for k, v in result.items()
                          if isinstance(v, (int, float, str))})
    jsonl.close()
    return result
