#!/usr/bin/env python3
"""Flagship benchmark — the BASELINE.json headline on the ex_*-shaped config.

Metric: CBOW trained paths/sec (+ wall-clock to val-ACC >= 0.88, reported in
`config`) on the bundled-dataset shape: 7,523 common genes / ~216k restricted
edges / 135 samples / lenPath 80 / 10 repetitions / hidden 128 — synthetic
data of that shape (the reference's ex_EXPRESSION.txt is not distributable)
and random-init weights.

One timed "step" = one full reference training epoch: the full-batch
optimizer pass at W_t (forward + rank-1 backward + grad all-reduce + dense
TF1-Adam) plus the post-update train-ACC and val-ACC evaluations — exactly
the per-epoch work of the reference (G2Vec.py:262-267), which its ~2.2
s/epoch baseline also includes.

Scaling is WEAK over the walk budget of ONE shared study: every rank
builds the same dataset, the job runs num_repetition = reps x N of the
reference algorithm (exactly "the published run with more repetitions"),
each rank walks its source shard of all reps (C5), the walk shards are
all-gathered for GLOBAL dedup + common-path removal, and training is
DP-sharded over the global path set — bitwise the single-process
trajectory at the same global batch (tests/test_dist_gloo.py). Per-GPU
walk and training work stay ~constant as N grows; convergence IMPROVES
with N (more repetitions = denser path coverage). Per epoch exactly ONE
RCCL collective runs in the timed fixed-epoch region: the rank-1
backward's c vector (G floats; dW_ho is recomputed per rank from the
reduced c, and the accuracy counts accumulate in a device-side history
all-reduced once after the loop).

(An earlier design gave each rank an INDEPENDENT dataset; measured on a
2-rank rehearsal, DP training over conflicting per-rank ground truths
caps global val-ACC at ~0.85 — near-identical gene-set paths from
different cohorts carry conflicting labels, which the reference's
common-path removal deletes within one study but nothing can delete
across studies. See build_dataset's cohort_seed note.)

Usage:  python bench.py [--gpus N] [--steps K] [--warmup W]
Launched multi-GPU by the driver via torch.distributed.run (one rank/GPU).
Rank 0 prints ONE JSON line on stdout; progress goes to stderr.
"""
from __future__ import annotations

import argparse
import dataclasses
import json
import sys
import time

import numpy as np
import torch

from g2vec_amd.config import G2VecConfig
from g2vec_amd.models.cbow import CbowTrainer
from g2vec_amd.parallel.dist import init_dist
from g2vec_amd.pipeline import generate_paths
from g2vec_amd.utils import synth

BASELINE_PATHS_PER_SEC = 16500.0   # BASELINE.md derived reference number


def log(*a):
    print(*a, file=sys.stderr, flush=True)


def build_dataset(seed: int, n_genes: int = 7523, n_edges: int = 298799,
                  n_extra: int = 2381, n_modules: int = 16,
                  shared_frac: float = 0.2, off_frac: float = 0.55,
                  cohort_seed: int | None = None):
    """In-memory ex_*-shaped dataset (identical on every rank: the bench
    scales the walk budget of ONE shared study, not the dataset count).

    shared_frac/off_frac set the convergence difficulty (calibrated with
    tools/calibrate_difficulty.py so the seeded val-ACC trajectory climbs
    gradually to 0.88 over tens of epochs like the published transcript,
    instead of crossing at epoch 0 on a separable dataset).

    seed fixes the GROUND TRUTH (module assignment + network topology);
    cohort_seed (default = seed) draws the patient cohort (clinical
    labels + expression sampling) — a diagnostics knob. DP training over
    DIFFERENT datasets was measured and rejected for the bench: with
    per-rank structure seeds global val-ACC caps at 0.8757, and even
    with a shared structure but per-rank cohorts at 0.8524 (2-rank
    rehearsal, 120 epochs) — near-identical gene-set paths from
    different cohorts carry conflicting labels that common-path removal
    can only delete within one study."""
    if cohort_seed is None:
        cohort_seed = seed
    rng = np.random.default_rng(seed)
    n_net = n_genes + n_extra
    module = np.full(n_net, -1, dtype=np.int64)
    live = rng.choice(n_genes, size=int(n_genes * 0.5), replace=False)
    module[live] = rng.integers(0, n_modules, size=live.size)
    _, edge_idx, _ = synth.synth_network(n_net, n_edges, n_modules, seed,
                                         module=module)
    # restrict to the common genes (network extras have no expression)
    keep = (edge_idx[:, 0] < n_genes) & (edge_idx[:, 1] < n_genes)
    edge_idx = edge_idx[keep]
    _, labels = synth.synth_clinical(135, 58, cohort_seed)
    expr = synth.synth_expression(
        [f"G{i}" for i in range(n_genes)], labels, module[:n_genes],
        cohort_seed, shared_frac=shared_frac, off_frac=off_frac)
    return expr, np.asarray(labels), edge_idx.astype(np.int32), n_genes


def _launch_health_probe() -> None:
    """~8% of fresh processes on this pool come up with a degraded HIP
    submission path: every launch/copy costs ~0.3 ms instead of ~3 us,
    so the 0.064 ms epoch measures ~2.4 ms (observed twice in ~25 runs,
    profiles/README.md; the same box is normal in the neighboring
    processes). Detect it BEFORE any measurement by timing 200 trivial
    launches; if degraded, re-exec once — the replacement process
    re-initializes HIP cleanly. Runs before init_dist so a re-exec'd
    torchrun rank simply rejoins the rendezvous."""
    import os
    if os.environ.get("G2VEC_HEALTH_REEXEC") == "1":
        return
    if not torch.cuda.is_available():
        return
    # probe THIS rank's GPU (LOCAL_RANK not yet consumed by init_dist) —
    # probing cuda:0 from every rank would pin 8 stray contexts there
    dev = int(os.environ.get("LOCAL_RANK", "0")) % torch.cuda.device_count()
    torch.cuda.set_device(dev)
    t = torch.zeros(1, device=f"cuda:{dev}")
    for _ in range(5):
        t.fill_(1.0)
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(200):
        t.fill_(1.0)
    torch.cuda.synchronize()
    per_launch = (time.perf_counter() - t0) / 200
    if per_launch > 100e-6:
        log(f"[bench] WARNING: degraded HIP launch path "
            f"({per_launch * 1e6:.0f} us/launch vs ~3 normal); "
            f"re-executing once for a clean process")
        os.environ["G2VEC_HEALTH_REEXEC"] = "1"
        sys.stderr.flush()
        sys.stdout.flush()
        os.execv(sys.executable, [sys.executable] + sys.argv)


def main() -> int:
    _launch_health_probe()
    ap = argparse.ArgumentParser()
    ap.add_argument("--gpus", type=int, default=1)
    ap.add_argument("--steps", type=int, default=30)
    ap.add_argument("--warmup", type=int, default=5)
    ap.add_argument("--hidden", type=int, default=128)
    ap.add_argument("--len-path", type=int, default=80)
    ap.add_argument("--reps", type=int, default=10)
    ap.add_argument("--seed", type=int, default=0)
    ap.add_argument("--trainer-path", choices=["fast", "general"],
                    default="fast")
    ap.add_argument("--dtype", choices=["fp32", "bf16", "fp16"],
                    default="fp32",
                    help="W_ih gather storage dtype (general path only; "
                         "the fast path computes fp32 regardless)")
    ap.add_argument("--n-genes", type=int, default=7523,
                    help="scale configs: 50000 / 200000 / 1000000")
    ap.add_argument("--n-edges", type=int, default=298799)
    ap.add_argument("--n-extra", type=int, default=2381)
    ap.add_argument("--n-modules", type=int, default=16)
    ap.add_argument("--shared-frac", type=float, default=0.2,
                    help="difficulty: fraction of modules co-expressed in "
                         "both classes (calibrated default)")
    ap.add_argument("--off-frac", type=float, default=0.55,
                    help="difficulty: inactive-class loading fraction "
                         "(calibrated default)")
    ap.add_argument("--acc-target-epochs", type=int, default=120)
    ap.add_argument("--conv-seeds", type=int, default=5,
                    help="training seeds for the wall-to-ACC>=0.88 probe "
                         "(same dataset, reshuffled split + fresh init, "
                         "like the unseeded reference)")
    ap.add_argument("--real-data", action="store_true",
                    help="real ex_NETWORK/ex_CLINICAL topology (9,904 "
                         "genes, 298,799 edges; synthesized expression "
                         "over the 7,523-gene published intersection)")
    ap.add_argument("--no-hipgraph", action="store_true")
    ap.add_argument("--no-pipeline", action="store_true",
                    help="time the synchronous run_epoch loop instead of "
                         "run_epochs_pipelined (A/B diagnostics)")
    args = ap.parse_args()

    ctx = init_dist("auto")
    device = ctx.device
    on_gpu = device.type == "cuda"
    if on_gpu:
        import g2vec_amd.ops as ops
        assert ops.native_available(), \
            "bench on GPU requires the native extension (build_ext --inplace)"
    world = ctx.world
    rank = ctx.rank
    log(f"[bench] rank {rank}/{world} device {device}")

    # ONE study per job: the global walk budget is reps x world of the
    # SAME dataset (weak scaling — each rank walks its source shard of
    # all repetitions, so per-rank walk work is ~constant in N)
    cfg = G2VecConfig(hidden=args.hidden, len_path=args.len_path,
                      num_repetition=args.reps * world, epochs=500,
                      seed=args.seed,
                      device=str(device.type), trainer_path=args.trainer_path,
                      dtype=args.dtype, use_hipgraph=not args.no_hipgraph)

    # ---- dataset + graphs + walks (input pipeline; measured, not the metric)
    if args.real_data:
        from g2vec_amd.utils import refdata
        ds = refdata.make_real_dataset(seed=args.seed)
        g2i = {g: i for i, g in enumerate(ds["net_genes"])}
        keep = np.array([g2i[g] for g in ds["expr_genes"]])
        idx_of = np.full(len(ds["net_genes"]), -1, np.int64)
        idx_of[keep] = np.arange(len(keep))
        e = ds["edge_idx"]
        m = (idx_of[e[:, 0]] >= 0) & (idx_of[e[:, 1]] >= 0)
        edge_idx = np.stack([idx_of[e[m, 0]], idx_of[e[m, 1]]],
                            1).astype(np.int32)
        expr, labels = ds["expr"], np.asarray(ds["labels"])
        n_genes = len(ds["expr_genes"])
    else:
        expr, labels, edge_idx, n_genes = build_dataset(
            args.seed, args.n_genes, args.n_edges, args.n_extra,
            args.n_modules, shared_frac=args.shared_frac,
            off_frac=args.off_frac)
    expr_t = torch.from_numpy(expr).to(device)
    labels_t = torch.from_numpy(labels).to(device)
    edge_t = torch.from_numpy(edge_idx).to(device)

    # C5 sharded generation: rank r walks sources [lo, hi) of the shared
    # study at the global repetition count; _gather_walks all-gathers the
    # shards so every rank holds the GLOBAL walks before dedup/common
    # removal (world 1: identical to a single-process context)
    sctx = ctx
    if on_gpu:
        # one-time hipModule/dispatcher loads + allocator arenas for the
        # step-3 op set (~1.2 s on a fresh process,
        # profiles/step3_cold.json) happen in a full-size dummy round so
        # walks_per_sec reports the pipeline, not runtime init
        # two rounds: the first pays module loads + arena growth, the
        # second settles allocator reuse for the timed shapes
        for wseed in (999, cfg.seed):
            warm_cfg = dataclasses.replace(cfg, seed=wseed)
            generate_paths(warm_cfg, expr_t, labels_t, edge_t, n_genes, sctx,
                           log=(lambda *a, **k2: None))
        torch.cuda.synchronize()

    t0 = time.perf_counter()
    ps, _freq, _nip, stats = generate_paths(
        cfg, expr_t, labels_t, edge_t, n_genes, sctx,
        log=(lambda *a, **k2: None))
    if on_gpu:
        torch.cuda.synchronize()
    walk_s = time.perf_counter() - t0   # END-TO-END step 3: both groups'
                                        # graph builds + walks + dedup
    n_walks = int(stats["n_walks"])
    log(f"[bench] rank {rank}: {n_walks} walks in {walk_s:.3f}s "
        f"({n_walks / walk_s:.0f} walks/s end-to-end), {ps.n_paths} paths "
        f"after dedup")

    # ---- convergence probe: epochs + wall to val-ACC >= 0.88, over
    # --conv-seeds independent training seeds (same dataset; fresh split
    # shuffle + weight init per seed, mirroring the unseeded reference's
    # run-to-run variation). Untimed work for the throughput metric, but
    # itself the secondary headline.
    conv_runs = []
    acc_val = 0.0
    for k in range(max(args.conv_seeds, 1)):
        cfg_k = dataclasses.replace(cfg, seed=args.seed + 7919 * k)
        trainer = CbowTrainer(cfg_k, n_genes, device, ctx,
                              log=(lambda *a, **k2: None))
        st = trainer.setup(ps, pre_sharded=False)
        wall_k = None
        acc_k = 0.0
        ep_k = None
        hist_k = []
        conv_t0 = time.perf_counter()
        for ep in range(args.acc_target_epochs):
            _acc_tr, a = trainer.run_epoch(st)
            hist_k.append(round(a, 4))
            acc_k = max(acc_k, a)
            if a >= 0.88:
                if on_gpu:
                    torch.cuda.synchronize()
                wall_k = time.perf_counter() - conv_t0
                ep_k = ep
                break
        conv_runs.append({"seed": cfg_k.seed, "epochs_to_0.88": ep_k,
                          "wall_s": (round(wall_k, 4) if wall_k else None),
                          "best_acc": round(acc_k, 4)})
        if k == 0:
            acc_val = acc_k
            wall_to_acc = wall_k
            log(f"[bench] rank {rank}: seed {cfg_k.seed} trajectory "
                f"{hist_k[:40]}")
        log(f"[bench] rank {rank}: conv seed {cfg_k.seed}: best ACC "
            f"{acc_k:.4f}, 0.88 at epoch {ep_k} ({wall_k} s)")
    crossed = [r["wall_s"] for r in conv_runs if r["wall_s"] is not None]
    cross_ep = [r["epochs_to_0.88"] for r in conv_runs
                if r["epochs_to_0.88"] is not None]

    # ---- timed throughput region: fresh state, W warmup + K timed epochs
    trainer = CbowTrainer(cfg, n_genes, device, ctx,
                          log=(lambda *a, **k2: None))
    st = trainer.setup(ps, pre_sharded=False)
    n_tr_global = trainer.n_tr_global
    pipelined = (on_gpu and args.trainer_path == "fast"
                 and not args.no_pipeline)
    if pipelined:
        trainer.run_epochs_pipelined(st, args.warmup, early_stop=False)
        if trainer.kblock_eligible(st, args.steps, False) and st.epoch_idx:
            # one-time capture stays untimed; block size tiles the timed
            # run so no epoch falls to the slower eager tail
            trainer._ensure_kgraph(st, k=trainer.pick_kblock(args.steps))
    else:
        for _ in range(args.warmup):
            trainer.run_epoch(st)
    ctx.barrier()
    if on_gpu:
        torch.cuda.synchronize()
    t0 = time.perf_counter()
    if pipelined:
        trainer.run_epochs_pipelined(st, args.steps, early_stop=False)
    else:
        for _ in range(args.steps):
            trainer.run_epoch(st)
    if on_gpu:
        torch.cuda.synchronize()
    ctx.barrier()
    elapsed = time.perf_counter() - t0
    emax = torch.tensor([elapsed], dtype=torch.float64, device=device)
    ctx.allreduce_max_(emax)
    elapsed = float(emax.item())

    ms_per_step = elapsed * 1000.0 / args.steps
    value = n_tr_global * args.steps / elapsed

    if ctx.is_primary:
        out = {
            "metric": "cbow_paths_per_sec",
            "value": round(value, 1),
            "unit": "paths/s",
            "n_gpus": world,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": round(ms_per_step, 3),
            "higher_is_better": True,
            "scaling": "weak",
            "vs_baseline": round(value / BASELINE_PATHS_PER_SEC, 2),
            "dtype": args.dtype,
            "data": (f"real ex_NETWORK topology ({n_genes} common genes of "
                     f"9904/135 real clinical samples; synthetic expression; "
                     f"random-init weights)" if args.real_data else
                     f"synthetic ({args.n_genes} genes/135 samples/"
                     f"{args.n_edges} network edges; random-init weights)"),
            "config": {
                "model": "g2vec-cbow",
                "global_batch": n_tr_global,
                "seq_len": args.len_path,
                "parallelism": f"dp{world}",
                "hidden": args.hidden,
                "num_repetition": args.reps,
                "num_repetition_global": args.reps * world,
                "trainer_path": args.trainer_path,
                "hipgraph_active": bool(
                    getattr(st, "graph", None) is not None
                    or getattr(st, "kgraph", None) is not None),
                "pipe_pinned": (bool(st.pipe_bufs[0][0].is_pinned())
                                if getattr(st, "pipe_bufs", None) else None),
                "val_acc": round(acc_val, 4),
                "wall_to_val_acc_0.88_s": (round(wall_to_acc, 4)
                                           if wall_to_acc else None),
                "conv_seeds": len(conv_runs),
                "conv_crossed": len(crossed),
                "wall_to_0.88_s_median": (round(float(np.median(crossed)), 4)
                                          if crossed else None),
                "wall_to_0.88_speedup_vs_ref": (
                    round(56.7 / float(np.median(crossed)), 1)
                    if crossed else None),   # reference: ~56.7 s
                "epochs_to_0.88_median": (float(np.median(cross_ep))
                                          if cross_ep else None),
                "conv_runs": conv_runs,
                "walks_per_sec": round(n_walks / walk_s, 1),
                "grad_allreduce_bytes_per_epoch": n_genes * 4,
                "step_includes": "full-batch fwd+bwd+allreduce+dense-Adam "
                                 "+ post-update train/val ACC evals",
            },
        }
        print(json.dumps(out), flush=True)
    return 0


if __name__ == "__main__":
    raise SystemExit(main())
