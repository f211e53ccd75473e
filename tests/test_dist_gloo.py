"""Multi-process DP coverage on CPU (gloo, world_size=2): the C1-C5
collective path must produce the same training result as single-process
(SURVEY §4.4)."""
import socket

import numpy as np
import pytest
import torch
import torch.distributed as dist
import torch.multiprocessing as mp

from g2vec_amd.config import G2VecConfig
from g2vec_amd.models.cbow import CbowTrainer
from g2vec_amd.parallel.dist import DistContext
from g2vec_amd.paths import PathSet


def _free_port():
    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    p = s.getsockname()[1]
    s.close()
    return p


def _pathset(G=40, P=160, seed=11):
    rng = np.random.default_rng(seed)
    genes, offs, labels = [], [0], []
    for _ in range(P):
        L = int(rng.integers(1, 10))
        genes += rng.choice(G, size=L, replace=False).tolist()
        offs.append(offs[-1] + L)
        labels.append(float(rng.integers(0, 2)))
    return PathSet(torch.tensor(genes, dtype=torch.int32),
                   torch.tensor(offs, dtype=torch.int32),
                   torch.tensor(labels), G)


def _worker(rank, world, port, out):
    dist.init_process_group("gloo", rank=rank, world_size=world,
                            init_method=f"tcp://127.0.0.1:{port}")
    try:
        ctx = DistContext(rank, world, torch.device("cpu"), True)
        ps = _pathset()
        cfg = G2VecConfig(hidden=64, epochs=6, early_stop=False, seed=4,
                          device="cpu", dtype="fp32")
        tr = CbowTrainer(cfg, ps.n_genes, torch.device("cpu"), ctx,
                         log=lambda *a, **k: None)
        res = tr.train(ps)

        # varlen allgather smoke (C5)
        t = torch.arange((rank + 1) * 3, dtype=torch.float32).reshape(-1, 1)
        parts = ctx.allgather_varlen(t)
        assert [p.shape[0] for p in parts] == [3, 6]
        assert torch.allclose(parts[1][:, 0], torch.arange(6, dtype=torch.float32))

        if rank == 0:
            out.put((res.W_ih.numpy(), res.acc_val_history))
    finally:
        dist.destroy_process_group()


@pytest.mark.timeout(300)
def test_dp2_matches_single_process():
    port = _free_port()
    ctxm = mp.get_context("spawn")
    out = ctxm.Queue()
    procs = [ctxm.Process(target=_worker, args=(r, 2, port, out))
             for r in range(2)]
    for p in procs:
        p.start()
    W_dp, hist_dp = out.get(timeout=240)
    for p in procs:
        p.join(timeout=120)
        assert p.exitcode == 0

    ps = _pathset()
    cfg = G2VecConfig(hidden=64, epochs=6, early_stop=False, seed=4,
                      device="cpu", dtype="fp32")
    tr = CbowTrainer(cfg, ps.n_genes, torch.device("cpu"),
                     log=lambda *a, **k: None)
    res = tr.train(ps)
    assert hist_dp == pytest.approx(res.acc_val_history, abs=1e-6)
    assert np.allclose(W_dp, res.W_ih.numpy(), atol=1e-5)


def _pipeline_worker(rank, world, port, files, outdir, q):
    dist.init_process_group("gloo", rank=rank, world_size=world,
                            init_method=f"tcp://127.0.0.1:{port}")
    try:
        from g2vec_amd.pipeline import run
        ctx = DistContext(rank, world, torch.device("cpu"), True)
        cfg = G2VecConfig(expression_file=files["expression"],
                          clinical_file=files["clinical"],
                          network_file=files["network"],
                          result_name=f"{outdir}/dp{rank}",
                          len_path=12, num_repetition=2, epochs=8,
                          device="cpu", seed=0, early_stop=False)
        res = run(cfg, ctx)
        if rank == 0:
            q.put((res["n_paths"], res["n_genes_in_paths"], res["acc_val"]))
    finally:
        dist.destroy_process_group()


@pytest.mark.timeout(300)
def test_dp2_pipeline_end_to_end(tiny_files, tmp_path):
    """Full pipeline under DP=2 (gloo): the sharded walk generation +
    all-gather must reproduce the single-process path set EXACTLY (the walk
    RNG is keyed on the global (source, repetition) pair)."""
    port = _free_port()
    ctxm = mp.get_context("spawn")
    q = ctxm.Queue()
    procs = [ctxm.Process(target=_pipeline_worker,
                          args=(r, 2, port, tiny_files, str(tmp_path), q))
             for r in range(2)]
    for p in procs:
        p.start()
    n_paths_dp, n_gip_dp, acc_dp = q.get(timeout=240)
    for p in procs:
        p.join(timeout=120)
        assert p.exitcode == 0

    from g2vec_amd.pipeline import run
    cfg = G2VecConfig(expression_file=tiny_files["expression"],
                      clinical_file=tiny_files["clinical"],
                      network_file=tiny_files["network"],
                      result_name=str(tmp_path / "sp"),
                      len_path=12, num_repetition=2, epochs=8,
                      device="cpu", seed=0, early_stop=False)
    res = run(cfg)
    assert n_paths_dp == res["n_paths"]
    assert n_gip_dp == res["n_genes_in_paths"]
    assert abs(acc_dp - res["acc_val"]) < 0.05


class _CountingCtx(DistContext):
    """DistContext that counts allreduce_ calls (collective-schedule tests)."""

    def __init__(self, rank, world, device):
        super().__init__(rank, world, device, True)
        self.n_allreduce = 0

    def allreduce_(self, t):
        self.n_allreduce += 1
        return super().allreduce_(t)


def _kblocked_worker(rank, world, port, n_epochs, out):
    dist.init_process_group("gloo", rank=rank, world_size=world,
                            init_method=f"tcp://127.0.0.1:{port}")
    try:
        ctx = _CountingCtx(rank, world, torch.device("cpu"))
        ps = _pathset()
        cfg = G2VecConfig(hidden=64, epochs=n_epochs, early_stop=False,
                          seed=4, device="cpu", dtype="fp32")
        tr = CbowTrainer(cfg, ps.n_genes, torch.device("cpu"), ctx,
                         log=lambda *a, **k: None)
        st = tr.setup(ps)
        n_setup = ctx.n_allreduce
        hist, stop, W, _who, _ = tr.run_epochs_pipelined(
            st, n_epochs, early_stop=False)
        assert stop == -1
        if rank == 0:
            out.put((np.asarray(W), hist, ctx.n_allreduce - n_setup))
    finally:
        dist.destroy_process_group()


@pytest.mark.timeout(300)
def test_dp2_kblocked_fixed_epochs_deferred_metrics():
    """Fixed-epoch deferred-readback runner under world_size=2 (the exact
    schedule the multi-GPU SCALE bench runs, minus RCCL): per-epoch
    accuracy counts stay local and are all-reduced ONCE after the loop.
    Checks (a) the trajectory and final weights match the single-process
    sync loop bit-for-bit within fp32 tolerance, and (b) the collective
    schedule is <= 1 all-reduce per steady-state epoch: epoch 0 (the
    eager warm epoch) issues grad + counts, epochs 1..N-1 issue grad
    only, plus the single final history reduce."""
    n_epochs = 10
    port = _free_port()
    ctxm = mp.get_context("spawn")
    out = ctxm.Queue()
    procs = [ctxm.Process(target=_kblocked_worker,
                          args=(r, 2, port, n_epochs, out)) for r in range(2)]
    for p in procs:
        p.start()
    W_dp, hist_dp, n_ar = out.get(timeout=240)
    for p in procs:
        p.join(timeout=120)
        assert p.exitcode == 0

    # collective schedule: warm epoch 2 (grad+counts) + (N-1) grads + 1 final
    assert n_ar == 2 + (n_epochs - 1) + 1

    ps = _pathset()
    cfg = G2VecConfig(hidden=64, epochs=n_epochs, early_stop=False, seed=4,
                      device="cpu", dtype="fp32")
    tr = CbowTrainer(cfg, ps.n_genes, torch.device("cpu"),
                     log=lambda *a, **k: None)
    res = tr.train(ps)
    assert hist_dp == pytest.approx(res.acc_val_history, abs=1e-6)
    assert np.allclose(W_dp, res.W_ih.numpy(), atol=1e-5)


def _pipelined_epochs_worker(rank, world, port, out):
    dist.init_process_group("gloo", rank=rank, world_size=world,
                            init_method=f"tcp://127.0.0.1:{port}")
    try:
        ctx = DistContext(rank, world, torch.device("cpu"), True)
        ps = _pathset()
        cfg = G2VecConfig(hidden=64, epochs=12, early_stop=True, seed=4,
                          device="cpu", dtype="fp32")
        tr = CbowTrainer(cfg, ps.n_genes, torch.device("cpu"), ctx,
                         log=lambda *a, **k: None)
        res_sync = tr.train(ps)             # CPU train() is the sync loop
        st = tr.setup(ps)
        hist, stop, W, _who, _ = tr.run_epochs_pipelined(
            st, cfg.epochs, early_stop=True)
        # pipelined DP epochs == sync DP epochs, on every rank
        assert hist == pytest.approx(res_sync.acc_val_history, abs=1e-6)
        assert stop == res_sync.stop_epoch
        assert torch.allclose(W, res_sync.W_ih, atol=1e-6)
        if rank == 0:
            out.put((np.asarray(W), hist))
    finally:
        dist.destroy_process_group()


@pytest.mark.timeout(300)
def test_dp2_pipelined_epochs_match_sync():
    """The speculative epoch pipeline under world_size=2 (the path the
    multi-GPU SCALE bench runs, minus RCCL): per-epoch collectives issued
    from queued epochs must give the sync loop's exact trajectory and
    keep-last-good weights."""
    port = _free_port()
    ctxm = mp.get_context("spawn")
    out = ctxm.Queue()
    procs = [ctxm.Process(target=_pipelined_epochs_worker,
                          args=(r, 2, port, out)) for r in range(2)]
    for p in procs:
        p.start()
    W_dp, hist_dp = out.get(timeout=240)
    for p in procs:
        p.join(timeout=120)
        assert p.exitcode == 0
    assert len(hist_dp) >= 2 and np.isfinite(W_dp).all()


def _kgranular_worker(rank, world, port, out):
    dist.init_process_group("gloo", rank=rank, world_size=world,
                            init_method=f"tcp://127.0.0.1:{port}")
    try:
        ctx = _CountingCtx(rank, world, torch.device("cpu"))
        ps = _pathset(G=60, P=260, seed=21)
        cfg = G2VecConfig(hidden=64, epochs=30, early_stop=True, seed=6,
                          device="cpu", dtype="fp32")
        tr = CbowTrainer(cfg, ps.n_genes, torch.device("cpu"), ctx,
                         log=lambda *a, **k: None)
        res_sync = tr.train(ps)            # CPU train() = sync loop
        tr2 = CbowTrainer(cfg, ps.n_genes, torch.device("cpu"), ctx,
                          log=lambda *a, **k: None)
        st = tr2.setup(ps)
        n0 = ctx.n_allreduce
        hist, stop, W, _who, _ = tr2.run_epochs_kgranular(st, cfg.epochs, 8)
        n_ar = ctx.n_allreduce - n0
        # exact parity with the per-epoch sync loop, on every rank
        assert stop == res_sync.stop_epoch
        assert hist == pytest.approx(res_sync.acc_val_history, abs=0)
        assert torch.allclose(W, res_sync.W_ih, atol=0)
        if rank == 0:
            out.put((stop, len(hist), n_ar))
    finally:
        dist.destroy_process_group()


@pytest.mark.timeout(300)
def test_dp2_kgranular_early_stop_parity_and_schedule():
    """k-granular early stop under world_size=2: bitwise the sync
    trajectory/stop/weights, with one metric all-reduce per 8-epoch
    block instead of one per epoch."""
    port = _free_port()
    ctxm = mp.get_context("spawn")
    out = ctxm.Queue()
    procs = [ctxm.Process(target=_kgranular_worker, args=(r, 2, port, out))
             for r in range(2)]
    for p in procs:
        p.start()
    stop, n_hist, n_ar = out.get(timeout=240)
    for p in procs:
        p.join(timeout=120)
        assert p.exitcode == 0
    assert stop >= 0 or n_hist == 30
    # schedule: warm epoch (grad+counts) + per-epoch grads + one metric
    # reduce per block (+ replay grads on stop) — strictly fewer than
    # the 2-per-epoch per-epoch schedule
    assert n_ar < 2 * n_hist


def _minibatch_uneven_worker(rank, world, port, out):
    dist.init_process_group("gloo", rank=rank, world_size=world,
                            init_method=f"tcp://127.0.0.1:{port}")
    try:
        ctx = DistContext(rank, world, torch.device("cpu"), True)
        # P=161: strided shards of the 128-path train split differ by one
        # path, so per-rank batch counts differ -> the lockstep padding
        # must keep the collective schedule aligned (no hang)
        ps = _pathset(G=40, P=161, seed=13)
        cfg = G2VecConfig(hidden=64, epochs=4, early_stop=False, seed=4,
                          device="cpu", dtype="fp32", batch_size=33)
        tr = CbowTrainer(cfg, ps.n_genes, torch.device("cpu"), ctx,
                         log=lambda *a, **k: None)
        res = tr.train(ps)
        assert np.isfinite(res.acc_val)
        if rank == 0:
            out.put(res.acc_val)
    finally:
        dist.destroy_process_group()


@pytest.mark.timeout(300)
def test_dp2_minibatch_uneven_shards_no_deadlock():
    port = _free_port()
    ctxm = mp.get_context("spawn")
    out = ctxm.Queue()
    procs = [ctxm.Process(target=_minibatch_uneven_worker,
                          args=(r, 2, port, out)) for r in range(2)]
    for p in procs:
        p.start()
    acc = out.get(timeout=240)
    for p in procs:
        p.join(timeout=120)
        assert p.exitcode == 0
    assert 0.0 <= acc <= 1.0


def _relabel_worker(rank, world, port, out):
    dist.init_process_group("gloo", rank=rank, world_size=world,
                            init_method=f"tcp://127.0.0.1:{port}")
    try:
        ctx = DistContext(rank, world, torch.device("cpu"), True)
        ps = _pathset(G=50, P=200, seed=17)
        cfg = G2VecConfig(hidden=64, epochs=8, early_stop=False, seed=4,
                          device="cpu", dtype="fp32", gene_relabel="on")
        tr = CbowTrainer(cfg, ps.n_genes, torch.device("cpu"), ctx,
                         log=lambda *a, **k: None)
        res = tr.train(ps)
        if rank == 0:
            out.put((res.W_ih.numpy(), res.acc_val_history))
    finally:
        dist.destroy_process_group()


@pytest.mark.timeout(300)
def test_dp2_gene_relabel_matches_single_process():
    """gene_relabel under DP: rank 0's first-touch order is broadcast so
    the c all-reduce lives in one id space — results must match the
    single-process relabeled run (and hence the unrelabeled one)."""
    port = _free_port()
    ctxm = mp.get_context("spawn")
    out = ctxm.Queue()
    procs = [ctxm.Process(target=_relabel_worker, args=(r, 2, port, out))
             for r in range(2)]
    for p in procs:
        p.start()
    W_dp, hist_dp = out.get(timeout=240)
    for p in procs:
        p.join(timeout=120)
        assert p.exitcode == 0
    ps = _pathset(G=50, P=200, seed=17)
    cfg = G2VecConfig(hidden=64, epochs=8, early_stop=False, seed=4,
                      device="cpu", dtype="fp32", gene_relabel="on")
    tr = CbowTrainer(cfg, ps.n_genes, torch.device("cpu"),
                     log=lambda *a, **k: None)
    res = tr.train(ps)
    assert hist_dp == pytest.approx(res.acc_val_history, abs=1e-6)
    assert np.allclose(W_dp, res.W_ih.numpy(), atol=1e-5)


def _general_worker(rank, world, port, out):
    dist.init_process_group("gloo", rank=rank, world_size=world,
                            init_method=f"tcp://127.0.0.1:{port}")
    try:
        ctx = DistContext(rank, world, torch.device("cpu"), True)
        ps = _pathset(G=40, P=150, seed=23)
        for act in ("none", "relu"):
            cfg = G2VecConfig(hidden=64, epochs=5, early_stop=False, seed=4,
                              device="cpu", trainer_path="general",
                              activation=act)
            tr = CbowTrainer(cfg, ps.n_genes, torch.device("cpu"), ctx,
                             log=lambda *a, **k: None)
            res = tr.train(ps)
            if rank == 0:
                out.put((act, res.W_ih.numpy(), res.acc_val_history))
    finally:
        dist.destroy_process_group()


@pytest.mark.timeout(300)
def test_dp2_general_path_matches_single_process():
    """The general kernel chain (dense dW + grad_who all-reduces) under
    world_size=2, linear AND relu — must match single-process."""
    port = _free_port()
    ctxm = mp.get_context("spawn")
    out = ctxm.Queue()
    procs = [ctxm.Process(target=_general_worker, args=(r, 2, port, out))
             for r in range(2)]
    for p in procs:
        p.start()
    got = {}
    for _ in range(2):
        act, W, hist = out.get(timeout=240)
        got[act] = (W, hist)
    for p in procs:
        p.join(timeout=120)
        assert p.exitcode == 0
    ps = _pathset(G=40, P=150, seed=23)
    for act in ("none", "relu"):
        cfg = G2VecConfig(hidden=64, epochs=5, early_stop=False, seed=4,
                          device="cpu", trainer_path="general",
                          activation=act)
        tr = CbowTrainer(cfg, ps.n_genes, torch.device("cpu"),
                         log=lambda *a, **k: None)
        res = tr.train(ps)
        W_dp, hist_dp = got[act]
        assert hist_dp == pytest.approx(res.acc_val_history, abs=1e-6), act
        assert np.allclose(W_dp, res.W_ih.numpy(), atol=1e-5), act


def _unseeded_worker(rank, world, port, q):
    dist.init_process_group("gloo", rank=rank, world_size=world,
                            init_method=f"tcp://127.0.0.1:{port}")
    try:
        ctx = DistContext(rank, world, torch.device("cpu"), True)
        ps = _pathset()
        cfg = G2VecConfig(hidden=64, epochs=4, early_stop=False, seed=None,
                          device="cpu", dtype="fp32")
        tr = CbowTrainer(cfg, ps.n_genes, torch.device("cpu"), ctx,
                         log=lambda *a, **k: None)
        res = tr.train(ps)
        q.put((rank, res.acc_val_history, res.W_ih.numpy()))
    finally:
        dist.destroy_process_group()


@pytest.mark.timeout(300)
def test_dp2_unseeded_ranks_agree():
    """seed=None at world>1: ranks must broadcast-agree on the global
    shuffle seed and the weight init (rank-0 broadcast), or the shards
    would overlap/miss paths and the trajectories diverge. Asserts both
    ranks return identical histories and weights (the reduced trajectory
    is global, so agreement proves the collectives lined up)."""
    port = _free_port()
    ctxm = mp.get_context("spawn")
    q = ctxm.Queue()
    procs = [ctxm.Process(target=_unseeded_worker, args=(r, 2, port, q))
             for r in range(2)]
    for p in procs:
        p.start()
    got = {}
    for _ in range(2):
        rank, hist, W = q.get(timeout=240)
        got[rank] = (hist, W)
    for p in procs:
        p.join(timeout=120)
        assert p.exitcode == 0
    assert got[0][0] == got[1][0]
    assert np.array_equal(got[0][1], got[1][1])
