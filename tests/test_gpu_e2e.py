"""Full-pipeline GPU integration at the ex_* scale (needs an MI355X)."""
import pytest
import torch

from g2vec_amd.config import G2VecConfig
from g2vec_amd.pipeline import run
from g2vec_amd.utils.synth import make_ex_style_files

pytestmark = pytest.mark.gpu


@pytest.mark.timeout(900)
@pytest.mark.parametrize("seed", [0, 1])
def test_ex_scale_pipeline_on_gpu(tmp_path, seed):
    """Statistical acceptance: seeded full-scale runs must
    clear the published val-ACC bar."""
    files = make_ex_style_files(str(tmp_path), n_genes=7523, n_extra=2381,
                                n_edges=298799, n_samples=135, n_poor=58,
                                n_modules=16, seed=seed)
    cfg = G2VecConfig(expression_file=files["expression"],
                      clinical_file=files["clinical"],
                      network_file=files["network"],
                      result_name=str(tmp_path / "out"),
                      len_path=80, num_repetition=10, epochs=500,
                      device="cuda", seed=seed)
    res = run(cfg)
    # README invariants that are deterministic + statistical bands
    assert res["n_samples"] == 135
    assert res["n_genes"] == 7523
    assert 30000 < res["n_paths"] < 80000
    assert 3000 < res["n_genes_in_paths"] < 4600
    assert res["acc_val"] >= 0.88          # the BASELINE.md headline bar
    # the hot phases must have run on the GPU in sane time
    t = res["timers"]
    assert t["walks_g0"] + t["walks_g1"] < 60.0
    assert t["train"] < 120.0


@pytest.mark.timeout(600)
def test_general_trainer_path_on_gpu(tmp_path):
    files = make_ex_style_files(str(tmp_path), n_genes=1000, n_extra=100,
                                n_edges=30000, n_samples=100, n_poor=43,
                                n_modules=8, seed=1)
    for dtype in ("fp32", "bf16"):
        cfg = G2VecConfig(expression_file=files["expression"],
                          clinical_file=files["clinical"],
                          network_file=files["network"],
                          result_name=str(tmp_path / f"out_{dtype}"),
                          len_path=40, num_repetition=5, epochs=20,
                          device="cuda", seed=0, trainer_path="general",
                          dtype=dtype)
        res = run(cfg)
        assert res["acc_val"] > 0.5


@pytest.mark.timeout(600)
def test_fast_equals_general_on_gpu(tmp_path):
    import numpy as np
    from g2vec_amd.models.cbow import CbowTrainer
    from g2vec_amd.paths import PathSet
    rng = np.random.default_rng(3)
    G, P = 200, 400
    genes, offs, labels = [], [0], []
    for _ in range(P):
        L = int(rng.integers(1, 15))
        genes += rng.choice(G, size=L, replace=False).tolist()
        offs.append(offs[-1] + L)
        labels.append(float(rng.integers(0, 2)))
    dev = torch.device("cuda", 0)
    ps = PathSet(torch.tensor(genes, dtype=torch.int32, device=dev),
                 torch.tensor(offs, dtype=torch.int32, device=dev),
                 torch.tensor(labels, device=dev), G)
    hists = {}
    for path in ("fast", "general"):
        cfg = G2VecConfig(hidden=128, epochs=6, early_stop=False, seed=5,
                          device="cuda", trainer_path=path, dtype="fp32")
        res = CbowTrainer(cfg, G, dev, log=lambda *a, **k: None).train(ps)
        hists[path] = res.acc_val_history
    assert hists["fast"] == pytest.approx(hists["general"], abs=1e-5)


@pytest.mark.timeout(600)
def test_hipgraph_matches_eager(tmp_path):
    """Graph-captured epochs must reproduce the eager trajectory exactly."""
    import numpy as np
    from g2vec_amd.models.cbow import CbowTrainer
    from g2vec_amd.paths import PathSet
    rng = np.random.default_rng(11)
    G, P = 300, 600
    genes, offs, labels = [], [0], []
    for _ in range(P):
        L = int(rng.integers(1, 20))
        genes += rng.choice(G, size=L, replace=False).tolist()
        offs.append(offs[-1] + L)
        labels.append(float(rng.integers(0, 2)))
    dev = torch.device("cuda", 0)
    ps = PathSet(torch.tensor(genes, dtype=torch.int32, device=dev),
                 torch.tensor(offs, dtype=torch.int32, device=dev),
                 torch.tensor(labels, device=dev), G)
    hists, weights = {}, {}
    for graphed in (True, False):
        cfg = G2VecConfig(hidden=128, epochs=10, early_stop=False, seed=3,
                          device="cuda", use_hipgraph=graphed)
        res = CbowTrainer(cfg, G, dev, log=lambda *a, **k: None).train(ps)
        hists[graphed] = res.acc_val_history
        weights[graphed] = res.W_ih
    assert hists[True] == hists[False]
    assert torch.equal(weights[True], weights[False])


@pytest.mark.gpu
def test_kblock_epochs_match_sync_loop(tmp_path):
    """The k-epoch block graph (fixed-epoch fast path) must reproduce the
    synchronous per-epoch loop's accuracy trajectory exactly."""
    from g2vec_amd.models.cbow import CbowTrainer
    from g2vec_amd.paths import PathSet
    import numpy as np

    rng = np.random.default_rng(11)
    G, P = 400, 3000
    genes, offs, labels = [], [0], []
    for _ in range(P):
        L = int(rng.integers(2, 14))
        genes += rng.choice(G, size=L, replace=False).tolist()
        offs.append(offs[-1] + L)
        labels.append(float(rng.integers(0, 2)))
    dev = torch.device("cuda")
    ps = PathSet(torch.tensor(genes, dtype=torch.int32, device=dev),
                 torch.tensor(offs, dtype=torch.int32, device=dev),
                 torch.tensor(labels, device=dev), G)
    N_EP = 29    # 1 warm + 3 blocks of 8 + 4 eager tail epochs:
                 # covers the replay loop AND the deferred tail
    cfg = G2VecConfig(hidden=64, epochs=N_EP, early_stop=False, seed=3,
                      device="cuda")
    tr1 = CbowTrainer(cfg, G, dev, log=lambda *a, **k: None)
    st1 = tr1.setup(ps)
    hist_sync = [tr1.run_epoch(st1)[1] for _ in range(N_EP)]
    tr2 = CbowTrainer(cfg, G, dev, log=lambda *a, **k: None)
    st2 = tr2.setup(ps)
    hist_k, stop, W, _who, _ = tr2.run_epochs_pipelined(
        st2, N_EP, early_stop=False)
    assert getattr(st2, "kgraph", None) is not None, "k-block graph not used"
    assert stop == -1 and len(hist_k) == N_EP
    assert hist_k == pytest.approx(hist_sync, abs=1e-6)
    assert torch.allclose(W, st1.W, atol=1e-6)


@pytest.mark.timeout(600)
def test_kgranular_early_stop_matches_per_epoch_on_gpu():
    """--earlystop-every on GPU (hipGraph blocks + snapshot/replay) must
    reproduce the per-epoch pipelined runner's trajectory, stop epoch,
    and weights bitwise."""
    import numpy as np
    from g2vec_amd.models.cbow import CbowTrainer
    from g2vec_amd.paths import PathSet
    rng = np.random.default_rng(23)
    G, P = 250, 500
    genes, offs, labels = [], [0], []
    for _ in range(P):
        L = int(rng.integers(1, 14))
        genes += rng.choice(G, size=L, replace=False).tolist()
        offs.append(offs[-1] + L)
        labels.append(float(rng.integers(0, 2)))
    dev = torch.device("cuda", 0)
    ps = PathSet(torch.tensor(genes, dtype=torch.int32, device=dev),
                 torch.tensor(offs, dtype=torch.int32, device=dev),
                 torch.tensor(labels, device=dev), G)
    cfg = G2VecConfig(hidden=128, epochs=40, early_stop=True, seed=8,
                      device="cuda", dtype="fp32")
    ref = CbowTrainer(cfg, G, dev, log=lambda *a, **k: None)
    res = ref.train(ps)                       # per-epoch pipelined runner
    tr = CbowTrainer(cfg, G, dev, log=lambda *a, **k: None)
    st = tr.setup(ps)
    hist, stop, W, _who, _ = tr.run_epochs_kgranular(st, cfg.epochs, 8)
    assert stop == res.stop_epoch
    assert hist == pytest.approx(res.acc_val_history, abs=0)
    assert torch.equal(W.cpu(), res.W_ih.cpu())


@pytest.mark.timeout(600)
def test_rccl_allreduce_capturable_in_hipgraph():
    """The SCALE path captures the grad all-reduce inside epoch graphs
    at world>1 (behind DistContext.graph_capture_ok). Multi-rank needs
    multiple GPUs, but a WORLD=1 RCCL process group still launches real
    RCCL kernels — capture one into a hipGraph, replay it, and verify:
    this exercises the same RCCL stream-capture machinery the probe
    relies on."""
    import socket

    import torch.distributed as dist
    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    port = s.getsockname()[1]
    s.close()
    dist.init_process_group("nccl", rank=0, world_size=1,
                            init_method=f"tcp://127.0.0.1:{port}")
    try:
        t = torch.ones(64, device="cuda")
        dist.all_reduce(t)              # connect communicator
        torch.cuda.synchronize()
        g = torch.cuda.CUDAGraph()
        with torch.cuda.graph(g, capture_error_mode="thread_local"):
            dist.all_reduce(t)
            t.mul_(2.0)
        for _ in range(3):
            g.replay()
        torch.cuda.synchronize()
        assert torch.allclose(t, torch.full_like(t, 8.0))
    finally:
        dist.destroy_process_group()


@pytest.mark.timeout(600)
def test_train_state_resume_bitwise_on_gpu(tmp_path):
    """Mid-training checkpoint/resume on GPU: exact continuation."""
    import numpy as np
    from g2vec_amd.models.cbow import CbowTrainer
    from g2vec_amd.paths import PathSet
    rng = np.random.default_rng(41)
    G, P = 200, 400
    genes, offs, labels = [], [0], []
    for _ in range(P):
        L = int(rng.integers(1, 15))
        genes += rng.choice(G, size=L, replace=False).tolist()
        offs.append(offs[-1] + L)
        labels.append(float(rng.integers(0, 2)))
    dev = torch.device("cuda", 0)
    ps = PathSet(torch.tensor(genes, dtype=torch.int32, device=dev),
                 torch.tensor(offs, dtype=torch.int32, device=dev),
                 torch.tensor(labels, device=dev), G)

    def mk(**kw):
        base = dict(hidden=128, epochs=16, early_stop=False, seed=6,
                    device="cuda")
        base.update(kw)
        return G2VecConfig(**base)

    full = CbowTrainer(mk(), G, dev, log=lambda *a, **k: None).train(ps)
    ck = str(tmp_path / "gs.pt")
    CbowTrainer(mk(epochs=8, train_ckpt=ck, train_ckpt_every=8), G, dev,
                log=lambda *a, **k: None).train(ps)
    resumed = CbowTrainer(mk(resume_train=ck), G, dev,
                          log=lambda *a, **k: None).train(ps)
    assert resumed.acc_val_history[8:] == pytest.approx(
        full.acc_val_history[8:], abs=0)
    assert torch.equal(resumed.W_ih.cpu(), full.W_ih.cpu())


@pytest.mark.timeout(600)
def test_cli_end_to_end_on_gpu(tmp_path):
    """The user-facing CLI on cuda: `python -m g2vec_amd ...` end to end,
    reference transcript + output triple."""
    import subprocess
    import sys as _sys

    from g2vec_amd.utils.synth import make_ex_style_files
    files = make_ex_style_files(str(tmp_path), n_genes=800, n_extra=80,
                                n_edges=20000, n_samples=100, n_poor=43,
                                n_modules=8, seed=2)
    r = subprocess.run(
        [_sys.executable, "-m", "g2vec_amd", files["expression"],
         files["clinical"], files["network"], str(tmp_path / "out"),
         "-p", "30", "-r", "3", "-e", "25", "--seed", "0",
         "--device", "cuda"],
        capture_output=True, text=True, timeout=500)
    assert r.returncode == 0, r.stderr[-2000:]
    assert "Optimization Finish" in r.stdout
    assert ">>> 7. Save results" in r.stdout
    for sfx in ("_biomarkers.txt", "_lgroups.txt", "_vectors.txt"):
        assert (tmp_path / f"out{sfx}").exists()
