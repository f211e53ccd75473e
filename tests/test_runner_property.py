"""Property-based checks of the epoch runners (hypothesis): for random
path sets, epoch budgets, granularities and seeds, every runner variant
must reproduce the synchronous reference loop exactly."""
import os

import numpy as np
import pytest
import torch
from hypothesis import given, settings
from hypothesis import strategies as st

_SOAK = max(int(os.environ.get("G2VEC_SOAK", "1")), 1)  # soak runs scale the example budget

from g2vec_amd.config import G2VecConfig
from g2vec_amd.models.cbow import CbowTrainer
from g2vec_amd.paths import PathSet


def _pathset(G, P, seed):
    rng = np.random.default_rng(seed)
    genes, offs, labels = [], [0], []
    for _ in range(P):
        L = int(rng.integers(1, min(10, G)))
        genes += rng.choice(G, size=L, replace=False).tolist()
        offs.append(offs[-1] + L)
        labels.append(float(rng.integers(0, 2)))
    return PathSet(torch.tensor(genes, dtype=torch.int32),
                   torch.tensor(offs, dtype=torch.int32),
                   torch.tensor(labels), G)


def _train_sync(cfg, ps):
    tr = CbowTrainer(cfg, ps.n_genes, torch.device("cpu"),
                     log=lambda *a, **k: None)
    return tr, tr.train(ps)


@settings(max_examples=12 * _SOAK, deadline=None)
@given(seed=st.integers(0, 50), epochs=st.integers(2, 25),
       k=st.integers(2, 9), data_seed=st.integers(0, 20))
def test_kgranular_property(seed, epochs, k, data_seed):
    ps = _pathset(40, 120, data_seed)
    cfg = G2VecConfig(hidden=64, epochs=epochs, early_stop=True, seed=seed,
                      device="cpu", dtype="fp32")
    ref_tr, ref = _train_sync(cfg, ps)
    tr = CbowTrainer(cfg, ps.n_genes, torch.device("cpu"),
                     log=lambda *a, **k2: None)
    st_ = tr.setup(ps)
    hist, stop, W, _who, _ = tr.run_epochs_kgranular(st_, epochs, k)
    assert stop == ref.stop_epoch
    assert hist == pytest.approx(ref.acc_val_history, abs=0)
    assert torch.equal(W, ref.W_ih)


@settings(max_examples=10 * _SOAK, deadline=None)
@given(seed=st.integers(0, 50), epochs=st.integers(3, 20),
       cut=st.floats(0.2, 0.8), data_seed=st.integers(0, 20))
def test_checkpoint_resume_property(tmp_path_factory, seed, epochs, cut,
                                    data_seed):
    ps = _pathset(40, 110, data_seed)
    base = dict(hidden=64, epochs=epochs, early_stop=False, seed=seed,
                device="cpu")
    _, full = _train_sync(G2VecConfig(**base), ps)
    ck = str(tmp_path_factory.mktemp("ck") / "s.pt")
    at = max(1, min(epochs - 1, int(epochs * cut)))
    _train_sync(G2VecConfig(**{**base, "epochs": at, "train_ckpt": ck,
                               "train_ckpt_every": at}), ps)
    _, resumed = _train_sync(G2VecConfig(**{**base, "resume_train": ck}), ps)
    assert resumed.acc_val_history == pytest.approx(full.acc_val_history,
                                                    abs=0)
    assert torch.equal(resumed.W_ih, full.W_ih)


@settings(max_examples=10 * _SOAK, deadline=None)
@given(seed=st.integers(0, 50), epochs=st.integers(2, 18),
       data_seed=st.integers(0, 20))
def test_pipelined_property(seed, epochs, data_seed):
    ps = _pathset(40, 120, data_seed)
    cfg = G2VecConfig(hidden=64, epochs=epochs, early_stop=True, seed=seed,
                      device="cpu", dtype="fp32")
    _, ref = _train_sync(cfg, ps)
    tr = CbowTrainer(cfg, ps.n_genes, torch.device("cpu"),
                     log=lambda *a, **k: None)
    st_ = tr.setup(ps)
    hist, stop, W, _who, _ = tr.run_epochs_pipelined(st_, epochs, True)
    assert stop == ref.stop_epoch
    assert hist == pytest.approx(ref.acc_val_history, abs=0)
    assert torch.equal(W, ref.W_ih)
