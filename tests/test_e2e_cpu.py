"""End-to-end pipeline on tiny synthetic ex_*-style files (integration tier,
SURVEY §4.3)."""
import os

import numpy as np
import pytest
import torch

from g2vec_amd.config import G2VecConfig
from g2vec_amd.pipeline import run


def _cfg(tiny_files, tmp_path, **kw):
    base = dict(expression_file=tiny_files["expression"],
                clinical_file=tiny_files["clinical"],
                network_file=tiny_files["network"],
                result_name=str(tmp_path / "out"),
                len_path=15, num_repetition=3, epochs=25, device="cpu",
                seed=0)
    base.update(kw)
    return G2VecConfig(**base)


def test_pipeline_end_to_end(tiny_files, tmp_path):
    res = run(_cfg(tiny_files, tmp_path))
    assert res["n_samples"] == 80
    assert res["n_genes"] == 300
    assert res["n_paths"] > 50
    assert 0 < res["n_genes_in_paths"] <= 300
    assert 0.4 <= res["acc_val"] <= 1.0

    # output files exist and parse in the documented formats
    bio = (tmp_path / "out_biomarkers.txt").read_text().splitlines()
    assert bio[0] == "GeneSymbol" and len(bio) > 1
    assert bio[1:] == sorted(bio[1:])

    lg = (tmp_path / "out_lgroups.txt").read_text().splitlines()
    assert lg[0] == "GeneSymbol\tLgroup(0:good,1:poor,2:other)"
    assert len(lg) == 1 + res["n_genes"]
    vals = {line.split("\t")[1] for line in lg[1:]}
    assert vals <= {"0", "1", "2"}

    vec = (tmp_path / "out_vectors.txt").read_text().splitlines()
    assert vec[0].startswith("GeneSymbol\tV0\t")
    assert len(vec) == 1 + res["n_genes"]
    assert len(vec[1].split("\t")) == 1 + 128


def test_pipeline_compat_bug_flag(tiny_files, tmp_path):
    res = run(_cfg(tiny_files, tmp_path, compat_lgroup_bug=True))
    assert set(np.unique(res["lgroups"])) <= {0, 1, 2}


def test_save_load_paths(tiny_files, tmp_path):
    cache = str(tmp_path / "paths.pt")
    r1 = run(_cfg(tiny_files, tmp_path, save_paths=cache))
    assert os.path.exists(cache)
    r2 = run(_cfg(tiny_files, tmp_path, load_paths=cache))
    assert r2["n_paths"] == r1["n_paths"]
    assert r2["acc_val"] == pytest.approx(r1["acc_val"], abs=1e-6)


def test_cli_surface(tiny_files, tmp_path):
    from g2vec_amd.cli import args_to_config, build_parser
    argv = [tiny_files["expression"], tiny_files["clinical"],
            tiny_files["network"], str(tmp_path / "o"),
            "-p", "20", "-r", "2", "-s", "64", "-e", "10", "-l", "0.01",
            "-n", "7", "--device", "cpu", "--compat-lgroup-bug",
            "--gene-relabel", "off", "--trainer-path", "general",
            "--activation", "relu", "--train-ckpt", "ck.pt",
            "--train-ckpt-every", "3", "--earlystop-every", "4"]
    cfg = args_to_config(build_parser().parse_args(argv))
    assert cfg.len_path == 20 and cfg.num_repetition == 2
    assert cfg.hidden == 64 and cfg.epochs == 10
    assert cfg.lr == 0.01 and cfg.num_biomarker == 7
    assert cfg.compat_lgroup_bug
    assert cfg.gene_relabel == "off" and cfg.activation == "relu"
    assert cfg.train_ckpt == "ck.pt" and cfg.train_ckpt_every == 3
    assert cfg.earlystop_every == 4
    cfg.validate()


def test_jsonl_metrics(tiny_files, tmp_path):
    log = str(tmp_path / "m.jsonl")
    run(_cfg(tiny_files, tmp_path, log_jsonl=log))
    import json
    with open(log) as fh:
        events = [json.loads(l) for l in fh]
    names = {e["event"] for e in events}
    assert {"counts", "paths", "train", "phase"} <= names


def test_unseeded_mode_runs(tiny_files, tmp_path):
    """seed=None reproduces the reference's nondeterministic behaviour
    without crashing (manual.pdf p.4 documents run-to-run variation)."""
    res = run(_cfg(tiny_files, tmp_path, seed=None, epochs=5,
                   num_repetition=1))
    assert res["n_paths"] > 0


def test_missing_file_raises(tmp_path):
    cfg = G2VecConfig(expression_file=str(tmp_path / "nope.tsv"),
                      clinical_file=str(tmp_path / "nope2.tsv"),
                      network_file=str(tmp_path / "nope3.tsv"),
                      result_name=str(tmp_path / "out"), device="cpu")
    with pytest.raises(Exception):
        run(cfg)


def test_console_transcript_format(tiny_files, tmp_path, capsys):
    """The phase-numbered console lines ARE the reference's published
    baseline format (README.md:22-49) — keep them shaped identically."""
    import re
    run(_cfg(tiny_files, tmp_path, epochs=6))
    out = capsys.readouterr().out
    for pat in (r">>> 1\. Load data",
                r">>> 2\. Preprocess data",
                r"    n_samples: \d+",
                r"    n_genes  : \d+\t\(common genes in both EXPRESSION and NETWORK\)",
                r"    n_edges  : \d+\t\(edges with the common genes\)",
                r">>> 3\. Generate random paths from each group",
                r"    \*\*\* most time consuming step \*\*\*",
                r"    n_paths : \d+",
                r">>> 4\. Compute distributed representations using modified CBOW",
                r"     Start training the modified CBOW with early stopping",
                r"    - Epoch: 000\tACC\[val\]=0\.\d{4}\tACC\[tr\]=0\.\d{4} \(\d+\.\d{3} sec\)",
                r"    Optimization Finish",
                r">>> 5\. Find L-groups",
                r">>> 6\. Select biomarkers with gene scores",
                r">>> 7\. Save results"):
        assert re.search(pat, out), pat


def test_save_model_checkpoint(tiny_files, tmp_path):
    ckpt = str(tmp_path / "model.pt")
    res = run(_cfg(tiny_files, tmp_path, save_model=ckpt, epochs=5))
    blob = torch.load(ckpt, weights_only=False)
    assert blob["W_ih"].shape == (res["n_genes"], 128)
    assert blob["acc_val"] == res["acc_val"]
    assert len(blob["gene_index"]) == res["n_genes"]


def test_config_validation_rejects_bad_values():
    import pytest as _pt

    from g2vec_amd.config import G2VecConfig

    ok = dict(expression_file="e", clinical_file="c", network_file="n",
              result_name="r")
    G2VecConfig(**ok).validate()
    for bad in (dict(hidden=100), dict(dtype="int8"), dict(len_path=0),
                dict(len_path=1000), dict(pcc_mode="dense"),
                dict(trainer_path="medium"), dict(kmeans_backend="cuml"),
                dict(epochs=0), dict(epochs=-3)):
        with _pt.raises(ValueError):
            G2VecConfig(**{**ok, **bad}).validate()


def test_pathset_nnz_int32_guard():
    """Total gene instances >= 2^31 must fail loudly before the silent
    int32 wrap in the offsets cast (paths.py)."""
    import pytest as _pt
    import torch

    from g2vec_amd.paths import PathSet, _check_i32_nnz, subset

    _check_i32_nnz(2 ** 31 - 1)          # just under: fine
    with _pt.raises(OverflowError):
        _check_i32_nnz(2 ** 31)

    # subset() path: fake offsets whose selected lengths overflow int32
    big = 2 ** 30
    ps = PathSet(genes=torch.zeros(4, dtype=torch.int32),
                 offsets=torch.tensor([0, big, 2 * big, 3 * big],
                                      dtype=torch.int64),
                 labels=torch.zeros(3), n_genes=10)
    with _pt.raises(OverflowError):
        subset(ps, torch.tensor([0, 1]))


def test_load_model_resumes_without_training(tiny_files, tmp_path):
    """--save-model then --load-model: steps 5-7 reproduce the same
    biomarkers without retraining; shape mismatch raises."""
    ckpt = str(tmp_path / "w.pt")
    res1 = run(_cfg(tiny_files, tmp_path, save_model=ckpt))
    out2 = tmp_path / "resume"
    out2.mkdir()
    res2 = run(_cfg(tiny_files, out2, load_model=ckpt))
    assert res2["biomarkers"] == res1["biomarkers"]
    assert np.allclose(res2["W_ih"], res1["W_ih"])
    bad = _cfg(tiny_files, tmp_path, load_model=ckpt)
    bad.hidden = 256
    with pytest.raises(ValueError, match="load-model"):
        run(bad)


def test_earlystop_every_cli_plumb(tiny_files, tmp_path):
    """--earlystop-every K must plumb through the CLI and reproduce the
    per-epoch early-stop results exactly (k-granular replay semantics),
    never acting as a silent no-op."""
    import subprocess
    import sys as _sys

    outs = {}
    for tag, extra in (("k1", []), ("k4", ["--earlystop-every", "4"])):
        r = subprocess.run(
            [_sys.executable, "-m", "g2vec_amd",
             tiny_files["expression"], tiny_files["clinical"],
             tiny_files["network"], str(tmp_path / tag),
             "-p", "12", "-r", "2", "-e", "30", "--seed", "0",
             "--device", "cpu"] + extra,
            capture_output=True, text=True, timeout=300)
        assert r.returncode == 0, r.stderr[-1500:]
        outs[tag] = r.stdout
        assert (tmp_path / f"{tag}_vectors.txt").exists()
    # identical epoch transcript lines (trajectory + stop epoch)
    lines1 = [l for l in outs["k1"].splitlines() if "Epoch" in l]
    lines4 = [l for l in outs["k4"].splitlines() if "Epoch" in l]
    def strip_t(ls):
        import re
        return [re.sub(r"\(\d+\.\d+ sec\)", "", l) for l in ls]
    assert strip_t(lines1) == strip_t(lines4)
    v1 = (tmp_path / "k1_vectors.txt").read_text()
    v4 = (tmp_path / "k4_vectors.txt").read_text()
    assert v1 == v4


def test_degenerate_inputs_fail_loudly(tiny_files, tmp_path):
    """Empty prognosis group and empty path set raise typed errors with
    actionable messages instead of nan-propagating (the reference's
    behavior) or crashing obscurely."""
    # (a) one-class clinical file
    bad_cli = tmp_path / "bad_CLINICAL.txt"
    lines = open(tiny_files["clinical"]).read().splitlines()
    with open(bad_cli, "w") as f:
        f.write(lines[0] + "\n")
        for ln in lines[1:]:
            f.write(ln.split("\t")[0] + "\t0\n")
    cfg = _cfg(tiny_files, tmp_path, clinical_file=str(bad_cli))
    with pytest.raises(ValueError, match="prognosis group"):
        run(cfg)

    # (b) empty path set reaching the trainer
    from g2vec_amd.models.cbow import CbowTrainer
    from g2vec_amd.paths import PathSet
    ps = PathSet(torch.zeros(0, dtype=torch.int32),
                 torch.zeros(1, dtype=torch.int32),
                 torch.zeros(0), 10)
    tr = CbowTrainer(G2VecConfig(hidden=64, device="cpu"), 10,
                     torch.device("cpu"), log=lambda *a, **k: None)
    with pytest.raises(ValueError, match="no paths"):
        tr.train(ps)


def test_failure_recovery_paths_cache_plus_train_resume(tiny_files, tmp_path):
    """The full failure-recovery story: a run that
    dies mid-training resumes from the step-3 path cache + the
    mid-training state file and produces byte-identical outputs to an
    uninterrupted run."""
    paths_cache = str(tmp_path / "paths.pt")
    ck = str(tmp_path / "state.pt")
    base = dict(len_path=15, num_repetition=3, device="cpu", seed=0,
                early_stop=False)

    # uninterrupted 12-epoch run (also writes the path cache)
    full = run(_cfg(tiny_files, tmp_path, epochs=12, save_paths=paths_cache,
                    result_name=str(tmp_path / "full"), **base))

    # "crashed" run: 6 epochs, checkpoint at 6
    run(_cfg(tiny_files, tmp_path, epochs=6, load_paths=paths_cache,
             train_ckpt=ck, train_ckpt_every=6,
             result_name=str(tmp_path / "crash"), **base))
    # recovery: same path cache + training state, continue to 12
    rec = run(_cfg(tiny_files, tmp_path, epochs=12, load_paths=paths_cache,
                   resume_train=ck,
                   result_name=str(tmp_path / "rec"), **base))
    assert rec["acc_val"] == pytest.approx(full["acc_val"], abs=0)
    assert (tmp_path / "rec_vectors.txt").read_text() == \
           (tmp_path / "full_vectors.txt").read_text()
    assert (tmp_path / "rec_biomarkers.txt").read_text() == \
           (tmp_path / "full_biomarkers.txt").read_text()


def test_compare_outputs_tool(tiny_files, tmp_path):
    """tools/compare_outputs.py: exit 0 on identical triples, 1 on a
    perturbed one, with a per-file report."""
    import subprocess
    import sys as _sys

    run(_cfg(tiny_files, tmp_path, result_name=str(tmp_path / "a"), epochs=4))
    run(_cfg(tiny_files, tmp_path, result_name=str(tmp_path / "b"), epochs=4))
    r = subprocess.run([_sys.executable, "tools/compare_outputs.py",
                        str(tmp_path / "a"), str(tmp_path / "b")],
                       capture_output=True, text=True, timeout=120)
    assert r.returncode == 0, r.stdout + r.stderr
    assert "MATCH" in r.stdout
    # perturb one lgroup assignment
    p = tmp_path / "b_lgroups.txt"
    lines = p.read_text().splitlines()
    g, v = lines[1].split("\t")
    lines[1] = f"{g}\t{(int(v) + 1) % 3}"
    p.write_text("\n".join(lines) + "\n")
    r = subprocess.run([_sys.executable, "tools/compare_outputs.py",
                        str(tmp_path / "a"), str(tmp_path / "b")],
                       capture_output=True, text=True, timeout=120)
    assert r.returncode == 1
    assert "DIFFER" in r.stdout


def test_golden_output_triple(tmp_path):
    """Byte-exact regression pin: the seeded tiny pipeline reproduces
    the committed golden triple (tests/golden/). Guards every layer that
    shapes outputs — parsers, graph, walks, trainer numerics, L-groups,
    scoring, writers. If an INTENTIONAL semantic change breaks this,
    regenerate the goldens with the snippet in tests/golden/README."""
    from g2vec_amd.utils.synth import make_ex_style_files
    files = make_ex_style_files(str(tmp_path), n_genes=120, n_extra=20,
                                n_edges=2500, n_samples=60, n_poor=26,
                                n_modules=5, seed=9)
    cfg = G2VecConfig(expression_file=files["expression"],
                      clinical_file=files["clinical"],
                      network_file=files["network"],
                      result_name=str(tmp_path / "g"),
                      len_path=12, num_repetition=3, hidden=64, epochs=15,
                      num_biomarker=10, seed=7, device="cpu")
    run(cfg)
    import pathlib
    golden = pathlib.Path(__file__).parent / "golden"
    for sfx in ("biomarkers", "lgroups", "vectors"):
        got = (tmp_path / f"g_{sfx}.txt").read_text()
        want = (golden / f"tiny_{sfx}.txt").read_text()
        assert got == want, f"{sfx} drifted from the golden pin"
