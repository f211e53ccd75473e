"""bench.py contract test: the driver depends on the one-JSON-line stdout
protocol and its field set — run the real script as a subprocess on a tiny
config and validate the schema."""
import json
import socket
import subprocess
import sys

import pytest


@pytest.mark.timeout(600)
def test_bench_json_contract(tmp_path):
    out = subprocess.run(
        [sys.executable, "bench.py", "--n-genes", "500", "--n-edges", "6000",
         "--n-extra", "50", "--n-modules", "6", "--reps", "2",
         "--len-path", "12", "--steps", "2", "--warmup", "1",
         "--acc-target-epochs", "2"],
        capture_output=True, text=True, timeout=500)
    assert out.returncode == 0, out.stderr[-2000:]
    lines = [l for l in out.stdout.strip().splitlines() if l.strip()]
    assert len(lines) == 1, f"stdout must be ONE JSON line, got: {lines}"
    d = json.loads(lines[0])

    for key in ("metric", "value", "unit", "n_gpus", "steps", "warmup",
                "ms_per_step", "higher_is_better", "scaling", "vs_baseline",
                "dtype", "data", "config"):
        assert key in d, key
    assert d["metric"] == "cbow_paths_per_sec"
    assert d["higher_is_better"] is True
    assert d["scaling"] == "weak"
    assert d["n_gpus"] == 1 and d["steps"] == 2 and d["warmup"] == 1
    assert d["value"] > 0 and d["ms_per_step"] > 0
    assert d["vs_baseline"] == pytest.approx(d["value"] / 16500.0, rel=0.01)
    cfg = d["config"]
    for key in ("model", "global_batch", "seq_len", "parallelism", "val_acc",
                "walks_per_sec"):
        assert key in cfg, key
    assert cfg["parallelism"] == "dp1"
    assert "synthetic" in d["data"]


def _free_port() -> int:
    with socket.socket() as s:
        s.bind(("127.0.0.1", 0))
        return s.getsockname()[1]


@pytest.mark.timeout(600)
def test_bench_torchrun_dp2_contract():
    """The driver's SCALE launch path: torch.distributed.run with 2 ranks
    (gloo on CPU) must still produce ONE valid JSON line from rank 0."""
    out = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
         "--master-port", str(_free_port()), "bench.py", "--gpus", "2",
         "--n-genes", "500", "--n-edges", "6000", "--n-extra", "50",
         "--n-modules", "6", "--reps", "2", "--len-path", "12",
         "--steps", "2", "--warmup", "1", "--acc-target-epochs", "2"],
        capture_output=True, text=True, timeout=500)
    assert out.returncode == 0, out.stderr[-2000:]
    lines = [l for l in out.stdout.strip().splitlines()
             if l.strip().startswith("{")]
    assert len(lines) == 1, f"exactly one JSON line expected: {lines}"
    d = json.loads(lines[0])
    assert d["n_gpus"] == 2 and d["config"]["parallelism"] == "dp2"
    assert d["value"] > 0
    # C5 weak scaling: one shared study at num_repetition = reps x world;
    # the global batch must come from the GLOBAL dedup of both ranks'
    # source-sharded walks (reps x world > reps of a single rank's set)
    assert d["config"]["num_repetition_global"] == 4
    single = subprocess.run(
        [sys.executable, "bench.py",
         "--n-genes", "500", "--n-edges", "6000", "--n-extra", "50",
         "--n-modules", "6", "--reps", "2", "--len-path", "12",
         "--steps", "2", "--warmup", "1", "--acc-target-epochs", "2"],
        capture_output=True, text=True, timeout=400)
    assert single.returncode == 0, single.stderr[-2000:]
    d1 = json.loads([l for l in single.stdout.strip().splitlines()
                     if l.strip().startswith("{")][0])
    # same study at 2x the walk budget: strictly more unique paths, and
    # roughly linear growth (dedup saturation would show up here)
    assert d["config"]["global_batch"] > 1.5 * d1["config"]["global_batch"]
