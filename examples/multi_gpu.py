#!/usr/bin/env python3
"""Data-parallel run on one node (1-8 MI355X GPUs over RCCL/xGMI).

Generates a seeded synthetic dataset triple, then launches the standard
CLI under torch.distributed.run with one rank per GPU:

    python examples/multi_gpu.py [outdir] [nproc]

Each rank walks its shard of the source genes (the walk shards are
all-gathered for global dedup — C5), training shards the path set with
a single fused grad all-reduce per epoch, and rank 0 writes the output
triple (bitwise-deterministic discrete outputs; the vectors file can
differ from a single-GPU run in low decimal digits only, from the fp32
all-reduce summation order — see profiles/dp2_rehearsal.md).

On a machine without GPUs this still runs: ranks fall back to CPU and
the gloo backend (same code path the CI covers at world_size=2).
"""
import os
import subprocess
import sys

sys.path.insert(0, __file__.rsplit("/", 2)[0])

from g2vec_amd.utils.synth import make_ex_style_files


def main() -> int:
    outdir = sys.argv[1] if len(sys.argv) > 1 else "mgpu_out"
    nproc = int(sys.argv[2]) if len(sys.argv) > 2 else 2
    os.makedirs(outdir, exist_ok=True)
    files = make_ex_style_files(outdir, n_genes=800, n_extra=80,
                                n_edges=20000, n_samples=100, n_poor=43,
                                n_modules=8, seed=2)
    cmd = [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
           "--nproc-per-node", str(nproc),
           "--master-addr", "127.0.0.1", "--master-port", "29533",
           "-m", "g2vec_amd", files["expression"], files["clinical"],
           files["network"], os.path.join(outdir, "result"),
           "-p", "30", "-r", "3", "-e", "25", "--seed", "0"]
    print("launching:", " ".join(cmd), flush=True)
    return subprocess.call(cmd)


if __name__ == "__main__":
    raise SystemExit(main())
