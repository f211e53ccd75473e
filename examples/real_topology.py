#!/usr/bin/env python3
"""Run the full pipeline on the REAL bundled reference data.

Parses the reference's real ex_NETWORK.txt (9,904 genes, 298,799 edges)
and ex_CLINICAL.txt (135 samples) — from /root/reference when mounted,
else from the committed cache — and synthesizes a seeded expression
matrix over the published 7,523-gene intersection (the original
ex_EXPRESSION.txt is not distributable). Reproduces the published run's
count invariants and trains to val-ACC ~0.88 on the real topology.

    python examples/real_topology.py [outdir]   # CPU or GPU (auto)
"""
import sys

sys.path.insert(0, __file__.rsplit("/", 2)[0])

from g2vec_amd.config import G2VecConfig
from g2vec_amd.pipeline import run
from g2vec_amd.utils import refdata


def main():
    outdir = sys.argv[1] if len(sys.argv) > 1 else "./real_out"
    print(f"materializing the real-topology dataset under {outdir} ...")
    files = refdata.write_dataset_files(outdir, seed=0)
    cfg = G2VecConfig(expression_file=files["expression"],
                      clinical_file=files["clinical"],
                      network_file=files["network"],
                      result_name=f"{outdir}/result",
                      seed=0)
    res = run(cfg)
    print(f"\nn_samples={res['n_samples']} n_genes={res['n_genes']} "
          f"n_paths={res['n_paths']} val_acc={res['acc_val']:.4f}")


if __name__ == "__main__":
    main()
