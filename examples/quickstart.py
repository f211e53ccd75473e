#!/usr/bin/env python3
"""Quickstart: generate an example dataset (the reference's bundled
ex_EXPRESSION.txt is not distributable, so a seeded synthetic triple with
the same structure is generated locally) and run the full pipeline.

    python examples/quickstart.py [outdir]   # CPU or GPU (auto)

Equivalent CLI run afterwards:
    python -m g2vec_amd <outdir>/syn_EXPRESSION.txt <outdir>/syn_CLINICAL.txt \
        <outdir>/syn_NETWORK.txt <outdir>/result -p 80 -r 10 -s 128
"""
import sys

sys.path.insert(0, __file__.rsplit("/", 2)[0])

from g2vec_amd.config import G2VecConfig
from g2vec_amd.pipeline import run
from g2vec_amd.utils.synth import make_ex_style_files


def main():
    outdir = sys.argv[1] if len(sys.argv) > 1 else "./quickstart_out"
    print(f"generating example dataset under {outdir} ...")
    files = make_ex_style_files(outdir, n_genes=2000, n_extra=300,
                                n_edges=80000, n_samples=135, n_poor=58,
                                n_modules=8, seed=0)
    cfg = G2VecConfig(expression_file=files["expression"],
                      clinical_file=files["clinical"],
                      network_file=files["network"],
                      result_name=f"{outdir}/result",
                      len_path=80, num_repetition=10, seed=0)
    res = run(cfg)
    print(f"\nval-ACC {res['acc_val']:.4f}; outputs: {outdir}/result_*.txt")


if __name__ == "__main__":
    main()
