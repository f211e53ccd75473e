"""Output writers — byte-format-compatible with the reference.

  *_biomarkers.txt  G2Vec.py:127-131
  *_lgroups.txt     G2Vec.py:159-165
  *_vectors.txt     G2Vec.py:203-215 (values printed \t%.6f)
"""
from __future__ import annotations

from typing import Iterable, Sequence

import numpy as np


def write_biomarkers(result_name: str, biomarkers: Iterable[str]) -> str:
    out = result_name + "_biomarkers.txt"
    with open(out, "w") as f:
        f.write("GeneSymbol\n")
        for g in biomarkers:
            f.write(f"{g}\n")
    return out


def write_lgroups(result_name: str, lgroup_idx: Sequence[int], genes: Sequence[str]) -> str:
    out = result_name + "_lgroups.txt"
    with open(out, "w") as f:
        f.write("GeneSymbol\tLgroup(0:good,1:poor,2:other)\n")
        for gene, grp in zip(genes, lgroup_idx):
            f.write("%s\t%d\n" % (gene, int(grp)))
    return out


def write_vectors(result_name: str, mat: np.ndarray, genes: Sequence[str]) -> str:
    out = result_name + "_vectors.txt"
    h = mat.shape[1]
    header = "GeneSymbol" + "".join("\tV%d" % i for i in range(h)) + "\n"
    # vectorized "%.6f" formatting (a pure-Python loop takes seconds at
    # G x h = 1M values); byte format identical to the reference's \t%.6f
    cells = np.char.mod("%.6f", mat.astype(np.float64))
    rows = np.char.add(np.asarray(genes, dtype=str),
                       ["\t" + "\t".join(r) for r in cells])
    with open(out, "w") as f:
        f.write(header)
        f.write("\n".join(rows))
        f.write("\n")
    return out
