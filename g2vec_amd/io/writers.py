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


def write_vectors(result_name: str, mat: np.ndarray, genes: Sequence[str],
                  engine: str = "auto") -> str:
    """engine: 'numpy' | 'pandas' | 'auto' (pandas above 5M values — its C
    csv writer is ~10x faster on the 1M-gene x 512 = 3.7 GB output)."""
    out = result_name + "_vectors.txt"
    h = mat.shape[1]
    if engine == "auto":
        engine = "pandas" if mat.size > 5_000_000 else "numpy"
    if engine == "pandas":
        import pandas as pd
        df = pd.DataFrame(mat, index=np.asarray(genes, dtype=str),
                          columns=[f"V{i}" for i in range(h)])
        df.index.name = "GeneSymbol"
        df.to_csv(out, sep="\t", float_format="%.6f", lineterminator="\n")
        return out
    header = "GeneSymbol" + "".join("\tV%d" % i for i in range(h)) + "\n"
    # vectorized "%.6f" formatting; byte format identical to the
    # reference's \t%.6f (G2Vec.py:214)
    cells = np.char.mod("%.6f", mat.astype(np.float64))
    rows = np.char.add(np.asarray(genes, dtype=str),
                       ["\t" + "\t".join(r) for r in cells])
    with open(out, "w") as f:
        f.write(header)
        f.write("\n".join(rows))
        f.write("\n")
    return out
