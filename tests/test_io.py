"""Parser/writer round-trips against the documented formats."""
import numpy as np

from g2vec_amd.io import (load_clinical, load_expression, load_network,
                          write_biomarkers, write_lgroups, write_vectors)


def _write(p, text):
    p.write_text(text)
    return str(p)


def test_expression_roundtrip(tmp_path):
    path = _write(tmp_path / "e.tsv",
                  "PATIENT\tS1\tS2\nA1BG\t1.5\t-2.25\nB2M\t0.0\t4.5\n")
    d = load_expression(path)
    assert list(d["sample"]) == ["S1", "S2"]
    assert list(d["gene"]) == ["A1BG", "B2M"]
    # transposed to samples x genes (G2Vec.py:498)
    assert d["expr"].shape == (2, 2)
    assert d["expr"][0, 0] == 1.5 and d["expr"][1, 1] == 4.5


def test_expression_python_fallback_matches_native(tmp_path):
    path = _write(tmp_path / "e.tsv",
                  "PATIENT\tS1\tS2\tS3\nG1\t0.25\t-1\t3\nG2\t2\t0\t-0.5\n")
    d_native = load_expression(path, use_native=True)
    d_py = load_expression(path, use_native=False)
    assert np.allclose(d_native["expr"], d_py["expr"])
    assert list(d_native["gene"]) == list(d_py["gene"])
    assert list(d_native["sample"]) == list(d_py["sample"])


def test_clinical(tmp_path):
    path = _write(tmp_path / "c.tsv",
                  "PATIENT_BARCODE\tLABEL\nS1\t0\nS2\t1\nS3\t0\n")
    c = load_clinical(path)
    assert c == {"S1": 0, "S2": 1, "S3": 0}


def test_network_directed(tmp_path):
    path = _write(tmp_path / "n.tsv", "src\tdest\nA\tB\nB\tA\nA\tC\n")
    n = load_network(path)
    assert n["edge"] == [("A", "B"), ("B", "A"), ("A", "C")]
    assert n["gene"] == {"A", "B", "C"}


def test_writers_golden(tmp_path):
    base = str(tmp_path / "res")
    write_biomarkers(base, ["ADH1C", "AKAP13"])
    assert (tmp_path / "res_biomarkers.txt").read_text() == \
        "GeneSymbol\nADH1C\nAKAP13\n"

    write_lgroups(base, [2, 0, 1], ["A1CF", "A2M", "AAK1"])
    assert (tmp_path / "res_lgroups.txt").read_text() == (
        "GeneSymbol\tLgroup(0:good,1:poor,2:other)\n"
        "A1CF\t2\nA2M\t0\nAAK1\t1\n")

    mat = np.array([[0.1234567, -1.0]], dtype=np.float32)
    write_vectors(base, mat, ["A1CF"])
    assert (tmp_path / "res_vectors.txt").read_text() == (
        "GeneSymbol\tV0\tV1\nA1CF\t0.123457\t-1.000000\n")


def test_write_vectors_engines_identical(tmp_path):
    rng = np.random.default_rng(4)
    mat = rng.normal(size=(7, 5)).astype(np.float32)
    genes = [f"G{i}" for i in range(7)]
    from g2vec_amd.io.writers import write_vectors
    a = write_vectors(str(tmp_path / "a"), mat, genes, engine="numpy")
    b = write_vectors(str(tmp_path / "b"), mat, genes, engine="pandas")
    assert open(a).read() == open(b).read()


def test_malformed_inputs_raise_with_context(tmp_path):
    """Parser errors name the file, line, and offending value (the
    reference crashes with bare ValueErrors/IndexErrors)."""
    import pytest

    from g2vec_amd.io import readers
    e1 = tmp_path / "ragged.tsv"
    e1.write_text("PATIENT\tS1\tS2\nG1\t1.0\nG2\t1.0\t2.0\n")
    for native in (False, True):
        with pytest.raises((ValueError, RuntimeError),
                           match=r"ragged(\.tsv)?:2.*expected 2"):
            readers.load_expression(str(e1), use_native=native)
    e2 = tmp_path / "nonnum.tsv"
    e2.write_text("PATIENT\tS1\nG1\tabc\n")
    with pytest.raises(ValueError, match=r"nonnum\.tsv:2.*non-numeric"):
        readers.load_expression(str(e2), use_native=False)
    e3 = tmp_path / "empty.tsv"
    e3.write_text("")
    with pytest.raises(ValueError, match="no expression rows"):
        readers.load_expression(str(e3), use_native=False)
    c1 = tmp_path / "badlabel.tsv"
    c1.write_text("H\tL\nS1\tx\n")
    with pytest.raises(ValueError, match=r"badlabel\.tsv:2.*integer"):
        readers.load_clinical(str(c1))
