import numpy as np
import pytest

from g2vec_amd import preprocess as pp


def test_match_labels():
    clin = {"A": 0, "B": 1}
    out = pp.match_labels(clin, np.array(["B", "A"]))
    assert out.tolist() == [1, 0]
    with pytest.raises(pp.SampleMismatchError):
        pp.match_labels(clin, np.array(["A", "MISSING"]))


def test_common_genes_sorted():
    out = pp.find_common_genes({"Z", "B", "A"}, np.array(["B", "Z", "Q"]))
    assert out == ["B", "Z"]


def test_restrict_network_and_data():
    net = {"edge": [("A", "B"), ("A", "Q"), ("C", "A")], "gene": {"A", "B", "C", "Q"}}
    common = ["A", "B", "C"]
    rn = pp.restrict_network(net, common)
    assert rn["edge"] == [("A", "B"), ("C", "A")]

    data = {"sample": np.array(["S1", "S2"]),
            "label": np.array([0, 1]),
            "expr": np.array([[1., 2., 3., 4.], [5., 6., 7., 8.]], dtype=np.float32),
            "gene": np.array(["Q", "A", "C", "B"])}
    rd = pp.restrict_data(data, common)
    assert list(rd["gene"]) == common
    # columns reordered to common gene order
    assert rd["expr"][0].tolist() == [2., 4., 3.]

    eidx = pp.edges_to_indices(rn["edge"], common)
    assert eidx.tolist() == [[0, 1], [2, 0]]


def test_match_labels_rejects_non_binary():
    import numpy as np
    import pytest

    from g2vec_amd.preprocess import match_labels
    clinical = {"A": 0, "B": 1, "C": 2}
    with pytest.raises(ValueError, match="label 2"):
        match_labels(clinical, np.array(["A", "B", "C"]))
