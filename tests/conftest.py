import pytest
import torch


def pytest_configure(config):
    config.addinivalue_line(
        "markers", "gpu: tests that need an AMD GPU (run via gpurun / driver)")


def pytest_collection_modifyitems(config, items):
    if torch.cuda.is_available():
        return
    skip = pytest.mark.skip(reason="no GPU on this host")
    for item in items:
        if "gpu" in item.keywords:
            item.add_marker(skip)


@pytest.fixture
def tiny_files(tmp_path):
    """Small ex_*-style synthetic file triple."""
    from g2vec_amd.utils.synth import make_ex_style_files
    return make_ex_style_files(str(tmp_path), n_genes=300, n_extra=40,
                               n_edges=6000, n_samples=80, n_poor=34,
                               n_modules=6, seed=0)
