import os
import sys

import pytest

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def pytest_configure(config):
    config.addinivalue_line(
        "markers", "gpu: tests that require an MI355X (run with -m gpu)")


def pytest_collection_modifyitems(config, items):
    import torch
    if torch.cuda.is_available():
        return
    skip_gpu = pytest.mark.skip(reason="no GPU available")
    for item in items:
        if "gpu" in item.keywords:
            item.add_marker(skip_gpu)


@pytest.fixture(scope="session")
def toy_corpus(tmp_path_factory):
    """Deterministic toy corpus + dictionary in a session tmp dir."""
    from nats_amd.data.synthetic import make_toy_corpus
    d = tmp_path_factory.mktemp("toy_data")
    make_toy_corpus(str(d))
    return str(d)


@pytest.fixture()
def tiny_options():
    from nats_amd.models.distraction import default_options
    return default_options(dim_word=12, dim=16, dim_att=8, n_words=64,
                           maxlen=50, batch_size=4, valid_batch_size=4)
