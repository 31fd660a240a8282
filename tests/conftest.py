import os
import sys

import numpy as np
import pytest
import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def pytest_configure(config):
    config.addinivalue_line("markers", "gpu: needs a ROCm GPU (MI355X)")


def pytest_collection_modifyitems(config, items):
    if torch.cuda.is_available():
        return
    skip_gpu = pytest.mark.skip(reason="no GPU in this environment")
    for item in items:
        if "gpu" in item.keywords:
            item.add_marker(skip_gpu)


@pytest.fixture
def rng():
    return np.random.default_rng(1234)


@pytest.fixture
def tiny_corpus(tmp_path):
    """Small synthetic corpus on disk in the reference file format."""
    from code2vec_amd.data.synthetic import SyntheticSpec, write_synthetic_corpus

    spec = SyntheticSpec(n_methods=48, n_terminals=80, n_paths=60,
                         max_contexts=24, seed=7)
    files = write_synthetic_corpus(str(tmp_path / "data"), spec)
    return files
