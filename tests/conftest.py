import os
import sys

import pytest
import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def pytest_configure(config):
    config.addinivalue_line(
        "markers", "gpu: test requires an MI355X GPU (run via gpurun)")


def pytest_collection_modifyitems(config, items):
    if not torch.cuda.is_available():
        skip = pytest.mark.skip(reason="no GPU in this container")
        for item in items:
            if "gpu" in item.keywords:
                item.add_marker(skip)


@pytest.fixture(scope="session")
def toy_corpus(tmp_path_factory):
    """Generated parallel toy corpus (digit-words En->De) for pipeline and
    convergence tests — see tools/make_toy_corpus.py."""
    root = tmp_path_factory.mktemp("corpus")
    sys.path.insert(0, os.path.join(os.path.dirname(os.path.dirname(
        os.path.abspath(__file__))), "tools"))
    from make_toy_corpus import generate
    generate(str(root), n_lines=400, seed=7)
    return str(root)
