import os
import sys

import pytest

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def pytest_configure(config):
    config.addinivalue_line(
        "markers", "gpu: test requires an MI355X GPU (run via gpurun)")


def pytest_collection_modifyitems(config, items):
    import torch
    if torch.cuda.is_available():
        return
    skip = pytest.mark.skip(reason="no GPU in this container")
    for item in items:
        if "gpu" in item.keywords:
            item.add_marker(skip)


@pytest.fixture(scope="session")
def lenet_pb(tmp_path_factory):
    from tensorflow_web_deploy_amd.models.lenet import save_lenet
    p = tmp_path_factory.mktemp("models") / "lenet_frozen.pb"
    save_lenet(str(p), seed=7)
    return str(p)
