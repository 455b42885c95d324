import os
import sys

import pytest

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def pytest_configure(config):
    config.addinivalue_line(
        "markers", "gpu: test requires a real MI355X GPU (run via gpurun)")


def pytest_collection_modifyitems(config, items):
    import torch

    if torch.cuda.is_available():
        return
    skip = pytest.mark.skip(reason="no GPU in this container")
    for item in items:
        if "gpu" in item.keywords:
            item.add_marker(skip)


@pytest.fixture
def session():
    from spark_rapids_amd import Session

    return Session()


@pytest.fixture
def gpu_session():
    from spark_rapids_amd import Session

    s = Session()
    s.set("spark.rapids.sql.enabled", True)
    return s


@pytest.fixture
def cpu_session():
    from spark_rapids_amd import Session

    s = Session()
    s.set("spark.rapids.sql.enabled", False)
    return s
