import os
import sys

import pytest

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def pytest_configure(config):
    config.addinivalue_line(
        "markers", "gpu: test requires an AMD MI355X GPU (run with -m gpu)"
    )


def pytest_collection_modifyitems(config, items):
    # skip gpu-marked tests automatically when no GPU is present
    try:
        import torch

        has_gpu = torch.cuda.is_available()
    except Exception:
        has_gpu = False
    if has_gpu:
        return
    skip_gpu = pytest.mark.skip(reason="no GPU available")
    for item in items:
        if "gpu" in item.keywords:
            item.add_marker(skip_gpu)


@pytest.fixture()
def store(tmp_path):
    from clearml_serving_amd.store import ServingStore

    return ServingStore(str(tmp_path / "store"))


@pytest.fixture()
def processor(store):
    from clearml_serving_amd.serving.processor import ModelRequestProcessor

    return ModelRequestProcessor(
        store=store, name="test-service", project="tests", force_create=True
    )
