import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(
    os.path.abspath(__file__))))

# NOTE: no ensure_env_or_reexec() here — re-exec'ing inside pytest loses
# all output to pytest's fd-capture temp files (the child's stdout is the
# inherited capture fd). GPU tests simply tolerate MIOpen immediate mode;
# perf-sensitive entry points (bench.py, samples/*) do re-exec.

import numpy as np
import pytest
import torch


def pytest_configure(config):
    config.addinivalue_line(
        "markers", "gpu: tests that require an MI355X GPU (run via gpurun)")


def pytest_collection_modifyitems(config, items):
    if torch.cuda.is_available():
        return
    skip_gpu = pytest.mark.skip(reason="no GPU in this environment")
    for item in items:
        if "gpu" in item.keywords:
            item.add_marker(skip_gpu)


@pytest.fixture(autouse=True)
def _seed():
    np.random.seed(0)
    torch.manual_seed(0)


@pytest.fixture
def shapes_config():
    from maskrcnn_amd.data.synthetic import ShapesConfig

    class TestConfig(ShapesConfig):
        IMAGES_PER_GPU = 1
        PRE_NMS_LIMIT = 1000
        POST_NMS_ROIS_TRAINING = 200
        POST_NMS_ROIS_INFERENCE = 100
        DETECTION_MIN_CONFIDENCE = 0.0

    return TestConfig()


@pytest.fixture
def shapes_dataset():
    from maskrcnn_amd.data.synthetic import ShapesDataset
    ds = ShapesDataset()
    ds.load_shapes(8, 128, 128, seed=1)
    ds.prepare()
    return ds
