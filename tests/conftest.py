import os
import sys

import pytest
import torch

REPO_ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
if REPO_ROOT not in sys.path:
    sys.path.insert(0, REPO_ROOT)


def pytest_configure(config):
    config.addinivalue_line("markers", "gpu: requires an MI355X (HIP) device")


def pytest_collection_modifyitems(config, items):
    if torch.cuda.is_available():
        return
    skip_gpu = pytest.mark.skip(reason="no GPU available")
    for item in items:
        if "gpu" in item.keywords:
            item.add_marker(skip_gpu)


@pytest.fixture
def smoke_cfg():
    import types

    from dinov3_amd.configs import setup_config

    args = types.SimpleNamespace(
        config_file=os.path.join(REPO_ROOT, "dinov3_amd/configs/train/vits_smoke.yaml"),
        opts=[],
        output_dir="",
    )
    return setup_config(args, apply_scaling=False)
