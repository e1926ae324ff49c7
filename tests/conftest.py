import os
import sys

import pytest
import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def pytest_configure(config):
    config.addinivalue_line("markers", "gpu: requires an MI355X (or any ROCm GPU)")


def pytest_collection_modifyitems(config, items):
    if torch.cuda.is_available():
        return
    skip = pytest.mark.skip(reason="no GPU in this environment")
    for item in items:
        if "gpu" in item.keywords:
            item.add_marker(skip)


@pytest.fixture
def tiny_cfg():
    from ai_rtc_agent_amd.config import EngineConfig

    return EngineConfig(
        model_family="tiny",
        width=64,
        height=64,
        t_index_list=[18, 26, 35, 45],
        device="cpu",
        acceleration="eager",
        use_hip_graph=False,
        use_lcm_lora=False,
    )
