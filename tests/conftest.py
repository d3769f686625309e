import os
import sys

import pytest

sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))


def pytest_configure(config):
    config.addinivalue_line(
        "markers", "gpu: test requires a real MI355X GPU (run via gpurun)"
    )


@pytest.fixture
def mi355x_system():
    from simumax_amd import SystemConfig, get_simu_system_config

    return SystemConfig.init_from_config_file(get_simu_system_config("mi355x"))


@pytest.fixture
def llama3_8b():
    from simumax_amd import ModelConfig, get_simu_model_config

    return ModelConfig.init_from_config_file(get_simu_model_config("llama3-8b"))
