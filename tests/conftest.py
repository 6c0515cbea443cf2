import os
import sys

import pytest

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def pytest_configure(config):
    config.addinivalue_line(
        "markers", "gpu: test requires an MI355X GPU (run with -m gpu)"
    )


def pytest_collection_modifyitems(config, items):
    """Skip gpu tests automatically when no GPU is present."""
    if config.getoption("-m", default="") == "gpu":
        return
    import torch

    if torch.cuda.is_available():
        return
    skip_gpu = pytest.mark.skip(reason="no GPU available")
    for item in items:
        if "gpu" in item.keywords:
            item.add_marker(skip_gpu)


@pytest.fixture
def tiny_config():
    from vllm_amd.config import (
        CacheConfig,
        DeviceConfig,
        EngineConfig,
        ModelConfig,
        ParallelConfig,
        SchedulerConfig,
    )

    return EngineConfig(
        model_config=ModelConfig(model="tiny-llama", dtype="fp32",
                                 max_model_len=2048),
        cache_config=CacheConfig(block_size=16, num_gpu_blocks=128),
        scheduler_config=SchedulerConfig(
            max_num_batched_tokens=256, max_num_seqs=8
        ),
        parallel_config=ParallelConfig(),
        device_config=DeviceConfig(device="cpu"),
    )
