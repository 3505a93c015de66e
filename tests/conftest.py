import os
import sys

import pytest

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def pytest_configure(config):
    config.addinivalue_line(
        "markers", "gpu: tests that require an MI355X GPU (run via gpurun)")


def pytest_collection_modifyitems(config, items):
    try:
        import torch
        has_gpu = torch.cuda.is_available()
    except Exception:
        has_gpu = False
    if has_gpu:
        return
    skip = pytest.mark.skip(reason="no GPU in this environment")
    for item in items:
        if "gpu" in item.keywords:
            item.add_marker(skip)


@pytest.fixture
def endpoints():
    """A standard 4-endpoint pool: 3 decode + 1 prefill-decode."""
    from llm_d_inference_scheduler_amd.datalayer.datastore import make_endpoint
    eps = [
        make_endpoint("gpu0", 0, rank=0, role="decode"),
        make_endpoint("gpu1", 1, rank=1, role="decode"),
        make_endpoint("gpu2", 2, rank=2, role="decode"),
        make_endpoint("gpu3", 3, rank=3, role="prefill-decode"),
    ]
    return eps


@pytest.fixture
def request_factory():
    from llm_d_inference_scheduler_amd.scheduling.types import LLMRequest
    counter = [0]

    def make(prompt="hello world " * 20, model="llama-3-8b", **kw):
        counter[0] += 1
        return LLMRequest(request_id=f"req-{counter[0]}", model=model,
                          prompt=prompt, **kw)
    return make
