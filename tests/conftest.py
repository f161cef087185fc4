import os
import socket

import pytest
import torch


def pytest_configure(config):
    config.addinivalue_line(
        "markers", "gpu: tests that require a ROCm GPU (run with -m gpu on an MI355X box)"
    )


def pytest_collection_modifyitems(config, items):
    if torch.cuda.is_available():
        return
    skip_gpu = pytest.mark.skip(reason="no GPU available")
    for item in items:
        if "gpu" in item.keywords:
            item.add_marker(skip_gpu)


def free_port():
    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    port = s.getsockname()[1]
    s.close()
    return port


@pytest.fixture
def tiny_llama_config():
    from relora_amd.models.config import LlamaConfig

    return LlamaConfig(
        vocab_size=256,
        hidden_size=64,
        intermediate_size=176,
        num_hidden_layers=2,
        num_attention_heads=4,
        max_position_embeddings=128,
        rms_norm_eps=1e-6,
    )


@pytest.fixture
def tiny_pythia_config():
    from relora_amd.models.config import GPTNeoXConfig

    return GPTNeoXConfig(
        vocab_size=256,
        hidden_size=64,
        intermediate_size=256,
        num_hidden_layers=2,
        num_attention_heads=4,
        max_position_embeddings=128,
        rotary_pct=0.25,
        use_parallel_residual=True,
    )


@pytest.fixture(autouse=True)
def _no_tqdm_env(monkeypatch):
    monkeypatch.setenv("RELORA_AMD_NO_TQDM", "1")
