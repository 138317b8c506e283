import json
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
    skip = pytest.mark.skip(reason="no GPU in this environment")
    for item in items:
        if "gpu" in item.keywords:
            item.add_marker(skip)


TINY_QWEN2 = {
    "architectures": ["Qwen2ForCausalLM"],
    "model_type": "qwen2",
    "hidden_size": 64,
    "intermediate_size": 128,
    "num_hidden_layers": 2,
    "num_attention_heads": 4,
    "num_key_value_heads": 2,
    "vocab_size": 128,
    "max_position_embeddings": 2048,
    "rms_norm_eps": 1e-6,
    "rope_theta": 10000.0,
    "tie_word_embeddings": False,
    "eos_token_id": 0,
    "torch_dtype": "float32",
}


@pytest.fixture()
def tiny_model_dir(tmp_path):
    d = tmp_path / "tiny_qwen2"
    d.mkdir()
    with open(d / "config.json", "w") as f:
        json.dump(TINY_QWEN2, f)
    return str(d)


@pytest.fixture()
def tiny_config(tiny_model_dir):
    from gllm_amd.config import EngineConfig
    return EngineConfig(model=tiny_model_dir, load_format="dummy",
                        device="cpu", dtype="float32", page_size=4,
                        maxp=64, maxd=32, minp=8, iterp=4,
                        enable_prefix_caching=True,
                        schedule_method="chunked_prefill")


@pytest.fixture(autouse=True)
def _reset_parallel_state():
    yield
    from gllm_amd.parallel import state
    state._RANK = 0
    state._WORLD = 1
    state._PP_RANK = state._DP_RANK = state._TP_RANK = 0
    state._PP_SIZE = state._DP_SIZE = state._TP_SIZE = 1
    state._TP_GROUP = state._DP_GROUP = state._EP_GROUP = None
    state._INITIALIZED = False
