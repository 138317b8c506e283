"""Multimodal serving path: native image processor + OpenAI image_url
content end to end through the serving workers (CPU)."""

import base64
import io
import json

import pytest
import torch


def test_smart_resize_and_processor():
    import numpy as np
    from PIL import Image
    from gllm_amd.multimodal.processor import ImageProcessor, smart_resize
    h, w = smart_resize(30, 45, factor=28)
    assert h % 28 == 0 and w % 28 == 0
    proc = ImageProcessor(patch_size=14, temporal_patch_size=2,
                          spatial_merge_size=2)
    img = Image.fromarray(np.random.RandomState(0).randint(
        0, 255, (30, 45, 3), dtype=np.uint8))
    patches, grid = proc(img)
    t, gh, gw = grid
    assert t == 1 and patches.shape == (gh * gw, 3 * 2 * 14 * 14)
    assert proc.num_tokens(grid) == gh * gw // 4


def test_expand_image_tokens():
    from gllm_amd.multimodal.processor import expand_image_tokens
    out = expand_image_tokens([1, 9, 2, 9, 3], sentinel_id=9,
                              counts=[2, 3], pad_id=7)
    assert out == [1, 7, 7, 2, 7, 7, 7, 3]
    with pytest.raises(AssertionError):
        expand_image_tokens([1, 9], 9, [1, 1], 7)


VL_CFG = {
    "architectures": ["Qwen2VLForConditionalGeneration"],
    "model_type": "qwen2_vl",
    "hidden_size": 64,
    "intermediate_size": 128,
    "num_hidden_layers": 2,
    "num_attention_heads": 4,
    "num_key_value_heads": 2,
    "vocab_size": 160,
    "image_token_id": 150,
    "max_position_embeddings": 2048,
    "rms_norm_eps": 1e-6,
    "rope_theta": 10000.0,
    "rope_scaling": {"type": "mrope", "mrope_section": [2, 3, 3]},
    "eos_token_id": 1,
    "vision_config": {
        "depth": 2, "embed_dim": 32, "hidden_size": 64, "num_heads": 4,
        "mlp_ratio": 2.0, "patch_size": 14, "temporal_patch_size": 2,
        "in_channels": 3, "spatial_merge_size": 2,
    },
}


def _mk_vl_dir(tmp_path):
    d = tmp_path / "vl"
    d.mkdir()
    with open(d / "config.json", "w") as f:
        json.dump(VL_CFG, f)
    from tokenizers import Tokenizer
    from tokenizers.models import WordLevel
    from tokenizers.pre_tokenizers import WhitespaceSplit
    vocab = {"[UNK]": 0, "</s>": 1}
    for i in range(2, 150):
        vocab[f"w{i}"] = i
    vocab["<|image_pad|>"] = 150
    vocab["<|vision_start|>"] = 151
    vocab["<|vision_end|>"] = 152
    tok = Tokenizer(WordLevel(vocab, unk_token="[UNK]"))
    tok.pre_tokenizer = WhitespaceSplit()
    tok.save(str(d / "tokenizer.json"))
    with open(d / "tokenizer_config.json", "w") as f:
        json.dump({
            "tokenizer_class": "PreTrainedTokenizerFast",
            "eos_token": "</s>",
            "unk_token": "[UNK]",
            "model_max_length": 2048,
            "chat_template": (
                "{% for m in messages %}{{ m.content }} {% endfor %}"),
        }, f)
    return str(d)


def _img_b64(seed, size=(28, 28)):
    import numpy as np
    from PIL import Image
    arr = np.random.RandomState(seed).randint(0, 255,
                                              (size[0], size[1], 3),
                                              dtype=np.uint8)
    buf = io.BytesIO()
    Image.fromarray(arr).save(buf, format="PNG")
    return "data:image/png;base64," + \
        base64.b64encode(buf.getvalue()).decode()


@pytest.mark.timeout(300)
def test_mm_chat_completion_roundtrip(tmp_path):
    from fastapi.testclient import TestClient
    import gllm_amd.entrypoints.api_server as srv
    from gllm_amd.config import EngineConfig
    from gllm_amd.engine.server_engine import AsyncLLMEngine

    d = _mk_vl_dir(tmp_path)
    cfg = EngineConfig(model=d, load_format="dummy", device="cpu",
                       dtype="float32", page_size=4, maxp=64,
                       master_port=29695, enable_prefix_caching=False)
    srv.engine = AsyncLLMEngine(cfg)
    srv.served_model = "vl-tiny"
    # the WordLevel test tokenizer is whitespace-split: pad the image
    # marker strings so they tokenize as single words
    srv.engine.vision_wrap = ("<|vision_start|> ", " <|vision_end|>")
    srv.engine.start()
    try:
        app = srv.build_app()
        client = TestClient(app)

        def ask(seed):
            r = client.post("/v1/chat/completions", json={
                "messages": [{"role": "user", "content": [
                    {"type": "text", "text": "w5 w6 "},
                    {"type": "image_url",
                     "image_url": {"url": _img_b64(seed)}},
                    {"type": "text", "text": " w7"},
                ]}],
                "max_tokens": 6, "temperature": 0.0, "ignore_eos": True})
            assert r.status_code == 200, r.text
            body = r.json()
            assert body["usage"]["completion_tokens"] == 6
            return body["choices"][0]["message"]["content"]

        a1 = ask(0)
        a2 = ask(0)
        assert a1 == a2, "greedy mm serving must be deterministic"
        b = ask(99)  # different image content
        assert b != a1, "image pixels must influence the completion"

        # text-only still works on the same engine
        r = client.post("/v1/chat/completions", json={
            "messages": [{"role": "user", "content": "w9 w10"}],
            "max_tokens": 3, "temperature": 0.0, "ignore_eos": True})
        assert r.status_code == 200, r.text
    finally:
        srv.engine.stop()


def test_vit_output_lru_cache(tmp_path):
    """Repeated image content skips the tower via the content-hash LRU
    (reference MultiModalEmbeddingCache)."""
    import json as _json
    import torch
    from gllm_amd import multimodal
    from gllm_amd.multimodal import prepare as P
    from gllm_amd.config import EngineConfig
    from gllm_amd.engine.llm import LLM
    from gllm_amd.sequence import SamplingParams

    d = tmp_path / "vlc"
    d.mkdir()
    cfg_json = {**VL_CFG}
    with open(d / "config.json", "w") as f:
        _json.dump(cfg_json, f)
    cfg = EngineConfig(model=str(d), load_format="dummy", device="cpu",
                       dtype="float32", page_size=4, maxp=64,
                       enable_prefix_caching=False)
    llm = LLM(config=cfg, num_pages_override=128)
    torch.manual_seed(0)
    px = torch.randn(16, 3 * 2 * 14 * 14)
    mm = {"pixel_values": px, "grids": [(1, 4, 4)]}
    toks = [2, 3] + [150] * 4 + [4]
    sp = [SamplingParams(temperature=0.0, max_tokens=3, ignore_eos=True)]
    before = P.emb_cache_hits
    o1 = llm.generate([toks], sp, mm_inputs=[mm])[0].token_ids
    o2 = llm.generate([toks], sp, mm_inputs=[mm])[0].token_ids
    assert P.emb_cache_hits > before
    assert o1 == o2
