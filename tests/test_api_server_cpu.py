"""HTTP-level API server test: FastAPI TestClient against a real CPU
engine with a tiny WordLevel tokenizer built on the fly."""

import json

import pytest


def _mk_model_dir(tmp_path):
    d = tmp_path / "tinytok"
    d.mkdir()
    cfg = {
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
        "eos_token_id": 1,
    }
    with open(d / "config.json", "w") as f:
        json.dump(cfg, f)
    from tokenizers import Tokenizer
    from tokenizers.models import WordLevel
    from tokenizers.pre_tokenizers import Whitespace
    vocab = {"[UNK]": 0, "</s>": 1}
    for i in range(2, 128):
        vocab[f"w{i}"] = i
    tok = Tokenizer(WordLevel(vocab, unk_token="[UNK]"))
    tok.pre_tokenizer = Whitespace()
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


@pytest.mark.timeout(300)
def test_api_server_routes(tmp_path):
    from fastapi.testclient import TestClient
    import gllm_amd.entrypoints.api_server as srv
    from gllm_amd.config import EngineConfig
    from gllm_amd.engine.server_engine import AsyncLLMEngine

    d = _mk_model_dir(tmp_path)
    cfg = EngineConfig(model=d, load_format="dummy", device="cpu",
                       dtype="float32", page_size=4, maxp=64,
                       master_port=29690,
                       enable_prefix_caching=False)
    srv.engine = AsyncLLMEngine(cfg)
    srv.served_model = "tiny"
    srv.engine.start()
    try:
        app = srv.build_app()
        client = TestClient(app)
        assert client.get("/health").json()["status"] == "ok"
        assert client.get("/v1/models").json()["data"][0]["id"] == "tiny"
        assert "pp_size" in client.get("/server_info").json()

        r = client.post("/v1/chat/completions", json={
            "messages": [{"role": "user", "content": "w5 w6 w7"}],
            "max_tokens": 6, "temperature": 0.0, "ignore_eos": True})
        assert r.status_code == 200, r.text
        body = r.json()
        assert body["choices"][0]["message"]["role"] == "assistant"
        assert body["usage"]["completion_tokens"] == 6

        # streaming
        with client.stream("POST", "/v1/chat/completions", json={
                "messages": [{"role": "user", "content": "w9 w10"}],
                "max_tokens": 4, "temperature": 0.0, "stream": True,
                "ignore_eos": True}) as r2:
            lines = [ln for ln in r2.iter_lines() if ln]
        assert lines[-1] == "data: [DONE]"
        assert len([l for l in lines if l.startswith("data: {")]) >= 4

        # completions
        r3 = client.post("/v1/completions", json={
            "prompt": "w3 w4", "max_tokens": 3, "temperature": 0.0,
            "ignore_eos": True})
        assert r3.status_code == 200, r3.text
        assert r3.json()["usage"]["completion_tokens"] == 3

        # n>1: greedy choices must be identical and indexed 0..n-1
        r4 = client.post("/v1/chat/completions", json={
            "messages": [{"role": "user", "content": "w5 w6"}],
            "max_tokens": 4, "temperature": 0.0, "n": 2,
            "ignore_eos": True})
        assert r4.status_code == 200, r4.text
        ch = r4.json()["choices"]
        assert [c["index"] for c in ch] == [0, 1]
        assert ch[0]["message"]["content"] == ch[1]["message"]["content"]
        assert r4.json()["usage"]["completion_tokens"] == 8

        # prompt_logprobs on completions: prompt_len-1 entries
        r5 = client.post("/v1/completions", json={
            "prompt": "w3 w4 w5 w6", "max_tokens": 2, "temperature": 0.0,
            "prompt_logprobs": 3, "ignore_eos": True})
        assert r5.status_code == 200, r5.text
        plp = r5.json()["choices"][0]["prompt_logprobs"]
        assert plp is not None and len(plp) == 3
        assert all(len(entry) == 2 and len(entry[1]) == 3 for entry in plp)

        # metrics + stats
        m = client.get("/metrics").text
        assert "gllm_requests_total" in m
        assert client.get("/stats").json()["requests_total"] >= 3
    finally:
        srv.engine.stop()


@pytest.mark.timeout(300)
def test_api_token_logprobs(tmp_path):
    """Per-token logprobs served through chat (bool + top_logprobs) and
    completions (int), non-stream and stream."""
    from fastapi.testclient import TestClient
    import gllm_amd.entrypoints.api_server as srv
    from gllm_amd.config import EngineConfig
    from gllm_amd.engine.server_engine import AsyncLLMEngine

    d = _mk_model_dir(tmp_path)
    cfg = EngineConfig(model=d, load_format="dummy", device="cpu",
                       dtype="float32", page_size=4, maxp=64,
                       master_port=29693, enable_prefix_caching=False)
    srv.engine = AsyncLLMEngine(cfg)
    srv.served_model = "tiny"
    srv.engine.start()
    try:
        app = srv.build_app()
        client = TestClient(app)
        r = client.post("/v1/chat/completions", json={
            "messages": [{"role": "user", "content": "w5 w6"}],
            "max_tokens": 4, "temperature": 0.0, "ignore_eos": True,
            "logprobs": True, "top_logprobs": 3})
        assert r.status_code == 200, r.text
        lp = r.json()["choices"][0]["logprobs"]["content"]
        assert len(lp) == 4
        for e in lp:
            assert e["logprob"] <= 0.0 and len(e["top_logprobs"]) == 3

        r2 = client.post("/v1/completions", json={
            "prompt": "w3 w4", "max_tokens": 3, "temperature": 0.0,
            "ignore_eos": True, "logprobs": 2})
        assert r2.status_code == 200, r2.text
        lp2 = r2.json()["choices"][0]["logprobs"]["content"]
        assert len(lp2) == 3 and len(lp2[0]["top_logprobs"]) == 2

        with client.stream("POST", "/v1/chat/completions", json={
                "messages": [{"role": "user", "content": "w9"}],
                "max_tokens": 3, "temperature": 0.0, "stream": True,
                "ignore_eos": True, "logprobs": True}) as r3:
            lines = [ln for ln in r3.iter_lines()
                     if ln.startswith("data: {")]
        bodies = [json.loads(ln[len("data: "):]) for ln in lines]
        with_lp = [b for b in bodies
                   if b["choices"] and b["choices"][0].get("logprobs")]
        assert len(with_lp) == 3
    finally:
        srv.engine.stop()


@pytest.mark.timeout(300)
def test_api_sampling_controls(tmp_path):
    """logit_bias / allowed_token_ids / bad_words / skip_special_tokens /
    truncate_prompt_tokens are SERVED (change the sampled stream), not
    just accepted."""
    from fastapi.testclient import TestClient
    import gllm_amd.entrypoints.api_server as srv
    from gllm_amd.config import EngineConfig
    from gllm_amd.engine.server_engine import AsyncLLMEngine

    d = _mk_model_dir(tmp_path)
    cfg = EngineConfig(model=d, load_format="dummy", device="cpu",
                       dtype="float32", page_size=4, maxp=64,
                       master_port=29692,
                       enable_prefix_caching=False)
    srv.engine = AsyncLLMEngine(cfg)
    srv.served_model = "tiny"
    srv.engine.start()
    try:
        app = srv.build_app()
        client = TestClient(app)

        # allowed_token_ids: greedy decode restricted to one token
        r = client.post("/v1/completions", json={
            "prompt": "w3 w4", "max_tokens": 4, "temperature": 0.0,
            "ignore_eos": True, "allowed_token_ids": [42]})
        assert r.status_code == 200, r.text
        assert r.json()["choices"][0]["text"].split() == ["w42"] * 4

        # logit_bias: +100 forces the token everywhere
        r = client.post("/v1/completions", json={
            "prompt": "w3 w4", "max_tokens": 3, "temperature": 0.0,
            "ignore_eos": True, "logit_bias": {"37": 100.0}})
        assert r.json()["choices"][0]["text"].split() == ["w37"] * 3

        # bad_words: the unconstrained first token must change
        base = client.post("/v1/completions", json={
            "prompt": "w3 w4", "max_tokens": 1, "temperature": 0.0,
            "ignore_eos": True}).json()["choices"][0]["text"].strip()
        banned = client.post("/v1/completions", json={
            "prompt": "w3 w4", "max_tokens": 1, "temperature": 0.0,
            "ignore_eos": True,
            "bad_words": [base]}).json()["choices"][0]["text"].strip()
        assert banned != base

        # skip_special_tokens: force the eos token into the stream
        body = {"prompt": "w3 w4", "max_tokens": 2, "temperature": 0.0,
                "ignore_eos": True, "logit_bias": {"1": 100.0}}
        hide = client.post("/v1/completions", json=body).json()
        assert "</s>" not in hide["choices"][0]["text"]
        show = client.post("/v1/completions",
                           json={**body, "skip_special_tokens": False}
                           ).json()
        assert "</s>" in show["choices"][0]["text"]

        # truncate_prompt_tokens: usage reflects the kept tail
        r = client.post("/v1/completions", json={
            "prompt": "w3 w4 w5 w6 w7", "max_tokens": 1,
            "temperature": 0.0, "ignore_eos": True,
            "truncate_prompt_tokens": 2})
        assert r.json()["usage"]["prompt_tokens"] == 2

        # out-of-range ids are a 400, not an engine crash (advisor r1)
        r = client.post("/v1/completions", json={
            "prompt": "w3", "max_tokens": 1,
            "logit_bias": {"99999": 5.0}})
        assert r.status_code == 400 and "out of range" in r.text
        r = client.post("/v1/completions", json={
            "prompt": "w3", "max_tokens": 1,
            "allowed_token_ids": [-3]})
        assert r.status_code == 400 and "out of range" in r.text
        # non-positive truncate_prompt_tokens is rejected by pydantic
        r = client.post("/v1/completions", json={
            "prompt": "w3", "max_tokens": 1,
            "truncate_prompt_tokens": 0})
        assert r.status_code == 422
    finally:
        srv.engine.stop()


def test_bad_words_sequence_matching():
    """Multi-token bad words ban only the final token and only when the
    preceding context matches (vLLM semantics)."""
    from types import SimpleNamespace
    import torch
    from gllm_amd.layers.sampler import Sampler, build_sampling_metadata
    from gllm_amd.sequence import SamplingParams

    def mk_item(ctx, bad):
        sp = SamplingParams(temperature=0.0, bad_words_token_ids=bad)
        seq = SimpleNamespace(sampling=sp, token_ids=list(ctx),
                              prompt_len=len(ctx), num_output_tokens=0)
        return SimpleNamespace(seq=seq, ends_prompt=True)

    V = 16
    logits = torch.zeros(2, V)
    logits[:, 7] = 5.0          # argmax would be 7 everywhere
    logits[:, 3] = 4.0          # runner-up
    # row 0: context ends with [5, 6] and bad word is [5, 6, 7] -> 7 banned
    # row 1: context does NOT match -> 7 allowed
    items = [mk_item([1, 5, 6], [[5, 6, 7]]),
             mk_item([1, 2, 4], [[5, 6, 7]])]
    meta = build_sampling_metadata(items, "cpu")
    out = Sampler()(logits.clone(), meta)
    assert out.next_tokens.tolist() == [3, 7]


@pytest.mark.timeout(300)
def test_chat_template_knobs(tmp_path):
    """add_generation_prompt / continue_final_message reach the chat
    template (prompt token count shifts by the generation-prompt
    marker)."""
    import os

    from fastapi.testclient import TestClient
    import gllm_amd.entrypoints.api_server as srv
    from gllm_amd.config import EngineConfig
    from gllm_amd.engine.server_engine import AsyncLLMEngine

    d = _mk_model_dir(tmp_path)
    # template that actually honors the knob
    with open(os.path.join(d, "tokenizer_config.json")) as f:
        tc = json.load(f)
    tc["chat_template"] = ("{% for m in messages %}{{ m.content }} "
                           "{% endfor %}"
                           "{% if add_generation_prompt %}w99 {% endif %}")
    with open(os.path.join(d, "tokenizer_config.json"), "w") as f:
        json.dump(tc, f)
    cfg = EngineConfig(model=d, load_format="dummy", device="cpu",
                       dtype="float32", page_size=4, maxp=64,
                       master_port=29693,
                       enable_prefix_caching=False)
    srv.engine = AsyncLLMEngine(cfg)
    srv.served_model = "tiny"
    srv.engine.start()
    try:
        client = TestClient(srv.build_app())
        body = {"messages": [{"role": "user", "content": "w5 w6"}],
                "max_tokens": 1, "temperature": 0.0, "ignore_eos": True}
        with_gp = client.post("/v1/chat/completions", json=body).json()
        without = client.post("/v1/chat/completions", json={
            **body, "add_generation_prompt": False}).json()
        assert with_gp["usage"]["prompt_tokens"] == \
            without["usage"]["prompt_tokens"] + 1
        cont = client.post("/v1/chat/completions", json={
            **body,
            "messages": [{"role": "user", "content": "w5 w6"},
                         {"role": "assistant", "content": "w7 w8"}],
            "continue_final_message": True}).json()
        # continued prompt = both messages, no generation marker
        assert cont["usage"]["prompt_tokens"] == 4
    finally:
        srv.engine.stop()


@pytest.mark.timeout(300)
def test_completions_stream_usage(tmp_path):
    """stream_options.include_usage on /v1/completions emits a final
    usage chunk before [DONE]."""
    from fastapi.testclient import TestClient
    import gllm_amd.entrypoints.api_server as srv
    from gllm_amd.config import EngineConfig
    from gllm_amd.engine.server_engine import AsyncLLMEngine

    d = _mk_model_dir(tmp_path)
    cfg = EngineConfig(model=d, load_format="dummy", device="cpu",
                       dtype="float32", page_size=4, maxp=64,
                       master_port=29698,
                       enable_prefix_caching=False)
    srv.engine = AsyncLLMEngine(cfg)
    srv.served_model = "tiny"
    srv.engine.start()
    try:
        client = TestClient(srv.build_app())
        with client.stream("POST", "/v1/completions", json={
                "prompt": [1, 2, 3], "max_tokens": 4, "temperature": 0.0,
                "stream": True, "ignore_eos": True,
                "stream_options": {"include_usage": True}}) as r:
            lines = [ln for ln in r.iter_lines()
                     if ln.startswith("data: {")]
        last = json.loads(lines[-1][6:])
        assert last["usage"]["completion_tokens"] == 4
        assert last["usage"]["prompt_tokens"] == 3
    finally:
        srv.engine.stop()
