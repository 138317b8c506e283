"""GPU checks for the round-1 torch quant/MoE/sampling paths added late
in r1 (the CPU twins live in test_{int4,fp8,moe,api_server}_cpu.py):
int4/fp8 dequant device-safety, the sort+segment MoE dispatch, and the
sampler's row-adjustment controls, all on cuda:0."""

import pytest
import torch

pytestmark = pytest.mark.gpu


@pytest.mark.timeout(300)
@pytest.mark.parametrize("method", ["gptq", "awq"])
def test_int4_dequant_on_gpu_matches_cpu(method):
    from gllm_amd.layers.quantization.int4 import (dequant_awq,
                                                   dequant_gptq,
                                                   pack_awq, pack_gptq)
    torch.manual_seed(0)
    w = torch.randn(64, 128)
    pack = pack_gptq if method == "gptq" else pack_awq
    deq = dequant_gptq if method == "gptq" else dequant_awq
    qw, qz, s = pack(w, 32)
    cpu = deq(qw, qz, s, 32, torch.float32)
    gpu = deq(qw.cuda(), qz.cuda(), s.cuda(), 32, torch.float32)
    assert torch.equal(cpu, gpu.cpu())


@pytest.mark.timeout(300)
def test_fp8_dequant_on_gpu_matches_cpu():
    from gllm_amd.layers.quantization.fp8 import (block_quant_fp8,
                                                  dequant_block_fp8)
    torch.manual_seed(1)
    w = torch.randn(96, 64)
    q, s = block_quant_fp8(w, block=(16, 16))
    cpu = dequant_block_fp8(q, s, (16, 16), torch.float32)
    gpu = dequant_block_fp8(q.cuda(), s.cuda(), (16, 16), torch.float32)
    assert torch.equal(cpu, gpu.cpu())


@pytest.mark.timeout(300)
def test_moe_sort_dispatch_on_gpu():
    """The sort+segment expert dispatch (single host transfer) on cuda
    must match the dense per-token reference."""
    from gllm_amd.layers.moe.layer import FusedMoE
    torch.manual_seed(2)
    E, K, H, I, T = 8, 2, 64, 128, 33
    layer = FusedMoE(E, K, H, I, renormalize=True,
                     params_dtype=torch.float32).cuda()
    with torch.no_grad():
        layer.w13_weight.normal_(0, 0.1)
        layer.w2_weight.normal_(0, 0.1)
    x = torch.randn(T, H, device="cuda")
    logits = torch.randn(T, E, device="cuda")
    out = layer(x, logits)
    # dense reference on CPU
    probs = torch.softmax(logits.float(), -1)
    weights, ids = torch.topk(probs, K, -1)
    weights = weights / weights.sum(-1, keepdim=True)
    ref = torch.zeros_like(x)
    for t in range(T):
        for k in range(K):
            e = int(ids[t, k])
            h = torch.nn.functional.linear(x[t:t + 1],
                                           layer.w13_weight[e])
            d = h.shape[-1] // 2
            act = torch.nn.functional.silu(h[:, :d]) * h[:, d:]
            y = torch.nn.functional.linear(act, layer.w2_weight[e])
            ref[t] += weights[t, k] * y[0]
    assert torch.allclose(out, ref, atol=1e-4), (out - ref).abs().max()


@pytest.mark.timeout(300)
def test_sampler_row_controls_on_gpu():
    """logit_bias / allowed_token_ids / bad_words adjustments on cuda."""
    from types import SimpleNamespace

    from gllm_amd.layers.sampler import Sampler, build_sampling_metadata
    from gllm_amd.sequence import SamplingParams

    def mk(sp_kwargs, ctx):
        sp = SamplingParams(temperature=0.0, **sp_kwargs)
        seq = SimpleNamespace(sampling=sp, token_ids=list(ctx),
                              prompt_len=len(ctx), num_output_tokens=0)
        return SimpleNamespace(seq=seq, ends_prompt=True)

    V = 32
    logits = torch.zeros(3, V, device="cuda")
    logits[:, 7] = 5.0
    logits[:, 3] = 4.0
    items = [
        mk({"allowed_token_ids": [11]}, [1, 2]),         # -> 11
        mk({"logit_bias": {5: 100.0}}, [1, 2]),          # -> 5
        mk({"bad_words_token_ids": [[2, 7]]}, [1, 2]),   # ctx matches -> 3
    ]
    meta = build_sampling_metadata(items, "cuda")
    out = Sampler()(logits.clone(), meta)
    assert out.next_tokens.tolist() == [11, 5, 3]
