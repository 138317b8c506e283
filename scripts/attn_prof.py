"""Prefill/decode attention kernel timing."""
import os, sys, time
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from gllm_amd import ops

def main():
    B, Hq, Hkv, D, ps = 8, 40, 8, 128, 16
    S = 1024
    ctx = [S] * B
    n_pages = B * (S // ps) + 1
    torch.manual_seed(0)
    k_cache = torch.randn(n_pages, ps, Hkv, D, dtype=torch.bfloat16, device="cuda")
    v_cache = torch.randn(n_pages, ps, Hkv, D, dtype=torch.bfloat16, device="cuda")
    bt = torch.arange(1, n_pages, dtype=torch.int32, device="cuda").reshape(B, S // ps)
    T = B * S
    q = torch.randn(T, Hq, D, dtype=torch.bfloat16, device="cuda")
    seq_lens = torch.tensor(ctx, dtype=torch.int32, device="cuda")
    qsl = torch.arange(0, T + 1, S, dtype=torch.int32, device="cuda")
    sc = D ** -0.5
    for _ in range(3):
        ops.paged_attention(q, k_cache, v_cache, bt, seq_lens, qsl, sc, max_query_len=S)
    torch.cuda.synchronize()
    t0 = time.time(); it = 20
    for _ in range(it):
        ops.paged_attention(q, k_cache, v_cache, bt, seq_lens, qsl, sc, max_query_len=S)
    torch.cuda.synchronize()
    dt = (time.time() - t0) / it
    flops = 4.0 * Hq * D * B * (S * S / 2)
    print(f"prefill attn B{B}xS{S}: {dt*1e3:.2f} ms  {flops/dt/1e12:.1f} TF/s")

if __name__ == "__main__":
    main()


def decode_bench():
    B, Hq, Hkv, D, ps = 256, 40, 8, 128, 16
    S = 1024
    n_pages = B * (S // ps) + 1
    torch.manual_seed(0)
    k_cache = torch.randn(n_pages, ps, Hkv, D, dtype=torch.bfloat16, device="cuda")
    v_cache = torch.randn(n_pages, ps, Hkv, D, dtype=torch.bfloat16, device="cuda")
    bt = torch.arange(1, n_pages, dtype=torch.int32, device="cuda").reshape(B, S // ps)
    q = torch.randn(B, Hq, D, dtype=torch.bfloat16, device="cuda")
    seq_lens = torch.full((B,), S, dtype=torch.int32, device="cuda")
    qsl = torch.arange(B + 1, dtype=torch.int32, device="cuda")
    sc = D ** -0.5
    for _ in range(3):
        ops.paged_attention(q, k_cache, v_cache, bt, seq_lens, qsl, sc, max_query_len=1)
    torch.cuda.synchronize()
    t0 = time.time(); it = 30
    for _ in range(it):
        ops.paged_attention(q, k_cache, v_cache, bt, seq_lens, qsl, sc, max_query_len=1)
    torch.cuda.synchronize()
    dt = (time.time() - t0) / it
    kv_bytes = 2.0 * B * S * Hkv * D * 2
    print(f"decode attn B{B}xS{S}: {dt*1e3:.3f} ms  KV {kv_bytes/dt/1e12:.2f} TB/s")


if os.environ.get("ATTN_DECODE"):
    decode_bench()


def decode_sweep():
    """Decode attention TB/s across B/ctx/splits (GLLM_DECODE_SPLITS)."""
    import itertools
    Hq, Hkv, D, ps = 40, 8, 128, 16
    for B, S in itertools.product([32, 64, 256], [1024, 4096]):
        n_pages = B * (S // ps) + 1
        torch.manual_seed(0)
        k_cache = torch.randn(n_pages, ps, Hkv, D, dtype=torch.bfloat16,
                              device="cuda")
        v_cache = torch.randn(n_pages, ps, Hkv, D, dtype=torch.bfloat16,
                              device="cuda")
        bt = torch.arange(1, n_pages, dtype=torch.int32,
                          device="cuda").reshape(B, S // ps)
        q = torch.randn(B, Hq, D, dtype=torch.bfloat16, device="cuda")
        seq_lens = torch.full((B,), S, dtype=torch.int32, device="cuda")
        qsl = torch.arange(B + 1, dtype=torch.int32, device="cuda")
        sc = D ** -0.5
        for _ in range(5):
            ops.paged_attention(q, k_cache, v_cache, bt, seq_lens, qsl, sc)
        torch.cuda.synchronize()
        t0 = time.time()
        it = 30
        for _ in range(it):
            ops.paged_attention(q, k_cache, v_cache, bt, seq_lens, qsl, sc)
        torch.cuda.synchronize()
        dt = (time.time() - t0) / it
        kv_bytes = 1.0 * B * S * Hkv * D * 2 * 2  # K+V, bf16
        print(f"decode B={B:4d} S={S:5d}: {dt*1e6:8.1f} us  "
              f"KV {kv_bytes/dt/1e12:5.2f} TB/s")
if __name__ == "__main__" and os.environ.get("DECODE_SWEEP"): decode_sweep()
