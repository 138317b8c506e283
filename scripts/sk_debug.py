import os, sys
sys.path.insert(0, "/root/repo")
import torch
from gllm_amd import ops
torch.manual_seed(0)
for (M,N,K) in [(1,7168,5120),(17,5120,5120),(64,64,128),(64,64,64),(16,64,192),(64,5120,27648),(256,152064,5120)]:
    x = torch.randn(M, K, dtype=torch.bfloat16, device="cuda") * 0.1
    w = torch.randn(N, K, dtype=torch.bfloat16, device="cuda") * 0.1
    out = ops.skinny_gemm(x, w).float()
    ref = x.float() @ w.float().T
    diff = (out - ref).abs()
    rel = (diff / ref.abs().clamp_min(1.0))
    bad = rel > 3e-2
    print(f"M={M} N={N} K={K}: maxrel={float(rel.max()):.4f} frac_bad={float(bad.float().mean()):.4f}")
    if bad.any():
        bi = bad.nonzero()[:8]
        print("  sample bad idx:", bi.tolist())
        r, c = bi[0]
        print(f"  out={out[r,c]:.4f} ref={ref[r,c]:.4f}")
        # column pattern of badness
        colbad = bad.any(0).nonzero().flatten()
        rowbad = bad.any(1).nonzero().flatten()
        print(f"  bad cols: {colbad[:16].tolist()}{'...' if colbad.numel()>16 else ''} ({colbad.numel()} total)")
        print(f"  bad rows: {rowbad[:16].tolist()} ({rowbad.numel()} total)")

# MB=2 / MB=4 engine-like shapes + graph capture repro
def graph_repro():
    M, N, K = 256, 5120, 27648
    x = torch.randn(M, K, dtype=torch.bfloat16, device="cuda") * 0.1
    w = torch.randn(N, K, dtype=torch.bfloat16, device="cuda") * 0.1
    ops.skinny_gemm(x, w)  # warm ws
    torch.cuda.synchronize()
    g = torch.cuda.CUDAGraph()
    with torch.cuda.graph(g):
        out = ops.skinny_gemm(x, w)
    for _ in range(3):
        g.replay()
    torch.cuda.synchronize()
    ref = x.float() @ w.float().T
    rel = ((out.float() - ref).abs() / ref.abs().clamp_min(1.0)).max()
    print(f"graph repro M={M}: maxrel={float(rel):.4f}")

if os.environ.get("SK_EXTRA"):
    for (M,N,K) in [(100,5120,5120),(128,7168,5120),(200,5120,27648),
                    (256,7168,5120),(256,5120,5120),(65,5120,27648)]:
        x = torch.randn(M, K, dtype=torch.bfloat16, device="cuda") * 0.1
        w = torch.randn(N, K, dtype=torch.bfloat16, device="cuda") * 0.1
        out = ops.skinny_gemm(x, w).float()
        ref = x.float() @ w.float().T
        rel = ((out - ref).abs() / ref.abs().clamp_min(1.0)).max()
        print(f"extra M={M} N={N} K={K}: maxrel={float(rel):.4f}")
    graph_repro()

def graph_repro_multi():
    """Mimic GraphRunner: shared pool, side stream, multiple buckets,
    qkv(+bias)/o/down skinny + gate_up library per forward."""
    import torch.nn.functional as F
    K = 5120
    w_qkv = torch.randn(7168, K, dtype=torch.bfloat16, device="cuda") * 0.05
    b_qkv = torch.randn(7168, dtype=torch.bfloat16, device="cuda") * 0.05
    w_o = torch.randn(K, K, dtype=torch.bfloat16, device="cuda") * 0.05
    w_gu = torch.randn(55296, K, dtype=torch.bfloat16, device="cuda") * 0.05
    w_dn = torch.randn(K, 27648, dtype=torch.bfloat16, device="cuda") * 0.05

    def fwd(x):
        q = ops.linear(x, w_qkv, b_qkv)
        o = ops.linear(q[:, :K].contiguous(), w_o)
        g = ops.linear(o, w_gu)
        h = ops.silu_and_mul(g)
        y = ops.linear(h.contiguous(), w_dn)
        return y

    pool = torch.cuda.graphs.graph_pool_handle()
    stream = torch.cuda.Stream()
    graphs, outs, ins = {}, {}, {}
    for bs in [512, 384, 256, 192, 128, 96, 64, 32, 16, 8, 4, 2, 1]:
        x = torch.randn(bs, K, dtype=torch.bfloat16, device="cuda") * 0.05
        ins[bs] = x
        with torch.cuda.stream(stream):
            for _ in range(2):
                fwd(x)
        torch.cuda.current_stream().wait_stream(stream)
        torch.cuda.synchronize()
        g = torch.cuda.CUDAGraph()
        with torch.cuda.graph(g, pool=pool, stream=stream):
            outs[bs] = fwd(x)
        graphs[bs] = g
    torch.cuda.synchronize()
    for rep in range(5):
        for bs in [256, 64, 128, 1, 512]:
            graphs[bs].replay()
    torch.cuda.synchronize()
    for bs in [256, 64]:
        ref = fwd(ins[bs])
        graphs[bs].replay()
        torch.cuda.synchronize()
        d = (outs[bs].float() - ref.float()).abs().max()
        print(f"multi-graph bs={bs}: maxdiff={float(d):.4f}")
    print("graph_repro_multi OK")

if os.environ.get("SK_GRAPH2"):
    graph_repro_multi()
