"""Driver-contract checks for bench.py (CPU, gloo)."""

import json
import os
import subprocess
import sys

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def _last_json(out: str) -> dict:
    for line in reversed(out.strip().splitlines()):
        line = line.strip()
        if line.startswith("{"):
            return json.loads(line)
    raise AssertionError(f"no JSON line in output:\n{out[-2000:]}")


def test_bench_single_rank_contract():
    out = subprocess.run(
        [sys.executable, "bench.py", "--model", "debug", "--steps", "2",
         "--warmup", "1", "--batch", "4", "--prompt-len", "32"],
        cwd=REPO, capture_output=True, text=True, timeout=600)
    assert out.returncode == 0, out.stderr[-2000:]
    j = _last_json(out.stdout)
    assert j["n_gpus"] == 1 and j["steps"] == 2
    assert j["ttft_qps"] == 32.0 and j["ttft_p50_ms"] is not None
    assert j["higher_is_better"] is True


def test_bench_two_ranks_pp_gloo():
    """The driver's torchrun invocation shape: 2 ranks, PP=2 over gloo
    with the paced TTFT ramp broadcasting release counts."""
    out = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
         "--master-port", "29575", "bench.py", "--gpus", "2",
         "--model", "debug", "--steps", "2", "--warmup", "1",
         "--batch", "4", "--prompt-len", "32"],
        cwd=REPO, capture_output=True, text=True, timeout=900)
    assert out.returncode == 0, out.stderr[-2000:]
    j = _last_json(out.stdout)
    assert j["n_gpus"] == 2
    assert j["config"]["parallelism"] == "pp2"
