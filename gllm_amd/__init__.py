"""gllm_amd — an MI355X-native distributed LLM serving engine.

A from-scratch implementation of the capabilities of gty111/gLLM
(continuous batching, paged attention, chunked prefill, prefix caching,
token-throttling pipeline scheduling, PP/TP/EP/DP parallelism, OpenAI API)
designed for AMD Instinct MI355X (gfx950, CDNA4): PyTorch-ROCm
orchestration, hand-written HIP kernels on MFMA with LDS-staged tiles,
RCCL collectives over xGMI, and hipGraph-captured decode steps.
"""

__version__ = "0.1.0"

from gllm_amd.config import EngineConfig  # noqa: F401
