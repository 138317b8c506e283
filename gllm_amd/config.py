"""Engine configuration.

The reference threads ~40 kwargs through LLM.__init__ →  ModelRunner
(/root/reference/gllm/llm_engine.py:34-75); here everything lives in one
dataclass that is cheap to pickle across process spawn and carries both
serving knobs and parallel layout.
"""

import dataclasses
import os
from typing import List, Optional


@dataclasses.dataclass
class EngineConfig:
    # --- model ---
    model: str = ""                      # HF model path (dir with config.json)
    load_format: str = "auto"            # "auto" | "dummy"
    dtype: str = "bfloat16"
    trust_remote_code: bool = True
    model_max_length: Optional[int] = None
    quantization: Optional[str] = None   # None | "fp8"

    # --- parallelism ---
    pp_size: int = 1
    tp_size: int = 1
    dp_size: int = 1
    use_ep: bool = False
    # Explicit per-stage layer split, e.g. "9,9,9,9"; None = even split.
    assigned_layers: Optional[str] = None
    master_addr: str = "127.0.0.1"
    master_port: int = 29500
    # multi-node launch: "normal" (single node) | "master" | "slave"
    # (engine/multinode.py — control plane over TCP relay_port, data
    # plane over the torch.distributed rendezvous at master_port)
    launch_mode: str = "normal"
    nnodes: int = 1
    node_rank: int = 0
    # global ranks hosted by THIS node (master/slave modes); None in
    # normal mode = all ranks
    worker_ranks: Optional[List[int]] = None
    # control-plane TCP port on the master node; None = master_port + 1
    relay_port: Optional[int] = None
    # encoder disaggregation (disagg/): remote vision-encoder address
    # "host:port", or a discovery server to resolve one from
    mm_encoder_addr: Optional[str] = None
    # LM node of an encoder-disaggregated deployment: skip loading the
    # vision tower (reference lm_server.py --skip-visual); embeddings
    # arrive from the remote encoder
    skip_visual: bool = False
    discovery_addr: Optional[str] = None

    # --- MLA (DeepSeek family) ---
    # "absorbed" (default): 576-dim latent MQA cache, W_UK/W_UV folded
    # into q/out (models/deepseek_v2.py) — ~70x less KV than
    # decompressed per-head K/V on real configs.
    # "decompressed": per-head K/V cache (the numerics cross-check).
    mla_mode: str = "absorbed"

    # --- KV cache ---
    page_size: int = 16
    gpu_memory_util: float = 0.9
    enable_prefix_caching: bool = True
    kv_cache_dtype: str = "auto"         # "auto" | "fp8"

    # --- scheduler ---
    schedule_method: str = "token_throttling"  # chunked_prefill|token_throttling|split_pd
    maxp: int = 8192                     # max prefill tokens per tick
    maxd: int = 1024                     # max decode seqs (batch rows) per tick
    minp: int = 512                      # min prefill tokens (WT floor)
    iterp: int = 16                      # WT divisor: budget = wait_tokens//iterp
    init_new_token_ratio: float = 0.7
    min_new_token_ratio: float = 0.1

    # --- runtime ---
    use_graph: bool = True               # hipGraph-captured decode buckets
    enable_overlap: bool = True          # launch-first/collect-later worker loop
    # hipGraph decode buckets now cover the full skinny GEMM range:
    # the r1 MB>=2 replay fault traced to dynamic LDS under capture
    # (kernel uses static LDS since r2) and to the workspace growing
    # mid-capture (warmup at the largest bucket sizes it first).
    max_graph_bs: int = 512
    profile_batch: int = 2048            # tokens used for the peak profile run
    device: str = "cuda"                 # "cuda" (=ROCm HIP) | "cpu"
    seed: int = 0
    enforce_eager: bool = False

    # --- derived / internal ---
    port_offset: int = 0                 # zmq port base offset

    def __post_init__(self):
        if self.enforce_eager:
            self.use_graph = False
        if self.device == "cpu":
            self.use_graph = False
            self.enable_overlap = False

    @property
    def world_size(self) -> int:
        return self.pp_size * self.dp_size * self.tp_size

    @property
    def stage_size(self) -> int:
        return self.dp_size * self.tp_size

    def torch_dtype(self):
        import torch
        return {
            "bfloat16": torch.bfloat16, "bf16": torch.bfloat16,
            "float16": torch.float16, "fp16": torch.float16,
            "float32": torch.float32, "fp32": torch.float32,
        }[self.dtype]

    def pp_layer_range(self, pp_rank: int, num_layers: int):
        """[start, end) layer indices owned by pipeline stage ``pp_rank``.

        Mirrors the reference's get_pp_layers (dist_utils.py:494-528):
        an even split with the remainder going to the last stages, or an
        explicit ``assigned_layers`` override.
        """
        if self.assigned_layers:
            splits = [int(x) for x in self.assigned_layers.split(",")]
            assert len(splits) == self.pp_size and sum(splits) == num_layers, (
                f"assigned_layers {splits} does not cover {num_layers} layers "
                f"across {self.pp_size} stages")
            start = sum(splits[:pp_rank])
            return start, start + splits[pp_rank]
        base = num_layers // self.pp_size
        rem = num_layers % self.pp_size
        # earlier stages get the base count; the last `rem` stages get +1
        counts = [base + (1 if i >= self.pp_size - rem else 0)
                  for i in range(self.pp_size)]
        start = sum(counts[:pp_rank])
        return start, start + counts[pp_rank]


def env_flag(name: str, default: bool = False) -> bool:
    v = os.environ.get(name)
    if v is None:
        return default
    return v not in ("0", "false", "False", "")
