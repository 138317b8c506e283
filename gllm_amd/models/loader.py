"""Model registry + checkpoint loading.

Parity: reference model_loader.py (arch→class registry :501-535, lazy
safetensors shard reads :60-108, dummy load format, HF config).
"""

import glob
import json
import os
import time
from typing import Iterable, Tuple

import torch

from gllm_amd.logger import logger


def get_arch_registry():
    from gllm_amd.models.llama_family import (LlamaForCausalLM,
                                              Qwen2ForCausalLM,
                                              Qwen3ForCausalLM)
    from gllm_amd.models.moe_family import (MixtralForCausalLM,
                                            Qwen2MoeForCausalLM,
                                            Qwen3MoeForCausalLM)
    return {
        "LlamaForCausalLM": LlamaForCausalLM,
        "Qwen2ForCausalLM": Qwen2ForCausalLM,
        "Qwen3ForCausalLM": Qwen3ForCausalLM,
        "MixtralForCausalLM": MixtralForCausalLM,
        "Qwen2MoeForCausalLM": Qwen2MoeForCausalLM,
        "Qwen3MoeForCausalLM": Qwen3MoeForCausalLM,
    }


def load_hf_config(model_path: str):
    cfg_file = os.path.join(model_path, "config.json")
    if os.path.exists(cfg_file):
        try:
            from transformers import AutoConfig
            return AutoConfig.from_pretrained(model_path,
                                              trust_remote_code=True)
        except Exception:
            import types
            with open(cfg_file) as f:
                d = json.load(f)
            return types.SimpleNamespace(**d)
    raise FileNotFoundError(f"no config.json under {model_path}")


def iterate_safetensors(model_path: str
                        ) -> Iterable[Tuple[str, torch.Tensor]]:
    """Lazily yield (name, tensor) from every *.safetensors shard."""
    from safetensors import safe_open
    shards = sorted(glob.glob(os.path.join(model_path, "*.safetensors")))
    if not shards:
        # .bin fallback
        bins = sorted(glob.glob(os.path.join(model_path, "*.bin")))
        if not bins:
            raise FileNotFoundError(
                f"no safetensors/bin weights under {model_path}")
        for b in bins:
            sd = torch.load(b, map_location="cpu", weights_only=True)
            yield from sd.items()
        return
    for shard in shards:
        with safe_open(shard, framework="pt", device="cpu") as f:
            for name in f.keys():
                yield name, f.get_tensor(name)


def dummy_init(model: torch.nn.Module, seed: int = 0) -> None:
    """Random-init all params (reference --load-format dummy). Seeded so
    every TP rank holding a replicated param gets identical values, and
    numerically tame so bf16 forward passes stay finite."""
    gen = torch.Generator()
    for name, p in sorted(model.named_parameters()):
        gen.manual_seed(seed ^ (hash(name) & 0x7FFFFFFF))
        with torch.no_grad():
            if p.dim() >= 2:
                t = torch.empty(p.shape, dtype=torch.float32)
                t.normal_(0.0, 0.02, generator=gen)
                p.data.copy_(t.to(p.dtype))
            elif "norm" in name or "weight" in name and p.dim() == 1 and \
                    "bias" not in name:
                p.data.fill_(1.0)
            else:
                p.data.zero_()


def create_model(cfg, engine_config, device: str = "cpu") -> torch.nn.Module:
    archs = getattr(cfg, "architectures", None) or []
    registry = get_arch_registry()
    cls = None
    for a in archs:
        if a in registry:
            cls = registry[a]
            break
    if cls is None:
        raise ValueError(f"unsupported architectures {archs}; "
                         f"known: {sorted(registry)}")
    with torch.device(device):
        model = cls(cfg, engine_config)
    return model.eval()


def load_model(engine_config, device: str = "cpu"):
    t0 = time.time()
    cfg = load_hf_config(engine_config.model)
    model = create_model(cfg, engine_config, device)
    if engine_config.load_format == "dummy":
        dummy_init(model, engine_config.seed)
    else:
        model.load_weights(iterate_safetensors(engine_config.model))
    logger.info("model loaded in %.1fs (%s, %d local layers)",
                time.time() - t0, type(model).__name__,
                model.num_local_layers)
    return model, cfg
