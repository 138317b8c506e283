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
    from gllm_amd.models.chatglm import ChatGLMForCausalLM
    from gllm_amd.models.llama_family import MistralForCausalLM
    from gllm_amd.models.deepseek_v2 import (DeepseekV2ForCausalLM,
                                             DeepseekV3ForCausalLM)
    from gllm_amd.models.deepseek_v32 import DeepseekV32ForCausalLM
    from gllm_amd.models.kimi_k25 import KimiK25ForCausalLM
    from gllm_amd.models.hybrid_gdn import Qwen3_5ForCausalLM
    from gllm_amd.models.qwen2_vl import Qwen2VLForCausalLM
    from gllm_amd.models.qwen3_vl import (Qwen3VLForCausalLM,
                                          Qwen3VLMoeForCausalLM,
                                          Qwen3_5VLForCausalLM)
    return {
        "Qwen2VLForConditionalGeneration": Qwen2VLForCausalLM,
        "Qwen2_5_VLForConditionalGeneration": Qwen2VLForCausalLM,
        "Qwen3VLForConditionalGeneration": Qwen3VLForCausalLM,
        "Qwen3VLMoeForConditionalGeneration": Qwen3VLMoeForCausalLM,
        "Qwen3_5ForCausalLM": Qwen3_5ForCausalLM,
        "Qwen3_5ForConditionalGeneration": Qwen3_5VLForCausalLM,
        "Qwen3_5MoeForConditionalGeneration": Qwen3_5VLForCausalLM,
        "Qwen3NextForCausalLM": Qwen3_5ForCausalLM,
        "Qwen3_5MoeForCausalLM": Qwen3_5ForCausalLM,  # MoE via config
        "DeepseekV2ForCausalLM": DeepseekV2ForCausalLM,
        "DeepseekV3ForCausalLM": DeepseekV3ForCausalLM,
        "DeepseekV32ForCausalLM": DeepseekV32ForCausalLM,
        "KimiK25ForConditionalGeneration": KimiK25ForCausalLM,
        "ChatGLMModel": ChatGLMForCausalLM,
        "ChatGLMForConditionalGeneration": ChatGLMForCausalLM,
        "MistralForCausalLM": MistralForCausalLM,
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
            cfg = AutoConfig.from_pretrained(model_path,
                                             trust_remote_code=True)
            return _normalize_rope(_flatten_text_config(cfg))
        except Exception:
            import types
            with open(cfg_file) as f:
                d = json.load(f)
            return types.SimpleNamespace(**d)
    raise FileNotFoundError(f"no config.json under {model_path}")


def _flatten_text_config(cfg):
    """Newer transformers nest LM fields under ``text_config`` for
    multimodal configs (e.g. Qwen2_5_VLConfig). The model classes here
    read flat attributes, so merge text_config up when the top level
    lacks them."""
    txt = getattr(cfg, "text_config", None)
    if txt is None:
        return cfg
    try:
        if getattr(cfg, "num_hidden_layers", None) is not None:
            return cfg
    except Exception:
        pass
    import types
    d = dict(txt.to_dict() if hasattr(txt, "to_dict") else vars(txt))
    try:
        top = cfg.to_dict()
    except Exception:
        top = dict(vars(cfg))
    for k, v in top.items():
        if k != "text_config" and k not in d:
            d[k] = v
    # top-level identity fields win over text_config leftovers
    for k in ("architectures", "model_type"):
        if top.get(k):
            d[k] = top[k]
    return types.SimpleNamespace(**d)


def _normalize_rope(cfg):
    """transformers >= 5 renamed rope_scaling -> rope_parameters (with
    rope_theta folded in); the model classes read the classic names."""
    if getattr(cfg, "rope_scaling", None) is None:
        rp = getattr(cfg, "rope_parameters", None)
        if isinstance(rp, dict) and rp:
            try:
                cfg.rope_scaling = dict(rp)
                if rp.get("rope_theta") is not None and \
                        getattr(cfg, "rope_theta", None) is None:
                    cfg.rope_theta = rp["rope_theta"]
            except Exception:  # frozen config object
                pass
    return cfg


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
    for i, shard in enumerate(shards):
        # load-progress reporting (reference llm_engine.py:324-347 uses
        # shared mp arrays + a bar; per-shard logs serve the same
        # purpose across worker processes)
        logger.info("loading weights: shard %d/%d (%s)", i + 1,
                    len(shards), os.path.basename(shard))
        with safe_open(shard, framework="pt", device="cpu") as f:
            for name in f.keys():
                yield name, f.get_tensor(name)


def dummy_init(model: torch.nn.Module, seed: int = 0) -> None:
    """Random-init all params (reference --load-format dummy).

    Seeding must be (a) process-deterministic (crc32, not the salted
    built-in hash) and (b) PP-stage-aligned: a param named layers.0.* on
    stage 1 is GLOBAL layer `layer_start`, so the same global layer gets
    the same weights regardless of the pipeline split. TP shards still
    differ by rank (narrowed from differently-seeded full tensors is NOT
    done here — each rank draws its own shard; TP-replicated params get
    identical draws)."""
    import zlib
    layer_start = getattr(model, "layer_start", 0)

    def global_name(name: str) -> str:
        if name.startswith("layers."):
            parts = name.split(".")
            parts[1] = str(int(parts[1]) + layer_start)
            return ".".join(parts)
        return name

    params = sorted(model.named_parameters())
    on_gpu = any(p.is_cuda for _, p in params)
    gen = (torch.Generator(device="cuda") if on_gpu else torch.Generator())
    for name, p in params:
        gname = global_name(name)
        gen.manual_seed(seed ^ zlib.crc32(gname.encode()))
        with torch.no_grad():
            if p.dtype == torch.float8_e4m3fn:
                # fp8 dummy weights: draw bf16 then quantize (normal_
                # has no float8 kernel); keeps the packed layout live
                # on the --quant fp8 bench path
                tmp = torch.empty(p.shape, dtype=torch.bfloat16,
                                  device=p.device)
                tmp.normal_(0.0, 0.02, generator=gen)
                p.data.copy_(tmp.clamp(-0.4, 0.4).to(torch.float8_e4m3fn))
            elif "scale_inv" in gname:
                p.data.fill_(0.05)
            elif p.dtype in (torch.int32, torch.uint8):
                # packed int4 qweight/qzeros: random bits
                p.data.random_(generator=gen)
            elif ("scales" in gname or "weight_scale" in gname) \
                    and p.dim() >= 2:
                p.data.fill_(0.01)
            elif p.dim() >= 2:
                # draw in-place on device (32B params via a CPU RNG would
                # take minutes); bf16 normal_ is supported on ROCm
                p.data.normal_(0.0, 0.02, generator=gen)
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


def _normalize_compressed_tensors(cfg, qcfg):
    """Kimi-K2.5 ships compressed-tensors ``pack-quantized`` int4 on the
    ROUTED EXPERTS only (dense/shared layers stay bf16). Translate it to
    a moe_quantization_config hint and clear the dense-layer config
    (reference model_loader.py:538-591)."""
    num_bits, group = 4, 32
    for grp in (qcfg.get("config_groups") or {}).values():
        wc = (grp or {}).get("weights") or {}
        num_bits = int(wc.get("num_bits", num_bits))
        group = int(wc.get("group_size", group))
        break
    if num_bits != 4:
        raise ValueError(f"compressed-tensors: only int4, got {num_bits}")
    cfg.moe_quantization_config = {"quant_method": "int4_moe",
                                   "num_bits": 4, "group_size": group,
                                   "symmetric": True}
    cfg.quantization_config = None
    return None


def load_model(engine_config, device: str = "cpu"):
    t0 = time.time()
    cfg = load_hf_config(engine_config.model)
    qcfg = getattr(cfg, "quantization_config", None)
    if qcfg is not None and not isinstance(qcfg, dict):
        qcfg = getattr(qcfg, "to_dict", lambda: vars(qcfg))()
    if qcfg and qcfg.get("quant_method") == "compressed-tensors":
        qcfg = _normalize_compressed_tensors(cfg, qcfg)
    model = create_model(cfg, engine_config, device)
    if qcfg:  # dummy loads convert too: quantized EXECUTION
        # with random weights is exactly what bench --quant tests
        method = qcfg.get("quant_method")
        if method == "fp8":
            from gllm_amd.layers.quantization.fp8 import \
                convert_model_to_fp8
            n = convert_model_to_fp8(model, qcfg)
            logger.info("fp8 block-quant checkpoint: converted %d linears "
                        "(block %s)", n, qcfg.get("weight_block_size"))
        elif method in ("awq", "gptq"):
            from gllm_amd.layers.quantization.int4 import \
                convert_model_to_int4
            bits = qcfg.get("bits", qcfg.get("w_bit", 4))
            assert bits == 4, f"only 4-bit {method} supported"
            n = convert_model_to_int4(model, {
                "quant_method": method,
                "group_size": qcfg.get("group_size",
                                       qcfg.get("q_group_size", 128))})
            logger.info("%s int4 checkpoint: converted %d linears",
                        method, n)
    mq = getattr(cfg, "moe_quantization_config", None)
    if mq and mq.get("quant_method") == "int4_moe":
        from gllm_amd.layers.quantization.int4 import \
            convert_moe_to_int4_packed
        n = convert_moe_to_int4_packed(model, mq)
        logger.info("compressed-tensors int4 MoE: converted %d layers "
                    "(group %d); dense layers bf16", n, mq["group_size"])
    if qcfg or mq:
        # quant converters create replacement params on CPU; re-home
        # them next to the rest of the model
        model = model.to(device)
    if engine_config.load_format == "dummy":
        dummy_init(model, engine_config.seed)
    else:
        model.load_weights(iterate_safetensors(engine_config.model))
    logger.info("model loaded in %.1fs (%s, %d local layers)",
                time.time() - t0, type(model).__name__,
                model.num_local_layers)
    return model, cfg
