"""Qwen3-VL (+MoE): deepstack vision tower + Qwen3 LM with interleaved
MRoPE.

Parity target: reference models/qwen3_vl.py + qwen3_vl_moe.py. The
tower (models/qwen3_vl_vision.py) emits [N, out_hidden * (1 + D)]
multiscale embeddings; level 0 replaces the image-pad rows after
embed_tokens and level d is ADDED at those rows after decoder layer d
(llama_family.py forward, fctx.mm_deepstack)."""

from types import SimpleNamespace
from typing import Iterable, Tuple

import torch

from gllm_amd.models.hybrid_gdn import Qwen3_5ForCausalLM
from gllm_amd.models.llama_family import Qwen3ForCausalLM
from gllm_amd.models.moe_family import Qwen3MoeForCausalLM
from gllm_amd.models.qwen3_vl_vision import Qwen3VisionTransformer


class _Qwen3VLMixin:
    uses_mrope = True

    def _init_vision(self, cfg, engine_config):
        vcfg = getattr(cfg, "vision_config", None)
        if isinstance(vcfg, dict):
            vcfg = SimpleNamespace(**vcfg)
        self.image_token_id = getattr(cfg, "image_token_id", None)
        self.spatial_merge_size = getattr(vcfg, "spatial_merge_size", 2) \
            if vcfg is not None else 2
        if (self.is_first_stage and vcfg is not None
                and not getattr(engine_config, "skip_visual", False)):
            self.visual = Qwen3VisionTransformer(
                vcfg, dtype=engine_config.torch_dtype())
        else:
            self.visual = None

    def encode_images(self, pixel_values: torch.Tensor, grids):
        assert self.visual is not None
        dev = next(self.visual.parameters()).device
        return self.visual(pixel_values.to(dev), grids)

    def load_weights(self, weights: Iterable[Tuple[str, torch.Tensor]]):
        vis = []

        def split():
            for name, w in weights:
                if name.startswith("visual."):
                    vis.append((name[len("visual."):], w))
                else:
                    yield name, w

        super().load_weights(split())
        if self.visual is not None and vis:
            missing, _ = self.visual.load_state_dict(dict(vis),
                                                     strict=False)
            if missing:
                from gllm_amd.logger import logger
                logger.warning("vision weights missing: %s",
                               sorted(missing)[:5])


class Qwen3VLForCausalLM(_Qwen3VLMixin, Qwen3ForCausalLM):
    def __init__(self, cfg, engine_config):
        super().__init__(cfg, engine_config)
        self._init_vision(cfg, engine_config)


class Qwen3VLMoeForCausalLM(_Qwen3VLMixin, Qwen3MoeForCausalLM):
    def __init__(self, cfg, engine_config):
        super().__init__(cfg, engine_config)
        self._init_vision(cfg, engine_config)


class Qwen3_5VLForCausalLM(_Qwen3VLMixin, Qwen3_5ForCausalLM):
    """Qwen3.5-VL: Qwen3-VL vision tower + the hybrid-GDN text LM
    (reference qwen3_5.py:1093-1107 Qwen3_5ForConditionalGeneration —
    the MoE flavour routes through the same class via config)."""

    def __init__(self, cfg, engine_config):
        super().__init__(cfg, engine_config)
        self._init_vision(cfg, engine_config)
