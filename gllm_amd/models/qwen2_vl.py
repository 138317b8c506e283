"""Qwen2-VL: vision tower + mrope language model.

Parity target: reference models/qwen2_5_vl.py (ViT tower, mrope,
embed_multimodal merge). Full path: offline (pixel patches + grids or
precomputed embeddings per request) AND serving (native image
processor + OpenAI image_url content, docs/multimodal.md); the tower
runs at admission (ViT-output LRU) or in a disaggregated encoder
process; embeddings merge at the image-pad rows chunk-aware; prefix
caching uses content-hash keys for image runs. hipGraphs stay off for
mrope models ([3,B] graph position buffers are round 2)."""

from typing import Iterable, Tuple

import torch

from gllm_amd.models.llama_family import Qwen2ForCausalLM
from gllm_amd.models.qwen2_vl_vision import Qwen2VisionTransformer


class Qwen2VLForCausalLM(Qwen2ForCausalLM):
    uses_mrope = True

    def __init__(self, cfg, engine_config):
        # the LM side reuses Qwen2 (mrope comes from cfg.rope_scaling's
        # mrope_section via the rope factory)
        super().__init__(cfg, engine_config)
        vcfg = getattr(cfg, "vision_config", None)
        if isinstance(vcfg, dict):
            import types
            vcfg = types.SimpleNamespace(**vcfg)
        self.image_token_id = getattr(cfg, "image_token_id", None)
        self.spatial_merge_size = getattr(vcfg, "spatial_merge_size", 2) \
            if vcfg is not None else 2
        if (self.is_first_stage and vcfg is not None
                and not getattr(engine_config, "skip_visual", False)):
            # Qwen2.5-VL configs carry window attention fields; the
            # plain Qwen2-VL tower is full-attention LayerNorm/QuickGELU
            if getattr(vcfg, "window_size", None) is not None:
                from gllm_amd.models.qwen2_vl_vision import \
                    Qwen25VisionTransformer
                self.visual = Qwen25VisionTransformer(
                    vcfg, dtype=engine_config.torch_dtype())
            else:
                self.visual = Qwen2VisionTransformer(
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
            sd = dict(vis)
            missing, unexpected = self.visual.load_state_dict(sd,
                                                              strict=False)
            if missing:
                from gllm_amd.logger import logger
                logger.warning("vision weights missing: %s",
                               sorted(missing)[:5])
