"""Kimi-K2.5: MoonViT3d vision tower on a DeepSeek-V3 LM backbone.

Parity target: reference models/kimi_k25.py. Unlike the Qwen-VL family
this model uses PLAIN 1-D positions for vision tokens (no MRoPE,
reference model_runner.py:313-320), and the tower is replicated on
every first-stage rank. Image placeholders are runs of
``media_placeholder_token_id`` (one merged-token per run element —
callers expand placeholders before admission, mm_common.py style; the
server-side Kimi image chunker is round 2). Checkpoint prefixes:
``language_model.*`` / ``vision_tower.*`` / ``mm_projector.*``.
"""

from typing import Iterable, Tuple

import torch

from gllm_amd.models.deepseek_v2 import DeepseekV2ForCausalLM
from gllm_amd.models.kimi_k25_vision import KimiPatchMerger, KimiVisionTower


class KimiK25ForCausalLM(DeepseekV2ForCausalLM):
    uses_mrope = False

    def __init__(self, cfg, engine_config):
        super().__init__(cfg, engine_config)
        vcfg = getattr(cfg, "vision_config", None)
        if vcfg is not None and not isinstance(vcfg, dict):
            vcfg = dict(vars(vcfg))
        self.image_token_id = getattr(cfg, "media_placeholder_token_id",
                                      None)
        if (vcfg is not None and self.is_first_stage
                and not getattr(engine_config, "skip_visual", False)):
            dtype = engine_config.torch_dtype()
            vcfg.setdefault("text_hidden_size", cfg.hidden_size)
            vcfg.setdefault("mm_hidden_size", vcfg["vt_hidden_size"])
            self.vision_tower = KimiVisionTower(vcfg, dtype=dtype)
            self.mm_projector = KimiPatchMerger(vcfg, dtype=dtype)
        else:
            self.vision_tower = None
            self.mm_projector = None

    def encode_images(self, pixel_values: torch.Tensor, grids):
        assert self.vision_tower is not None
        items = self.vision_tower(pixel_values, grids)
        return torch.cat(self.mm_projector(items), dim=0)

    def load_weights(self, weights: Iterable[Tuple[str, torch.Tensor]]):
        vis, proj = [], []

        def split():
            for name, w in weights:
                if name.startswith("vision_tower."):
                    vis.append((name[len("vision_tower."):], w))
                elif name.startswith("mm_projector."):
                    proj.append((name[len("mm_projector."):], w))
                else:
                    if name.startswith("language_model."):
                        name = name[len("language_model."):]
                    yield name, w

        super().load_weights(split())
        for mod, sd in ((self.vision_tower, vis),
                        (self.mm_projector, proj)):
            if mod is not None and sd:
                missing, _ = mod.load_state_dict(dict(sd), strict=False)
                if missing:
                    from gllm_amd.logger import logger
                    logger.warning("kimi vision weights missing: %s",
                                   sorted(missing)[:5])
