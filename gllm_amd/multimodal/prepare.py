"""Multimodal admission: vision tower run + span/MRoPE bookkeeping.

Shared by the offline LLM (engine/llm.py) and the serving workers
(engine/worker.py). Reference role: model_runner.py:735-1245 (processor
outputs -> mrope positions -> embedding cache)."""

from typing import Optional

import torch

from gllm_amd.sequence import Sequence


def prepare_mm_seq(model, seq: Sequence, mm_input: dict) -> None:
    """Attach vision embeddings (when this rank hosts the tower), image
    spans and MRoPE positions to ``seq``.

    mm_input: {"pixel_values": [L, C*tps*ps*ps], "grids": [(t,h,w)...]}
    or {"embeds": [N, hidden], "grids": [...]}. Non-first PP stages skip
    the tower (the embedding merge happens on stage 0 only); they still
    need spans + mrope for batch building.
    """
    grids = mm_input["grids"]
    has_tower = getattr(model, "visual", None) is not None or \
        getattr(model, "vision_tower", None) is not None
    if "embeds" in mm_input:
        seq.mm_embeds = mm_input["embeds"]
    elif has_tower:
        with torch.no_grad():
            seq.mm_embeds = model.encode_images(
                mm_input["pixel_values"], grids).cpu()
    img_tok = model.image_token_id
    spans = []
    i = 0
    toks = seq.token_ids
    while i < len(toks):
        if toks[i] == img_tok:
            j = i
            while j < len(toks) and toks[j] == img_tok:
                j += 1
            spans.append((i, j - i))
            i = j
        else:
            i += 1
    if seq.mm_embeds is not None:
        assert sum(n for _, n in spans) == seq.mm_embeds.shape[0], \
            (spans, seq.mm_embeds.shape)
    seq.mm_spans = spans
    if getattr(model, "uses_mrope", False):
        from gllm_amd.layers.mrope import MRotaryEmbedding
        pos, delta = MRotaryEmbedding.get_input_positions(
            toks, img_tok, grids,
            spatial_merge_size=model.spatial_merge_size)
        seq.mrope_positions = pos
        seq.mrope_delta = delta
