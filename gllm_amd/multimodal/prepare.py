"""Multimodal admission: vision tower run + span/MRoPE bookkeeping.

Shared by the offline LLM (engine/llm.py) and the serving workers
(engine/worker.py). Reference role: model_runner.py:735-1245 (processor
outputs -> mrope positions -> per-item content hashing) +
MultiModalEmbeddingCache (:161-221): repeated images skip the tower via
a content-hash LRU of merged embeddings."""

import collections
import hashlib

import torch

from gllm_amd.sequence import Sequence

# content-hash LRU of tower outputs (per worker process)
_EMB_CACHE: "collections.OrderedDict[int, torch.Tensor]" = \
    collections.OrderedDict()
_EMB_CACHE_MAX = 64
emb_cache_hits = 0


def _content_hash(src: torch.Tensor, grids) -> int:
    digest = hashlib.sha256()
    digest.update(repr([tuple(g) for g in grids]).encode())
    digest.update(src.detach().to(torch.float32).cpu().numpy().tobytes())
    return int.from_bytes(digest.digest()[:8], "little")


def prepare_mm_seq(model, seq: Sequence, mm_input: dict) -> None:
    """Attach vision embeddings (when this rank hosts the tower), image
    spans, content-hash cache keys and MRoPE positions to ``seq``.

    mm_input: {"pixel_values": [L, C*tps*ps*ps], "grids": [(t,h,w)...]}
    or {"embeds": [N, hidden], "grids": [...]}. Non-first PP stages skip
    the tower (the embedding merge happens on stage 0 only); they still
    need spans + mrope for batch building.
    """
    global emb_cache_hits
    grids = mm_input["grids"]
    has_tower = getattr(model, "visual", None) is not None or \
        getattr(model, "vision_tower", None) is not None
    src = mm_input.get("pixel_values", mm_input.get("embeds"))
    h = _content_hash(src, grids) if src is not None else None
    if "embeds" in mm_input:
        seq.mm_embeds = mm_input["embeds"]
    elif has_tower:
        cached = _EMB_CACHE.get(h)
        if cached is not None:
            _EMB_CACHE.move_to_end(h)
            emb_cache_hits += 1
            seq.mm_embeds = cached
        else:
            with torch.no_grad():
                seq.mm_embeds = model.encode_images(
                    mm_input["pixel_values"], grids).cpu()
            _EMB_CACHE[h] = seq.mm_embeds
            while len(_EMB_CACHE) > _EMB_CACHE_MAX:
                _EMB_CACHE.popitem(last=False)
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
    # prefix-cache keys: substitute each image run's pad tokens with
    # content-derived pseudo-ids so identical text + different pixels
    # never alias a cached page (core/kv_cache.py _key_ids). sha digest:
    # deterministic across processes/ranks.
    if h is not None and spans:
        keys = list(toks)
        for si, (s, n) in enumerate(spans):
            base = h ^ (si * 0x9E3779B97F4A7C15 & (1 << 63) - 1)
            for j in range(n):
                keys[s + j] = -(
                    (base + j * 0x100000001B3) & ((1 << 62) - 1)) - 10
        seq.cache_key_ids = keys
    if getattr(model, "uses_mrope", False):
        from gllm_amd.layers.mrope import MRotaryEmbedding
        pos, delta = MRotaryEmbedding.get_input_positions(
            toks, img_tok, grids,
            spatial_merge_size=model.spatial_merge_size)
        seq.mrope_positions = pos
        seq.mrope_delta = delta
