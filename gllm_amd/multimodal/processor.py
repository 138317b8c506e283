"""Server-side image preprocessing for the Qwen2-VL family.

Parity target: the reference's HF-processor pipeline
(model_runner.py:735-1245 runs AutoProcessor) + mm_common.py sentinel
expansion. Implemented natively (PIL + torch) so the serving path does
not depend on a matching transformers processor version:

  1. smart-resize to multiples of ``patch_size * spatial_merge_size``
     (28 for Qwen2-VL), preserving aspect ratio within min/max pixel
     budgets;
  2. rescale + normalize (OpenAI-CLIP mean/std — Qwen2-VL's values);
  3. patchify to the tower's flat layout
     [grid_t*grid_h*grid_w, C * temporal_patch_size * ps * ps] with the
     (t, h//2, w//2, 2, 2) merge-group ordering the tower expects;
  4. ``expand_image_tokens`` replaces each image sentinel with
     grid_thw.prod() // merge**2 image-pad tokens (mm_common.py:128).
"""

import base64
import io
import math
from typing import List, Tuple

import torch

# Qwen2-VL normalization (CLIP mean/std)
_MEAN = (0.48145466, 0.4578275, 0.40821073)
_STD = (0.26862954, 0.26130258, 0.27577711)


def smart_resize(h: int, w: int, factor: int = 28,
                 min_pixels: int = 56 * 56,
                 max_pixels: int = 14 * 14 * 4 * 1280
                 ) -> Tuple[int, int]:
    """Nearest (h, w) multiples of ``factor`` with h*w inside the pixel
    budget and aspect ratio preserved."""
    if max(h, w) / min(h, w) > 200:
        raise ValueError("absurd aspect ratio")
    hb = max(factor, round(h / factor) * factor)
    wb = max(factor, round(w / factor) * factor)
    if hb * wb > max_pixels:
        beta = math.sqrt((h * w) / max_pixels)
        hb = max(factor, math.floor(h / beta / factor) * factor)
        wb = max(factor, math.floor(w / beta / factor) * factor)
    elif hb * wb < min_pixels:
        beta = math.sqrt(min_pixels / (h * w))
        hb = math.ceil(h * beta / factor) * factor
        wb = math.ceil(w * beta / factor) * factor
    return hb, wb


class ImageProcessor:
    def __init__(self, patch_size: int = 14, temporal_patch_size: int = 2,
                 spatial_merge_size: int = 2,
                 min_pixels: int = 56 * 56,
                 max_pixels: int = 14 * 14 * 4 * 1280):
        self.ps = patch_size
        self.tps = temporal_patch_size
        self.merge = spatial_merge_size
        self.min_pixels = min_pixels
        self.max_pixels = max_pixels

    @classmethod
    def from_config(cls, vcfg) -> "ImageProcessor":
        g = (lambda k, d: vcfg.get(k, d)) if isinstance(vcfg, dict) \
            else (lambda k, d: getattr(vcfg, k, d))
        return cls(patch_size=g("patch_size", 14),
                   temporal_patch_size=g("temporal_patch_size", 2),
                   spatial_merge_size=g("spatial_merge_size", 2))

    def __call__(self, image) -> Tuple[torch.Tensor, Tuple[int, int, int]]:
        """PIL image -> (pixel patches [T, C*tps*ps*ps], (t, h, w) grid)."""
        import numpy as np
        from PIL import Image
        if not isinstance(image, Image.Image):
            image = Image.open(io.BytesIO(image))
        image = image.convert("RGB")
        H, W = smart_resize(image.height, image.width,
                            factor=self.ps * self.merge,
                            min_pixels=self.min_pixels,
                            max_pixels=self.max_pixels)
        image = image.resize((W, H), Image.BICUBIC)
        x = torch.from_numpy(
            np.array(image, copy=True)).float() / 255.0      # [H,W,C]
        mean = torch.tensor(_MEAN)
        std = torch.tensor(_STD)
        x = (x - mean) / std
        x = x.permute(2, 0, 1)                                   # [C,H,W]
        # temporal: a still image repeats across tps frames
        x = x.unsqueeze(0).expand(self.tps, -1, -1, -1)          # [tps,C,H,W]
        gh, gw = H // self.ps, W // self.ps
        grid = (1, gh, gw)
        m = self.merge
        # [t, C, gh//m, m, ps, gw//m, m, ps] -> merge-group-major rows
        p = x.reshape(self.tps, 3, gh // m, m, self.ps, gw // m, m,
                      self.ps)
        p = p.permute(2, 5, 3, 6, 1, 0, 4, 7)
        # rows: (gh/m, gw/m, m, m) order == tower's expected layout
        p = p.reshape(gh * gw, 3 * self.tps * self.ps * self.ps)
        return p.contiguous(), grid

    def num_tokens(self, grid: Tuple[int, int, int]) -> int:
        t, h, w = grid
        return t * h * w // (self.merge ** 2)


def decode_image_url(url: str) -> bytes:
    """data: URL (base64) -> raw bytes. http(s) fetching is declined —
    this deployment has no egress; callers embed images."""
    if url.startswith("data:"):
        _, b64 = url.split(",", 1)
        return base64.b64decode(b64)
    raise ValueError("only data: image URLs are supported "
                     "(no network egress)")


def expand_image_tokens(token_ids: List[int], sentinel_id: int,
                        counts: List[int], pad_id: int) -> List[int]:
    """Replace each occurrence of ``sentinel_id`` (one per image, in
    order) with counts[i] copies of ``pad_id`` (mm_common.py:128-143)."""
    out: List[int] = []
    i = 0
    for t in token_ids:
        if t == sentinel_id:
            assert i < len(counts), "more image sentinels than images"
            out.extend([pad_id] * counts[i])
            i += 1
        else:
            out.append(t)
    assert i == len(counts), "fewer image sentinels than images"
    return out
