"""Qwen2-VL vision tower unit tests (multimodal staging)."""

import types

import torch

from gllm_amd.models.qwen2_vl_vision import Qwen2VisionTransformer


def _vcfg():
    return types.SimpleNamespace(
        depth=2, embed_dim=64, hidden_size=96, num_heads=4, mlp_ratio=2.0,
        patch_size=14, temporal_patch_size=2, in_channels=3,
        spatial_merge_size=2)


def test_vision_tower_shapes_and_determinism():
    torch.manual_seed(0)
    vit = Qwen2VisionTransformer(_vcfg(), dtype=torch.float32)
    grid = [(1, 4, 6)]                 # 24 patches -> 6 merged tokens
    L = sum(t * h * w for t, h, w in grid)
    px = torch.randn(L, 3 * 2 * 14 * 14)
    out = vit(px, grid)
    assert out.shape == (L // 4, 96)
    assert torch.isfinite(out).all()
    out2 = vit(px, grid)
    assert torch.equal(out, out2)


def test_vision_tower_two_images_independent():
    """Full attention must stay WITHIN each image (cu_seqlens): image 2's
    output is unchanged when image 1's pixels change."""
    torch.manual_seed(1)
    vit = Qwen2VisionTransformer(_vcfg(), dtype=torch.float32)
    grid = [(1, 2, 2), (1, 4, 4)]
    L1, L2 = 4, 16
    px = torch.randn(L1 + L2, 3 * 2 * 14 * 14)
    out = vit(px, grid)
    px2 = px.clone()
    px2[:L1] = torch.randn(L1, 3 * 2 * 14 * 14)
    out2 = vit(px2, grid)
    assert not torch.allclose(out[:L1 // 4], out2[:L1 // 4])
    assert torch.allclose(out[L1 // 4:], out2[L1 // 4:], atol=1e-5)


def test_vision_checkpoint_names_match_hf_layout():
    vit = Qwen2VisionTransformer(_vcfg(), dtype=torch.float32)
    names = {n for n, _ in vit.named_parameters()}
    for expect in [
        "patch_embed.proj.weight",
        "blocks.0.norm1.weight", "blocks.0.attn.qkv.weight",
        "blocks.0.attn.qkv.bias", "blocks.0.attn.proj.weight",
        "blocks.0.mlp.fc1.weight", "blocks.0.mlp.fc2.bias",
        "merger.ln_q.weight", "merger.mlp.0.weight", "merger.mlp.2.weight",
    ]:
        assert expect in names, expect


def test_qwen25_tower_windowed_attention():
    """Qwen2.5-VL tower: window partition correctness + the full-attn
    blocks actually change the result vs all-window."""
    import types
    import torch
    from gllm_amd.models.qwen2_vl_vision import (Qwen25VisionTransformer,
                                                 window_index_thw)
    # window math: 1x8x8 patches, merge 2, window 2 merge-units
    idx, cu = window_index_thw(1, 8, 8, 2, 2)
    assert sorted(idx.tolist()) == list(range(16))
    assert cu.tolist() == [16, 32, 48, 64]  # 4 windows x 4 units x 4

    vcfg = dict(depth=2, hidden_size=32, out_hidden_size=64, num_heads=4,
                intermediate_size=48, patch_size=14,
                temporal_patch_size=2, in_channels=3,
                spatial_merge_size=2, window_size=56,
                fullatt_block_indexes=[1])
    torch.manual_seed(0)
    tower = Qwen25VisionTransformer(types.SimpleNamespace(**vcfg))
    px = torch.randn(64, 3 * 2 * 14 * 14)
    out = tower(px, [(1, 8, 8)])
    assert out.shape == (16, 64)

    # identical weights, but every block windowed -> different output
    vcfg2 = dict(vcfg, fullatt_block_indexes=[])
    torch.manual_seed(0)
    tower2 = Qwen25VisionTransformer(types.SimpleNamespace(**vcfg2))
    out2 = tower2(px, [(1, 8, 8)])
    assert not torch.allclose(out, out2, atol=1e-5), \
        "full-attention blocks must see beyond their window"

    # permutation sanity: un-permuted output must be deterministic
    assert torch.allclose(out, tower(px, [(1, 8, 8)]), atol=1e-6)
