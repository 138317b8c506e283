"""MRoPE building-block tests (multimodal path, round-2 staging)."""

import torch

from gllm_amd.layers.mrope import MRotaryEmbedding
from gllm_amd.layers.rotary import RotaryEmbedding


def _mk(head_dim=32, sections=(4, 6, 6)):
    return MRotaryEmbedding(head_dim, head_dim, 1024, 10000.0,
                            list(sections))


def test_equal_streams_match_plain_rope():
    torch.manual_seed(0)
    m = _mk()
    plain = RotaryEmbedding(32, 32, 1024, 10000.0, is_neox=True)
    T = 9
    q = torch.randn(T, 2 * 32)
    k = torch.randn(T, 32)
    pos1 = torch.arange(T)
    pos3 = pos1.unsqueeze(0).expand(3, T).contiguous()
    q1, k1 = plain.forward(pos1, q.clone(), k.clone())
    q3, k3 = m.forward(pos3, q.clone(), k.clone())
    assert torch.allclose(q1, q3, atol=1e-5)
    assert torch.allclose(k1, k3, atol=1e-5)


def test_text_only_positions_are_arange():
    pos, delta = MRotaryEmbedding.get_input_positions(
        [5, 6, 7, 8], image_token_id=99, image_grids=[])
    assert torch.equal(pos, torch.arange(4).unsqueeze(0).expand(3, 4))
    assert delta == 4


def test_image_span_positions():
    # 2 text + image (t=1, 4x4 patches -> 2x2 merged = 4 tokens) + 1 text
    toks = [1, 2] + [99] * 4 + [3]
    pos, delta = MRotaryEmbedding.get_input_positions(
        toks, image_token_id=99, image_grids=[(1, 4, 4)],
        spatial_merge_size=2)
    # text prefix
    assert pos[:, 0].tolist() == [0, 0, 0]
    assert pos[:, 1].tolist() == [1, 1, 1]
    # vision tokens: t stream constant, h/w vary within the 2x2 grid
    assert pos[0, 2:6].tolist() == [2, 2, 2, 2]
    assert pos[1, 2:6].tolist() == [2, 2, 3, 3]
    assert pos[2, 2:6].tolist() == [2, 3, 2, 3]
    # trailing text continues after max stream advance (2 + max(1,2,2)=4)
    assert pos[:, 6].tolist() == [4, 4, 4]
    assert delta == 5
    nxt = MRotaryEmbedding.get_next_input_positions(delta, 0)
    assert nxt.flatten().tolist() == [5, 5, 5]


def test_mrope_differs_when_streams_differ():
    m = _mk()
    T = 4
    q = torch.randn(T, 64)
    k = torch.randn(T, 32)
    same = torch.arange(T).unsqueeze(0).expand(3, T).contiguous()
    diff = same.clone()
    diff[1] += 3
    q1, _ = m.forward(same, q.clone(), k.clone())
    q2, _ = m.forward(diff, q.clone(), k.clone())
    assert not torch.allclose(q1, q2)
