"""ScheduledBatch -> ForwardContext tensors.

Role parity with the reference InputData.cal_input (input_data.py:112,
vectorized numpy fills :338-533): build per-forward tensors on CPU with
numpy, stage through pinned memory, async-copy to the GPU.
"""

from typing import List, Optional, Tuple

import numpy as np
import torch

from gllm_amd.core.scheduler import ScheduledBatch, ScheduledSeq
from gllm_amd.runtime.forward_context import ForwardContext


class BatchBuilder:
    def __init__(self, page_size: int, device: str, pin: bool = True):
        self.page_size = page_size
        self.device = device
        self.pin = pin and device != "cpu"

    def build(self, batch: ScheduledBatch, k_caches, v_caches,
              need_logits: bool = True,
              use_mrope: bool = False) -> Tuple[torch.Tensor,
                                                ForwardContext]:
        items = batch.items
        ps = self.page_size
        B = len(items)
        lens = np.fromiter((it.num_tokens for it in items), dtype=np.int64,
                           count=B)
        T = int(lens.sum())
        qsl = np.zeros(B + 1, dtype=np.int32)
        np.cumsum(lens, out=qsl[1:])

        tokens = np.empty(T, dtype=np.int64)
        positions = np.empty((3, T) if use_mrope else T, dtype=np.int64)
        slots = np.empty(T, dtype=np.int64)
        seq_lens = np.empty(B, dtype=np.int32)
        max_pages = max(-(-(it.start + it.num_tokens) // ps) for it in items)
        block_table = np.zeros((B, max_pages), dtype=np.int32)

        for i, it in enumerate(items):
            s, n = it.start, it.num_tokens
            o = qsl[i]
            tokens[o:o + n] = it.seq.token_ids[s:s + n]
            if use_mrope:
                seq = it.seq
                mp = seq.mrope_positions
                for j in range(n):
                    pos = s + j
                    if mp is not None and pos < seq.prompt_len:
                        positions[:, o + j] = mp[:, pos].numpy()
                    elif mp is not None:
                        positions[:, o + j] = seq.mrope_delta + \
                            (pos - seq.prompt_len)
                    else:
                        positions[:, o + j] = pos
            else:
                positions[o:o + n] = np.arange(s, s + n)
            pt = np.asarray(it.seq.page_table, dtype=np.int64)
            pos = np.arange(s, s + n)
            slots[o:o + n] = pt[pos // ps] * ps + pos % ps
            seq_lens[i] = s + n
            block_table[i, :pt.shape[0]] = pt

        has_ph = bool((tokens < 0).any())
        max_q = int(lens.max())
        max_s = int(seq_lens.max())
        dev = self.device

        def to_dev(arr, dtype):
            t = torch.from_numpy(arr)
            if dev != "cpu":
                if self.pin:
                    t = t.pin_memory()
                t = t.to(dev, non_blocking=True)
            return t

        tokens_t = to_dev(tokens, None)
        fctx = ForwardContext(
            num_tokens=T,
            positions=to_dev(positions, None),
            slot_mapping=to_dev(slots, None),
            block_table=to_dev(block_table, None),
            seq_lens=to_dev(seq_lens, None),
            query_start_loc=to_dev(qsl, None),
            max_query_len=max_q,
            max_seq_len=max_s,
            k_caches=k_caches,
            v_caches=v_caches,
        )
        fctx.has_placeholders = has_ph
        fctx.seq_lens_cpu = seq_lens.tolist()
        fctx.query_start_loc_cpu = qsl.tolist()
        if need_logits:
            fctx.logits_indices = (fctx.query_start_loc[1:].long() - 1)
        return tokens_t, fctx
