"""Token sampler: greedy fast path, temperature, top-k/top-p, repetition
penalty, logprobs (reference: layers/sampler.py + repetition_penalty.py).

Round-1 note: top-k/top-p runs as a masked-sort composite on GPU; the
fused single-pass HIP sampling kernel replaces it in a later pass.
"""

import dataclasses
from typing import List, Optional

import torch

from gllm_amd.ops import torch_ref


@dataclasses.dataclass
class SamplingMetadata:
    temperatures: torch.Tensor          # [B] float32 (0 => greedy row)
    top_ps: torch.Tensor                # [B] float32
    top_ks: torch.Tensor                # [B] int32 (-1 disabled)
    penalties: torch.Tensor             # [B] float32
    all_greedy: bool
    any_penalty: bool
    min_ps: Optional[torch.Tensor] = None  # [B] float32 (0 disabled)
    # OpenAI presence/frequency penalties (additive over OUTPUT tokens);
    # applied on the host token-row path
    pres_freq_rows: Optional[List] = None  # [(row, out_ids, pres, freq)]
    # token history per row for repetition penalty (cpu LongTensors),
    # used only as the fallback when no mask pool slot is available
    token_id_rows: Optional[List[torch.Tensor]] = None
    # persistent mask pool path: pool ref + per-row slot ids (-1 = none)
    penalty_pool: Optional[object] = None
    penalty_slots: Optional[torch.Tensor] = None   # [B] long
    # rows that actually sample a kept token (ends_prompt), host list
    sample_rows: Optional[List[int]] = None
    # per-row logits adjustments (host-built, device tensors):
    # logit_bias additive entries, allowed-token whitelists and
    # bad-words bans (-inf outside/on the listed ids)
    bias_rows: Optional[List] = None     # [(row, ids, vals)]
    allowed_rows: Optional[List] = None  # [(row, ids)]
    ban_rows: Optional[List] = None      # [(row, ids)]
    max_logprobs: int = 0               # >0 => return top-k logprobs
    generators: Optional[List[Optional[torch.Generator]]] = None


@dataclasses.dataclass
class SamplerOutput:
    next_tokens: torch.Tensor           # [B] int64 (device)
    logprobs: Optional[torch.Tensor] = None        # [B] chosen-token logprob
    topk_logprobs: Optional[torch.Tensor] = None   # [B, K]
    topk_token_ids: Optional[torch.Tensor] = None  # [B, K]


class Sampler(torch.nn.Module):
    def forward(self, logits: torch.Tensor,
                meta: SamplingMetadata) -> SamplerOutput:
        # logits: [B, V] (already gathered to full vocab)
        if meta.any_penalty:
            logits = self._apply_penalties(logits, meta)
        if meta.bias_rows or meta.allowed_rows or meta.ban_rows:
            logits = self._apply_row_adjust(logits, meta)
        if meta.all_greedy:
            next_tokens = logits.argmax(dim=-1)
            return self._with_logprobs(logits, next_tokens, meta)

        logits = logits.float()
        temps = meta.temperatures.clamp_min(1e-5).unsqueeze(-1)
        scaled = logits / temps
        probs = torch.softmax(scaled, dim=-1)
        if probs.is_cuda:
            # fused sorting-free radix-select kernel (in-place),
            # replaces the [B, V] sort composite below
            from gllm_amd import ops
            if ops.has_kernels():
                probs = probs.contiguous()
                ops.topk_topp_filter(probs, meta.top_ks,
                                     meta.top_ps, meta.min_ps)
                sampled = self._multinomial(probs, meta)
                greedy_rows = meta.temperatures == 0.0
                if greedy_rows.any():
                    sampled = torch.where(greedy_rows,
                                          logits.argmax(dim=-1), sampled)
                return self._with_logprobs(logits, sampled, meta)
        probs = self._apply_top_k_top_p(probs, meta.top_ks, meta.top_ps)
        probs = self._apply_min_p(probs, meta.min_ps)
        sampled = self._multinomial(probs, meta)
        # greedy rows override
        greedy_rows = meta.temperatures == 0.0
        if greedy_rows.any():
            sampled = torch.where(greedy_rows, logits.argmax(dim=-1), sampled)
        return self._with_logprobs(logits, sampled, meta)

    @staticmethod
    def _apply_penalties(logits: torch.Tensor,
                         meta: SamplingMetadata) -> torch.Tensor:
        logits = logits.float()
        if meta.penalty_pool is not None and meta.penalty_slots is not None:
            if logits.is_cuda:
                # fused gather+scale HIP kernel (csrc/sampling.hip)
                from gllm_amd import ops
                logits = logits.contiguous()
                ops.apply_penalty_pool(
                    logits, meta.penalty_pool.mask, meta.penalty_slots,
                    meta.penalties.to(logits.device).float())
            else:
                slots = meta.penalty_slots.clamp_min(0)
                rows = meta.penalty_pool.mask[slots, :logits.shape[1]]
                pen = meta.penalties.to(logits.device).unsqueeze(1)
                active = (rows != 0) & (pen != 1.0) & \
                    (meta.penalty_slots.unsqueeze(1) >= 0)
                adj = torch.where(logits > 0, logits / pen, logits * pen)
                logits = torch.where(active, adj, logits)
        if meta.token_id_rows is not None:
            logits = torch_ref.apply_repetition_penalty(
                logits, meta.token_id_rows, meta.penalties)
        if meta.pres_freq_rows:
            for row, out_ids, pres, freq in meta.pres_freq_rows:
                if out_ids.numel() == 0:
                    continue
                counts = torch.bincount(out_ids,
                                        minlength=logits.shape[1]
                                        ).to(logits.device)
                seen = counts > 0
                logits[row] = logits[row] - pres * seen.float() \
                    - freq * counts.float()
        return logits

    @staticmethod
    def _apply_row_adjust(logits: torch.Tensor,
                          meta: SamplingMetadata) -> torch.Tensor:
        """logit_bias / allowed_token_ids / bad_words, applied before
        the greedy argmax and the softmax alike."""
        logits = logits.float()
        if meta.bias_rows:
            for row, ids, vals in meta.bias_rows:
                logits[row, ids] += vals
        if meta.allowed_rows:
            for row, ids in meta.allowed_rows:
                keep = logits[row, ids].clone()
                logits[row] = float("-inf")
                logits[row, ids] = keep
        if meta.ban_rows:
            for row, ids in meta.ban_rows:
                logits[row, ids] = float("-inf")
        return logits

    @staticmethod
    def _apply_min_p(probs: torch.Tensor, min_ps) -> torch.Tensor:
        """vLLM-style min_p: drop tokens whose probability is below
        min_p * max_prob of the row (applied after top-k/top-p)."""
        if min_ps is None or not bool((min_ps > 0).any()):
            return probs
        thresh = probs.amax(dim=-1, keepdim=True) * min_ps.unsqueeze(-1)
        probs = probs.masked_fill(probs < thresh, 0.0)
        return probs / probs.sum(-1, keepdim=True).clamp_min(1e-20)

    @staticmethod
    def _apply_top_k_top_p(probs: torch.Tensor, top_ks: torch.Tensor,
                           top_ps: torch.Tensor) -> torch.Tensor:
        B, V = probs.shape
        need_k = bool((top_ks > 0).any())
        need_p = bool((top_ps < 1.0).any())
        if not need_k and not need_p:
            return probs
        sorted_probs, idx = probs.sort(dim=-1, descending=True)
        if need_k:
            ks = torch.where(top_ks > 0, top_ks, torch.full_like(top_ks, V))
            rank = torch.arange(V, device=probs.device).unsqueeze(0)
            sorted_probs = sorted_probs.masked_fill(
                rank >= ks.unsqueeze(-1), 0.0)
        if need_p:
            cum = sorted_probs.cumsum(dim=-1)
            # keep tokens whose cumulative mass (exclusive) < top_p
            exclusive = cum - sorted_probs
            sorted_probs = sorted_probs.masked_fill(
                exclusive > top_ps.unsqueeze(-1), 0.0)
        out = torch.zeros_like(probs)
        out.scatter_(-1, idx, sorted_probs)
        return out / out.sum(-1, keepdim=True).clamp_min(1e-20)

    @staticmethod
    def _multinomial(probs: torch.Tensor, meta: SamplingMetadata):
        gens = meta.generators
        if gens and any(g is not None for g in gens):
            outs = []
            for i in range(probs.shape[0]):
                outs.append(torch.multinomial(
                    probs[i:i + 1], 1, generator=gens[i]).squeeze(-1))
            return torch.cat(outs)
        return torch.multinomial(probs, 1).squeeze(-1)

    @staticmethod
    def _with_logprobs(logits: torch.Tensor, next_tokens: torch.Tensor,
                       meta: SamplingMetadata) -> SamplerOutput:
        if meta.max_logprobs <= 0:
            return SamplerOutput(next_tokens)
        logp = torch.log_softmax(logits.float(), dim=-1)
        chosen = logp.gather(-1, next_tokens.unsqueeze(-1)).squeeze(-1)
        k = meta.max_logprobs
        topv, topi = logp.topk(k, dim=-1)
        return SamplerOutput(next_tokens, chosen, topv, topi)


def build_sampling_metadata(items, device,
                            penalty_pool=None) -> SamplingMetadata:
    """Build metadata from the scheduled batch items (one row per item).

    With a ``penalty_pool``, penalized seqs get persistent device mask
    slots (seeded from the prompt on first use, updated on-GPU after
    every sample); the host token-row fallback covers pool exhaustion."""
    temps, tps, tks, mps, pens, gens = [], [], [], [], [], []
    rows: List[torch.Tensor] = []
    slots: List[int] = []
    sample_rows: List[int] = []
    max_lp = 0
    any_pen = False
    need_rows = False
    pres_freq = []
    bias_rows, allowed_rows, ban_rows = [], [], []
    for i, it in enumerate(items):
        sp = it.seq.sampling
        if sp.logit_bias:
            ids = torch.tensor(list(sp.logit_bias.keys()),
                               dtype=torch.long, device=device)
            vals = torch.tensor(list(sp.logit_bias.values()),
                                dtype=torch.float32, device=device)
            bias_rows.append((i, ids, vals))
        if sp.allowed_token_ids:
            allowed_rows.append((i, torch.tensor(
                sp.allowed_token_ids, dtype=torch.long, device=device)))
        if sp.bad_words_token_ids:
            ctx = it.seq.token_ids
            banned = set()
            for w in sp.bad_words_token_ids:
                if not w:
                    continue
                if len(w) == 1 or (len(w) - 1 <= len(ctx) and
                                   ctx[len(ctx) - len(w) + 1:] == w[:-1]):
                    banned.add(w[-1])
            if banned:
                ban_rows.append((i, torch.tensor(
                    sorted(banned), dtype=torch.long, device=device)))
        temps.append(sp.temperature)
        tps.append(sp.top_p)
        tks.append(sp.top_k)
        mps.append(getattr(sp, "min_p", 0.0))
        pens.append(sp.repetition_penalty)
        if it.ends_prompt:
            sample_rows.append(i)
        slot = -1
        if sp.repetition_penalty != 1.0:
            any_pen = True
            if penalty_pool is not None:
                slot = penalty_pool.ensure(it.seq)
            if slot < 0:
                need_rows = True
                rows.append(torch.tensor(it.seq.token_ids,
                                         dtype=torch.long, device=device))
            else:
                rows.append(torch.empty(0, dtype=torch.long,
                                        device=device))
        else:
            rows.append(torch.empty(0, dtype=torch.long, device=device))
        slots.append(slot)
        pres = getattr(sp, "presence_penalty", 0.0) or 0.0
        freq = getattr(sp, "frequency_penalty", 0.0) or 0.0
        if pres or freq:
            any_pen = True
            out_ids = torch.tensor(
                it.seq.token_ids[it.seq.prompt_len:], dtype=torch.long)
            pres_freq.append((i, out_ids, pres, freq))
        if sp.logprobs:
            max_lp = max(max_lp, sp.logprobs)
        if sp.seed is not None:
            g = torch.Generator(device=device)
            g.manual_seed(sp.seed + it.seq.num_output_tokens)
            gens.append(g)
        else:
            gens.append(None)
    all_greedy = all(t == 0.0 for t in temps)
    return SamplingMetadata(
        temperatures=torch.tensor(temps, dtype=torch.float32, device=device),
        top_ps=torch.tensor(tps, dtype=torch.float32, device=device),
        top_ks=torch.tensor(tks, dtype=torch.int32, device=device),
        min_ps=torch.tensor(mps, dtype=torch.float32, device=device),
        penalties=torch.tensor(pens, dtype=torch.float32, device=device),
        all_greedy=all_greedy, any_penalty=any_pen,
        token_id_rows=rows if need_rows else None,
        penalty_pool=penalty_pool if any_pen else None,
        penalty_slots=torch.tensor(slots, dtype=torch.long, device=device)
        if any_pen else None,
        sample_rows=sample_rows,
        pres_freq_rows=pres_freq or None,
        bias_rows=bias_rows or None,
        allowed_rows=allowed_rows or None,
        ban_rows=ban_rows or None,
        max_logprobs=max_lp,
        generators=gens if any(g is not None for g in gens) else None)
