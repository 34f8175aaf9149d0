"""Speculative decoding: a small draft model proposes gamma tokens, the
target model verifies them in ONE forward, and rejection sampling keeps the
target distribution exact.

Reference behavior: the reference's speculative-decoding inference mode
(llm/predict, speculate_* configs).  MI355X rationale: decode is
memory-bandwidth-bound (one weight sweep per token); verification reads the
target weights once per gamma+1 tokens, so acceptance rate a gives ~a*gamma
tokens per sweep instead of 1.

Greedy mode accepts while the target argmax equals the draft token — output
is IDENTICAL to target-only greedy decoding.  Sampling mode implements the
Leviathan accept/reject rule: accept draft token x with prob
min(1, p_target(x)/p_draft(x)), else resample from max(0, p-q) normalized.
"""
from __future__ import annotations

from typing import Optional, Tuple

import torch

from .configuration_utils import GenerationConfig


def _truncate_cache(past, length: int):
    """Slice every layer's (k, v) cache to `length` positions ([B,S,H,D])."""
    if past is None:
        return None
    return [(k[:, :length], v[:, :length]) for k, v in past]


@torch.no_grad()
def speculative_generate(
    target, draft, input_ids: torch.Tensor,
    generation_config: Optional[GenerationConfig] = None,
    gamma: int = 4, **kwargs,
) -> Tuple[torch.Tensor, dict]:
    """Batch-1 speculative decode.  Returns (generated_ids [1, new], stats)."""
    gen = generation_config or GenerationConfig()
    for k, v in kwargs.items():
        if hasattr(gen, k):
            setattr(gen, k, v)
    assert input_ids.shape[0] == 1, "speculative decoding is per-sequence"
    eos = set(gen.eos_ids())
    do_sample = gen.do_sample
    temp = max(gen.temperature, 1e-6)

    def dist_of(logits):
        if do_sample:
            return (logits.float() / temp).softmax(-1)
        # greedy == sampling from a point mass
        p = torch.zeros_like(logits, dtype=torch.float32)
        p[..., logits.argmax(-1)] = 1.0
        return p

    device = input_ids.device
    all_ids = input_ids
    t_past = d_past = None
    t_len = d_len = 0  # positions already cached in each model
    n_accepted = n_drafted = 0
    new_tokens = []

    while len(new_tokens) < gen.max_new_tokens:
        # ------------------------------------------------- draft gamma tokens
        g = min(gamma, gen.max_new_tokens - len(new_tokens))
        draft_tokens = []
        draft_probs = []
        cur = all_ids[:, d_len:]
        for _ in range(g):
            logits, d_past = draft(input_ids=cur, use_cache=True,
                                   past_key_values=d_past)
            p = dist_of(logits[:, -1])
            tok = (torch.multinomial(p, 1) if do_sample
                   else logits[:, -1].argmax(-1, keepdim=True))
            draft_tokens.append(int(tok))
            draft_probs.append(p[0])
            cur = tok
        d_cached = all_ids.shape[1] + g - 1  # last drafted token never fed
        n_drafted += g

        # -------------------------------- verify with ONE target forward pass
        proposal = torch.tensor([draft_tokens], device=device)
        t_in = torch.cat([all_ids[:, t_len:], proposal], dim=1)
        t_logits, t_past = target(input_ids=t_in, use_cache=True,
                                  past_key_values=t_past)
        t_cached = t_len + t_in.shape[1]
        # t_logits[:, -g-1+i] predicts position after draft token i-1
        base = t_in.shape[1] - g - 1
        accepted = 0
        next_token = None
        for i in range(g):
            p = dist_of(t_logits[:, base + i])[0]
            q = draft_probs[i]
            x = draft_tokens[i]
            if do_sample:
                r = torch.rand((), device=device)
                ok = r < (p[x] / q[x].clamp(min=1e-20)).clamp(max=1.0)
            else:
                ok = int(p.argmax()) == x
            if ok:
                accepted += 1
                if x in eos:
                    next_token = None
                    break
            else:
                if do_sample:
                    resid = (p - q).clamp(min=0)
                    resid = resid / resid.sum().clamp(min=1e-20)
                    next_token = int(torch.multinomial(resid, 1))
                else:
                    next_token = int(p.argmax())
                break
        else:
            # all g accepted: bonus token from the target's last position
            p = dist_of(t_logits[:, -1])[0]
            next_token = (int(torch.multinomial(p, 1)) if do_sample
                          else int(p.argmax()))

        kept = draft_tokens[:accepted]
        new_tokens.extend(kept)
        n_accepted += accepted
        all_ids = torch.cat(
            [all_ids, torch.tensor([kept], device=device, dtype=torch.long)], dim=1)
        hit_eos = any(t in eos for t in kept)
        if next_token is not None and len(new_tokens) < gen.max_new_tokens \
                and not hit_eos:
            new_tokens.append(next_token)
            all_ids = torch.cat(
                [all_ids, torch.tensor([[next_token]], device=device)], dim=1)
            hit_eos = next_token in eos

        # roll both caches back to verified history (cache EXCLUDES at least
        # the last token so the next forward re-embeds it)
        hist = all_ids.shape[1] - 1
        t_len = min(t_cached, hist)
        d_len = min(d_cached, hist)
        t_past = _truncate_cache(t_past, t_len)
        d_past = _truncate_cache(d_past, d_len)
        if hit_eos:
            break

    stats = {"drafted": n_drafted, "accepted": n_accepted,
             "acceptance_rate": n_accepted / max(1, n_drafted)}
    out = torch.tensor([new_tokens], device=device, dtype=torch.long)
    return out, stats
