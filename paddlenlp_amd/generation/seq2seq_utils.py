"""Shared seq2seq decoding helpers (beam search for T5/BART-style models).

Reference behavior: paddlenlp/generation/utils.py beam_search applied to
encoder-decoder models — the encoder runs once, decoder self-attention
caches reorder by beam index each step, cross-attention caches ride along.
"""
from __future__ import annotations

import torch

from .configuration_utils import GenerationConfig


def _reorder_seq2seq_cache(past, beam_idx):
    """past: list per layer of (self_present, cross_present), each a (k, v)
    with batch on dim 0."""
    if past is None:
        return None
    out = []
    for self_p, cross_p in past:
        self_p = tuple(t.index_select(0, beam_idx) for t in self_p)
        cross_p = tuple(t.index_select(0, beam_idx) for t in cross_p) \
            if cross_p is not None else None
        out.append((self_p, cross_p))
    return out


@torch.no_grad()
def seq2seq_beam_search(model, input_ids: torch.Tensor, gen: GenerationConfig,
                        start_token_id: int, eos_token_id: int,
                        pad_token_id: int):
    """Length-penalized beam search over a model exposing
    `model.forward(decoder_input_ids=..., encoder_output=...,
    past_key_values=..., use_cache=True) -> (logits, past, enc)` and an
    encoder attribute reachable via model.<base>.encoder."""
    B = input_ids.shape[0]
    K = gen.num_beams
    device = input_ids.device
    base = getattr(model, model.base_model_prefix)
    enc = base.encoder(input_ids)
    # expand encoder states to beams
    enc = enc.repeat_interleave(K, dim=0)

    cur = torch.full((B * K, 1), start_token_id, dtype=torch.long, device=device)
    seqs = cur.clone()
    beam_scores = torch.full((B, K), float("-inf"), device=device)
    beam_scores[:, 0] = 0.0
    beam_scores = beam_scores.view(-1)
    past = None
    finished = [[] for _ in range(B)]

    for step in range(gen.max_new_tokens):
        logits, past, _ = model(decoder_input_ids=cur, encoder_output=enc,
                                past_key_values=past, use_cache=True)
        logp = logits[:, -1].float().log_softmax(-1)      # [B*K, V]
        vocab = logp.shape[-1]
        scores = (beam_scores[:, None] + logp).view(B, K * vocab)
        top_scores, top_idx = scores.topk(2 * K, dim=-1)
        beam_idx = top_idx // vocab
        token_idx = top_idx % vocab

        new_rows, new_scores, src_rows, new_tokens = [], [], [], []
        for b in range(B):
            live = 0
            for j in range(2 * K):
                tok = int(token_idx[b, j])
                src = b * K + int(beam_idx[b, j])
                seq = torch.cat([seqs[src], token_idx[b, j:j + 1]])
                if tok == eos_token_id:
                    lp = max(1, seq.shape[0] - 1) ** gen.length_penalty
                    finished[b].append((float(top_scores[b, j]) / lp, seq))
                elif live < K:
                    new_rows.append(seq)
                    new_scores.append(top_scores[b, j])
                    src_rows.append(src)
                    new_tokens.append(tok)
                    live += 1
            while live < K:
                new_rows.append(new_rows[-1])
                new_scores.append(new_scores[-1])
                src_rows.append(src_rows[-1])
                new_tokens.append(new_tokens[-1])
                live += 1
        seqs = torch.stack(new_rows)
        beam_scores = torch.stack(new_scores)
        sel = torch.tensor(src_rows, device=device)
        past = _reorder_seq2seq_cache(past, sel)
        cur = torch.tensor(new_tokens, device=device).view(-1, 1)
        if all(len(f) >= K for f in finished):
            break

    n_ret = max(1, min(gen.num_return_sequences, K))
    results = []
    for b in range(B):
        cands = sorted(finished[b], key=lambda x: -x[0])
        if len(cands) < n_ret:
            live = []
            for j in range(K):
                lp = max(1, seqs.shape[1] - 1) ** gen.length_penalty
                live.append((float(beam_scores[b * K + j]) / lp, seqs[b * K + j]))
            cands.extend(sorted(live, key=lambda x: -x[0]))
        for r in range(n_ret):
            results.append(cands[r][1][1:])  # drop the start token
    maxlen = max(r.shape[0] for r in results)
    out = torch.full((B * n_ret, maxlen), pad_token_id, dtype=torch.long,
                     device=device)
    for i, r in enumerate(results):
        out[i, :r.shape[0]] = r
    return out, None
