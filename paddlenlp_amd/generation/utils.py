"""GenerationMixin: training-free decoding loops.

Reference behavior: paddlenlp/generation/utils.py — GenerationMixin :319,
generate :609 dispatching to greedy_search :1036, sample :1137, beam_search
:1496; KV-cached decode via the model's use_cache/past_key_values protocol.
"""
from __future__ import annotations

from typing import Optional, Tuple

import torch

from .configuration_utils import GenerationConfig
from .logits_process import (
    ForcedBOSTokenLogitsProcessor,
    ForcedEOSTokenLogitsProcessor,
    LogitsProcessorList,
    MinNewTokensLengthLogitsProcessor,
    NoBadWordsLogitsProcessor,
    NoRepeatNGramLogitsProcessor,
    PrefixConstrainedLogitsProcessor,
    RepetitionPenaltyLogitsProcessor,
    SequenceBiasLogitsProcessor,
    TemperatureLogitsWarper,
    TopKLogitsWarper,
    TopPLogitsWarper,
)


class GenerationMixin:
    """Mixed into CausalLM models.  The model forward must support
    (input_ids, use_cache=True, past_key_values=...) -> (logits, presents)."""

    generation_config: Optional[GenerationConfig] = None

    def prepare_inputs_for_generation(self, input_ids, past_key_values=None, **kwargs):
        if past_key_values is not None:
            input_ids = input_ids[:, -1:]
        return {"input_ids": input_ids, "past_key_values": past_key_values}

    def _get_logits_processors(self, gen_config: GenerationConfig, prompt_len: int):
        procs = LogitsProcessorList()
        if gen_config.repetition_penalty != 1.0:
            procs.append(RepetitionPenaltyLogitsProcessor(gen_config.repetition_penalty))
        if gen_config.min_new_tokens > 0 and gen_config.eos_ids():
            procs.append(MinNewTokensLengthLogitsProcessor(
                prompt_len, gen_config.min_new_tokens, gen_config.eos_ids()))
        if getattr(gen_config, "no_repeat_ngram_size", 0):
            procs.append(NoRepeatNGramLogitsProcessor(
                gen_config.no_repeat_ngram_size))
        if getattr(gen_config, "forced_bos_token_id", None) is not None:
            procs.append(ForcedBOSTokenLogitsProcessor(
                prompt_len, gen_config.forced_bos_token_id))
        if getattr(gen_config, "forced_eos_token_id", None) is not None \
                and gen_config.max_new_tokens:
            procs.append(ForcedEOSTokenLogitsProcessor(
                prompt_len + gen_config.max_new_tokens,
                gen_config.forced_eos_token_id))
        if getattr(gen_config, "bad_words_ids", None):
            procs.append(NoBadWordsLogitsProcessor(gen_config.bad_words_ids))
        if getattr(gen_config, "sequence_bias", None):
            procs.append(SequenceBiasLogitsProcessor(gen_config.sequence_bias))
        if getattr(gen_config, "prefix_allowed_tokens_fn", None) is not None:
            procs.append(PrefixConstrainedLogitsProcessor(
                gen_config.prefix_allowed_tokens_fn, gen_config.num_beams))
        return procs

    def _get_logits_warpers(self, gen_config: GenerationConfig):
        warpers = LogitsProcessorList()
        if gen_config.temperature != 1.0:
            warpers.append(TemperatureLogitsWarper(gen_config.temperature))
        if gen_config.top_k > 0:
            warpers.append(TopKLogitsWarper(gen_config.top_k))
        if gen_config.top_p < 1.0:
            warpers.append(TopPLogitsWarper(gen_config.top_p))
        return warpers

    @torch.no_grad()
    def generate(
        self,
        input_ids: torch.Tensor,
        generation_config: Optional[GenerationConfig] = None,
        **kwargs,
    ) -> Tuple[torch.Tensor, torch.Tensor]:
        """Returns (generated_ids [B, new], scores placeholder)."""
        gen = generation_config or self.generation_config or GenerationConfig.from_model_config(self.config)
        for k, v in kwargs.items():
            if hasattr(gen, k):
                setattr(gen, k, v)
        if gen.max_length:
            gen.max_new_tokens = max(1, gen.max_length - input_ids.shape[1])

        if gen.num_return_sequences > 1 and gen.num_beams == 1 \
                and not gen.do_sample:
            raise ValueError(
                "num_return_sequences > 1 needs do_sample or beam search")
        if gen.num_beams > 1 and gen.num_beam_groups > 1:
            return self.group_beam_search(input_ids, gen)
        if gen.num_beams > 1:
            return self.beam_search(input_ids, gen)
        if gen.do_sample:
            if gen.num_return_sequences > 1:
                # reference semantics: expand the batch n times, sample
                # independently, return [B*n, L]
                input_ids = input_ids.repeat_interleave(
                    gen.num_return_sequences, dim=0)
            return self.sample(input_ids, gen)
        return self.greedy_search(input_ids, gen)

    def _decode_loop(self, input_ids, gen: GenerationConfig, select_fn,
                     stopping_criteria=None):
        B = input_ids.shape[0]
        device = input_ids.device
        eos_ids = gen.eos_ids()
        procs = self._get_logits_processors(gen, input_ids.shape[1])
        unfinished = torch.ones(B, dtype=torch.bool, device=device)
        pad_id = gen.pad_token_id if gen.pad_token_id is not None else (eos_ids[0] if eos_ids else 0)

        past = None
        all_ids = input_ids
        new_tokens = []
        cur = input_ids
        for step in range(gen.max_new_tokens):
            if gen.use_cache:
                out = self(input_ids=cur, use_cache=True, past_key_values=past)
                logits, past = out
            else:
                logits = self(input_ids=all_ids)
            next_logits = logits[:, -1].float()
            next_logits = procs(all_ids, next_logits)
            token = select_fn(all_ids, next_logits)
            token = torch.where(unfinished, token, torch.full_like(token, pad_id))
            new_tokens.append(token)
            all_ids = torch.cat([all_ids, token[:, None]], dim=1)
            cur = token[:, None]
            if eos_ids:
                for eos in eos_ids:
                    unfinished = unfinished & (token != eos)
                if not unfinished.any():
                    break
            if stopping_criteria is not None and stopping_criteria(all_ids, None):
                break
        gen_ids = torch.stack(new_tokens, dim=1) if new_tokens else input_ids.new_zeros(B, 0)
        return gen_ids, None

    def greedy_search(self, input_ids, gen: GenerationConfig,
                      stopping_criteria=None):
        return self._decode_loop(input_ids, gen, lambda ids, lg: lg.argmax(-1),
                                 stopping_criteria)

    def sample(self, input_ids, gen: GenerationConfig):
        warpers = self._get_logits_warpers(gen)

        def select(ids, lg):
            lg = warpers(ids, lg)
            probs = lg.softmax(-1)
            return torch.multinomial(probs, 1).squeeze(-1)

        return self._decode_loop(input_ids, gen, select)

    def beam_search(self, input_ids, gen: GenerationConfig):
        """Standard length-penalized beam search (no cache reordering
        subtleties: the KV cache is reindexed per step)."""
        B, prompt_len = input_ids.shape
        K = gen.num_beams
        device = input_ids.device
        eos_ids = gen.eos_ids()
        procs = self._get_logits_processors(gen, prompt_len)

        # expand to beams
        ids = input_ids.repeat_interleave(K, dim=0)  # [B*K, L]
        beam_scores = torch.full((B, K), float("-inf"), device=device)
        beam_scores[:, 0] = 0.0
        beam_scores = beam_scores.view(-1)
        past = None
        cur = ids
        finished = [[] for _ in range(B)]  # (score, seq)

        for step in range(gen.max_new_tokens):
            out = self(input_ids=cur, use_cache=True, past_key_values=past)
            logits, past = out
            logp = logits[:, -1].float().log_softmax(-1)
            logp = procs(ids, logp)
            vocab = logp.shape[-1]
            scores = beam_scores[:, None] + logp  # [B*K, V]
            scores = scores.view(B, K * vocab)
            top_scores, top_idx = scores.topk(2 * K, dim=-1)
            beam_idx = top_idx // vocab           # [B, 2K]
            token_idx = top_idx % vocab

            new_ids, new_scores, new_beam_src = [], [], []
            for b in range(B):
                row_ids, row_scores, row_src = [], [], []
                for j in range(2 * K):
                    tok = token_idx[b, j].item()
                    src = b * K + beam_idx[b, j].item()
                    seq = torch.cat([ids[src], token_idx[b, j:j + 1]])
                    if eos_ids and tok in eos_ids:
                        lp = (seq.shape[0] - prompt_len) ** gen.length_penalty
                        finished[b].append((top_scores[b, j].item() / lp, seq))
                    elif len(row_ids) < K:
                        row_ids.append(seq)
                        row_scores.append(top_scores[b, j])
                        row_src.append(src)
                while len(row_ids) < K:  # degenerate: pad with best
                    row_ids.append(row_ids[-1])
                    row_scores.append(row_scores[-1])
                    row_src.append(row_src[-1])
                new_ids.extend(row_ids)
                new_scores.extend(row_scores)
                new_beam_src.extend(row_src)

            ids = torch.stack(new_ids)
            beam_scores = torch.stack(new_scores)
            src = torch.tensor(new_beam_src, device=device)
            past = _reorder_cache(past, src)
            cur = ids[:, -1:]
            if all(len(f) >= K for f in finished):
                break

        n_ret = max(1, min(gen.num_return_sequences, K))
        results = []
        for b in range(B):
            cands = sorted(finished[b], key=lambda x: -x[0])
            if len(cands) < n_ret:
                # top up with live beams (finished candidates stay preferred)
                live = []
                for j in range(K):
                    lp = max(1, ids.shape[1] - prompt_len) ** gen.length_penalty
                    live.append((beam_scores[b * K + j].item() / lp,
                                 ids[b * K + j]))
                cands.extend(sorted(live, key=lambda x: -x[0]))
            for r in range(n_ret):
                results.append(cands[r][1][prompt_len:])
        maxlen = max(r.shape[0] for r in results)
        pad_id = gen.pad_token_id if gen.pad_token_id is not None else (eos_ids[0] if eos_ids else 0)
        out = torch.full((B * n_ret, maxlen), pad_id, dtype=torch.long,
                         device=device)
        for i, r in enumerate(results):
            out[i, : r.shape[0]] = r
        return out, None


    def group_beam_search(self, input_ids, gen: GenerationConfig):
        """Diverse (group) beam search: beams split into num_beam_groups
        groups; later groups pay `diversity_penalty` per use of a token an
        earlier group already picked at the same step (reference
        group_beam_search / HammingDiversityLogitsProcessor)."""
        B, prompt_len = input_ids.shape
        K = gen.num_beams
        G = gen.num_beam_groups
        assert K % G == 0, f"num_beams {K} not divisible by num_beam_groups {G}"
        Kg = K // G
        device = input_ids.device
        eos_ids = gen.eos_ids()
        procs = self._get_logits_processors(gen, prompt_len)

        ids = input_ids.repeat_interleave(K, dim=0)   # beam layout: group-major
        beam_scores = torch.full((B, K), float("-inf"), device=device)
        beam_scores[:, [g * Kg for g in range(G)]] = 0.0
        beam_scores = beam_scores.view(-1)
        past = None
        cur = ids
        finished = [[[] for _ in range(G)] for _ in range(B)]

        for step in range(gen.max_new_tokens):
            out = self(input_ids=cur, use_cache=True, past_key_values=past)
            logits, past = out
            logp_all = logits[:, -1].float().log_softmax(-1)   # [B*K, V]
            logp_all = procs(ids, logp_all)
            vocab = logp_all.shape[-1]

            new_ids = [None] * (B * K)
            new_scores = torch.empty(B * K, device=device)
            new_src = [0] * (B * K)
            used = torch.zeros(B, vocab, device=device)  # earlier groups' picks
            for g in range(G):
                rows = torch.cat([
                    torch.arange(b * K + g * Kg, b * K + (g + 1) * Kg)
                    for b in range(B)]).to(device)
                logp = logp_all[rows].view(B, Kg, vocab)
                if g > 0 and gen.diversity_penalty > 0:
                    logp = logp - gen.diversity_penalty * used[:, None, :]
                scores = beam_scores[rows].view(B, Kg, 1) + logp
                scores = scores.view(B, Kg * vocab)
                top_scores, top_idx = scores.topk(2 * Kg, dim=-1)
                beam_idx = top_idx // vocab
                token_idx = top_idx % vocab
                for b in range(B):
                    n_live = 0
                    for j in range(2 * Kg):
                        tok = int(token_idx[b, j])
                        src = b * K + g * Kg + int(beam_idx[b, j])
                        seq = torch.cat([ids[src], token_idx[b, j:j + 1]])
                        if eos_ids and tok in eos_ids:
                            lp = (seq.shape[0] - prompt_len) ** gen.length_penalty
                            finished[b][g].append((float(top_scores[b, j]) / lp, seq))
                            used[b, tok] += 1.0
                        elif n_live < Kg:
                            slot = b * K + g * Kg + n_live
                            new_ids[slot] = seq
                            new_scores[slot] = top_scores[b, j]
                            new_src[slot] = src
                            used[b, tok] += 1.0
                            n_live += 1
                    while n_live < Kg:  # degenerate fill
                        slot = b * K + g * Kg + n_live
                        new_ids[slot] = new_ids[slot - 1]
                        new_scores[slot] = new_scores[slot - 1]
                        new_src[slot] = new_src[slot - 1]
                        n_live += 1

            ids = torch.stack(new_ids)
            beam_scores = new_scores
            past = _reorder_cache(past, torch.tensor(new_src, device=device))
            cur = ids[:, -1:]
            if all(len(fg) >= Kg for f in finished for fg in f):
                break

        results = []
        for b in range(B):
            cands = [c for fg in finished[b] for c in fg]
            if not cands:
                for j in range(K):
                    lp = (ids.shape[1] - prompt_len) ** gen.length_penalty
                    cands.append((float(beam_scores[b * K + j]) / lp, ids[b * K + j]))
            cands.sort(key=lambda x: -x[0])
            results.append(cands[0][1][prompt_len:])
        maxlen = max(r.shape[0] for r in results)
        pad_id = gen.pad_token_id if gen.pad_token_id is not None else (eos_ids[0] if eos_ids else 0)
        out = torch.full((B, maxlen), pad_id, dtype=torch.long, device=device)
        for b, r in enumerate(results):
            out[b, : r.shape[0]] = r
        return out, None


def _reorder_cache(past, beam_idx):
    if past is None:
        return None
    return [
        (k.index_select(0, beam_idx), v.index_select(0, beam_idx))
        for k, v in past
    ]
