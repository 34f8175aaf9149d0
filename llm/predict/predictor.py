"""Inference predictor: dynamic-batching paged-KV serving loop.

Reference behavior: llm/predict/predictor.py — PredictorArgument :54,
DygraphPredictor :232 (vanilla model.generate), BlockInferencePredictorMixin
:727 (block tables, free list, dynamic insert), create_predictor :1163,
decode loop `while not_need_stop: _infer` :1003.

MI355X design: the block predictor drives the FusedMultiTransformer engine
(gfx950 paged decode kernels); sampling/penalties run as torch ops on-device;
the block manager is host-side.
"""
from __future__ import annotations

import os
import sys
from dataclasses import dataclass, field
from typing import List, Optional

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.dirname(os.path.abspath(__file__)))))

import torch

from paddlenlp_amd.experimental.block_manager import BlockManager
from paddlenlp_amd.experimental.fused_transformer import FusedMultiTransformer
from paddlenlp_amd.generation import GenerationConfig
from paddlenlp_amd.trainer import PdArgumentParser
from paddlenlp_amd.transformers import AutoConfig, AutoModelForCausalLM, AutoTokenizer
from paddlenlp_amd.utils.log import logger


@dataclass
class PredictorArgument:
    model_name_or_path: str = field(default=None)
    src_length: int = field(default=1024)
    max_length: int = field(default=1024)
    total_max_length: int = field(default=4096)
    batch_size: int = field(default=8)
    block_size: int = field(default=64)
    block_attn: bool = field(default=True)
    inference_model: bool = field(default=True)
    dtype: str = field(default="bfloat16")
    top_p: float = field(default=0.7)
    temperature: float = field(default=0.95)
    repetition_penalty: float = field(default=1.0)
    decode_strategy: str = field(default="sampling")  # or "greedy"
    mode: str = field(default="dynamic")
    benchmark: bool = field(default=False)
    quant_type: str = field(default="")  # "" | "fp8" | "weight_only_int8"
    use_hipgraph: bool = field(default=False)  # capture decode in a hipGraph
    cachekv_int8: bool = field(default=False)  # int8 paged KV (2x capacity)
    # speculative decoding: local path of a small draft model (dygraph mode)
    speculate_model: str = field(default="")
    speculate_gamma: int = field(default=4)


class BasePredictor:
    def __init__(self, config: PredictorArgument, tokenizer=None):
        self.config = config
        self.tokenizer = tokenizer

    def _apply_chat_template(self, item):
        """Accept raw strings or chat `messages` lists; message lists run
        through the tokenizer's chat template (reference predictor applies
        chat_template when the input is a conversation)."""
        if isinstance(item, str):
            return item
        return self.tokenizer.apply_chat_template(
            list(item), tokenize=False, add_generation_prompt=True)

    def _preprocess(self, texts: List[str]):
        texts = [self._apply_chat_template(t) for t in texts]
        out = self.tokenizer(list(texts), padding=True, return_tensors="pt",
                             truncation=True, max_length=self.config.src_length)
        return out


class DygraphPredictor(BasePredictor):
    """Vanilla model.generate path (reference DygraphPredictor :232);
    with `speculate_model` set, per-sequence speculative decoding (a small
    draft proposes speculate_gamma tokens, the target verifies in one
    forward — greedy output is identical to target-only decoding)."""

    def __init__(self, config, model=None, tokenizer=None, draft_model=None):
        super().__init__(config, tokenizer)
        self.model = model
        self.model.eval()
        self.draft_model = draft_model
        if draft_model is None and getattr(config, "speculate_model", ""):
            from paddlenlp_amd.transformers import AutoModelForCausalLM

            self.draft_model = AutoModelForCausalLM.from_pretrained(
                config.speculate_model)
        if self.draft_model is not None:
            self.draft_model = self.draft_model.to(
                next(self.model.parameters()).device).eval()

    @torch.no_grad()
    def predict(self, texts: List[str]) -> List[str]:
        inputs = self._preprocess(texts)
        device = next(self.model.parameters()).device
        ids = inputs["input_ids"].to(device)
        gen = GenerationConfig(
            max_new_tokens=self.config.max_length,
            do_sample=self.config.decode_strategy == "sampling",
            top_p=self.config.top_p,
            temperature=self.config.temperature,
            repetition_penalty=self.config.repetition_penalty,
            eos_token_id=self.tokenizer.eos_token_id,
            pad_token_id=self.tokenizer.pad_token_id or 0,
        )
        if self.draft_model is not None:
            from paddlenlp_amd.generation.speculative import (
                speculative_generate,
            )

            outs = []
            mask = inputs.get("attention_mask")
            for b in range(ids.shape[0]):
                row = ids[b:b + 1]
                if mask is not None:  # strip left padding per sequence
                    keep = mask[b].bool()
                    row = row[:, keep.to(row.device)]
                out, _ = speculative_generate(
                    self.model, self.draft_model, row, gen,
                    gamma=self.config.speculate_gamma)
                outs.append(out[0])
            return [self.tokenizer.decode(o, skip_special_tokens=True)
                    for o in outs]
        out, _ = self.model.generate(ids, gen)
        return self.tokenizer.batch_decode(out, skip_special_tokens=True)


class BlockInferencePredictor(BasePredictor):
    """Paged-KV dynamic-batching predictor over the fused engine."""

    def __init__(self, config: PredictorArgument, engine: FusedMultiTransformer,
                 tokenizer=None, device=None):
        super().__init__(config, tokenizer)
        self.engine = engine
        self.device = device or ("cuda:0" if torch.cuda.is_available() else "cpu")
        c = engine.config
        max_blocks_per_seq = (config.total_max_length + c.block_size - 1) // c.block_size
        # enough blocks for a full batch at total_max_length (288 GB HBM: be generous)
        num_blocks = max_blocks_per_seq * config.batch_size + 8
        self.manager = BlockManager(
            num_blocks, c.block_size, max_blocks_per_seq, config.batch_size)
        engine.allocate_caches(
            num_blocks, self.device,
            cachekv_dtype="int8" if config.cachekv_int8 else "bf16")
        self.engine = engine.to(self.device)
        if config.use_hipgraph and torch.cuda.is_available():
            from paddlenlp_amd.experimental.fused_transformer import GraphDecodeRunner

            self.decode_fn = GraphDecodeRunner(self.engine)
        else:
            self.decode_fn = self.engine.decode_step

    def _sample(self, logits, prev_ids=None, prev_lens=None):
        cfg = self.config
        # fused HIP path (reference token_penalty_multi_scores_v2 +
        # top_p_sampling_reject): penalties scatter + one histogram-threshold
        # sampling kernel instead of a 128k-vocab sort + multinomial
        if logits.is_cuda:
            try:
                from paddlenlp_amd.ops.functional import _load_extension

                C = _load_extension()
                B = logits.size(0)
                dev = logits.device
                logits = logits.contiguous().to(torch.bfloat16)
                if cfg.repetition_penalty != 1.0 and prev_ids is not None and prev_ids.numel():
                    if prev_lens is None:
                        prev_lens = torch.full((B,), prev_ids.size(1),
                                               dtype=torch.int32, device=dev)
                    C.apply_repetition_penalty(
                        logits, prev_ids.to(torch.int64),
                        prev_lens.to(torch.int32),
                        torch.full((B,), cfg.repetition_penalty,
                                   dtype=torch.float32, device=dev))
                greedy = cfg.decode_strategy == "greedy"
                temp = torch.full((B,), max(cfg.temperature, 1e-6),
                                  dtype=torch.float32, device=dev)
                top_p = torch.full((B,), 0.0 if greedy else min(cfg.top_p, 1.0),
                                   dtype=torch.float32, device=dev)
                u = torch.rand(B, dtype=torch.float32, device=dev)
                return C.topp_sample(logits, temp, top_p, u)
            except Exception:
                pass
        if cfg.repetition_penalty != 1.0 and prev_ids is not None and prev_ids.numel():
            score = torch.gather(logits, 1, prev_ids)
            score = torch.where(score < 0, score * cfg.repetition_penalty,
                                score / cfg.repetition_penalty)
            logits.scatter_(1, prev_ids, score)
        if cfg.decode_strategy == "greedy":
            return logits.argmax(-1)
        logits = logits / max(cfg.temperature, 1e-6)
        if cfg.top_p < 1.0:
            sorted_logits, sorted_idx = torch.sort(logits, descending=True)
            probs = sorted_logits.softmax(-1)
            cum = probs.cumsum(-1)
            remove = cum - probs > cfg.top_p
            mask = remove.scatter(1, sorted_idx, remove)
            logits = logits.masked_fill(mask, float("-inf"))
        return torch.multinomial(logits.softmax(-1), 1).squeeze(-1)

    @torch.no_grad()
    def predict(self, texts: List[str]) -> List[str]:
        """Dynamic batching: insert prompts while capacity allows, decode the
        active batch, refill slots as sequences finish."""
        cfg = self.config
        tok = self.tokenizer
        eos = tok.eos_token_id
        pending = list(enumerate(texts))
        results = {i: [] for i in range(len(texts))}
        slot_to_req = {}
        generated = {}
        mgr = self.manager
        prev_tokens = {}

        def prepare_inserts():
            """Tokenize + allocate slots for as many pending prompts as fit.
            Returns (inserted, batch, lens, bt) or None — no GPU compute."""
            inserted = []
            while pending:
                req_id, text = pending[0]
                text = self._apply_chat_template(text)
                ids = tok.encode(text)[-cfg.src_length:]
                slot = mgr.allocate_slot(len(ids))
                if slot is None:
                    break
                pending.pop(0)
                slot_to_req[slot] = req_id
                generated[slot] = 0
                prev_tokens[slot] = list(ids)
                inserted.append((slot, ids))
            if not inserted:
                return None
            T = max(len(ids) for _, ids in inserted)
            batch = torch.zeros(len(inserted), T, dtype=torch.long, device=self.device)
            lens = torch.zeros(len(inserted), dtype=torch.int32, device=self.device)
            bt = torch.empty(len(inserted), mgr.max_blocks_per_seq, dtype=torch.int32,
                             device=self.device)
            for i, (slot, ids) in enumerate(inserted):
                batch[i, :len(ids)] = torch.tensor(ids, device=self.device)
                lens[i] = len(ids)
                bt[i] = mgr.block_table[slot].to(self.device)
            return inserted, batch, lens, bt

        def finish_inserts(ins, logits):
            inserted, batch, lens, bt = ins
            T = batch.shape[1]
            prev = torch.tensor([prev_tokens[s] + [0] * (T - len(prev_tokens[s]))
                                 for s, _ in inserted], device=self.device, dtype=torch.long)
            plens = torch.tensor([len(prev_tokens[s]) for s, _ in inserted],
                                 dtype=torch.int32, device=self.device)
            tokens = self._sample(logits, prev, plens)
            for i, (slot, ids) in enumerate(inserted):
                self._commit_token(slot, int(tokens[i]), results, slot_to_req,
                                   generated, prev_tokens, eos)

        def _active_slots():
            return sorted(slot_to_req.keys())

        while slot_to_req or pending:
            ins = prepare_inserts()
            slots = _active_slots()
            # note: slots admitted by `ins` this round are in slot_to_req but
            # have no KV yet — exclude them from this round's decode batch
            if ins:
                new_slots = {s for s, _ in ins[0]}
                slots = [s for s in slots if s not in new_slots]
            if not slots:
                if ins:
                    logits = self.engine.prefill(ins[1], ins[3], ins[2])
                    finish_inserts(ins, logits)
                elif pending:
                    raise RuntimeError(
                        "cannot admit pending prompt: prompt larger than KV capacity")
                continue
            B = len(slots)
            input_ids = torch.tensor(
                [[prev_tokens[s][-1]] for s in slots], dtype=torch.long, device=self.device)
            lens_before = torch.tensor(
                [int(mgr.seq_lens[s]) for s in slots], dtype=torch.int32, device=self.device)
            # extend block tables for the incoming token
            for s in slots:
                if not mgr.extend(s, 1):
                    mgr.preempt_longest()  # (v1: preempted request is dropped back to pending)
            bt = torch.stack([mgr.block_table[s] for s in slots]).to(self.device, torch.int32)
            if ins:
                # admit concurrently: the prefill overlaps this decode step
                # on a side stream (engine.mixed_step) instead of stalling it
                logits, p_logits = self.engine.mixed_step(
                    input_ids, bt, lens_before, ins[1], ins[3], ins[2])
            else:
                logits = self.decode_fn(input_ids, bt, lens_before)
                p_logits = None
            maxlen = max(len(prev_tokens[s]) for s in slots)
            prev = torch.zeros(B, maxlen, dtype=torch.long, device=self.device)
            plens = torch.zeros(B, dtype=torch.int32, device=self.device)
            for i, s in enumerate(slots):
                prev[i, :len(prev_tokens[s])] = torch.tensor(
                    prev_tokens[s], device=self.device)
                plens[i] = len(prev_tokens[s])
            tokens = self._sample(logits, prev, plens)
            for i, s in enumerate(slots):
                self._commit_token(s, int(tokens[i]), results, slot_to_req,
                                   generated, prev_tokens, eos)
            if p_logits is not None:
                finish_inserts(ins, p_logits)

        return [tok.decode(results[i], skip_special_tokens=True) for i in range(len(texts))]

    def _commit_token(self, slot, token, results, slot_to_req, generated, prev_tokens, eos):
        req = slot_to_req[slot]
        generated[slot] += 1
        is_eos = (eos is not None and token == eos)
        if not is_eos:
            results[req].append(token)
            prev_tokens[slot].append(token)
        if is_eos or generated[slot] >= self.config.max_length:
            self.manager.release(slot)
            del slot_to_req[slot]
            del generated[slot]
            del prev_tokens[slot]
            return True
        return False


def create_predictor(predictor_args: PredictorArgument, model=None, tokenizer=None):
    """Factory (reference create_predictor :1163)."""
    if tokenizer is None:
        tokenizer = AutoTokenizer.from_pretrained(predictor_args.model_name_or_path)
    if model is None:
        model = AutoModelForCausalLM.from_pretrained(
            predictor_args.model_name_or_path,
            dtype=predictor_args.dtype,
        )
        if torch.cuda.is_available():
            model = model.to("cuda:0")
    if predictor_args.inference_model and predictor_args.block_attn:
        engine = FusedMultiTransformer.from_llama(
            model, block_size=predictor_args.block_size,
            max_seq_len=predictor_args.total_max_length)
        if predictor_args.quant_type:
            engine.quantize(predictor_args.quant_type)
        device = next(model.parameters()).device
        del model
        return BlockInferencePredictor(predictor_args, engine, tokenizer, device=device)
    return DygraphPredictor(predictor_args, model, tokenizer)


def predict():
    parser = PdArgumentParser((PredictorArgument,))
    (args,) = parser.parse_json_file_and_cmd_lines()
    predictor = create_predictor(args)
    import json as _json

    for line in sys.stdin:
        line = line.strip()
        if not line:
            continue
        outs = predictor.predict([line])
        print(_json.dumps({"src": line, "output": outs[0]}, ensure_ascii=False))


if __name__ == "__main__":
    predict()
