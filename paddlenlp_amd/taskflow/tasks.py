"""Encoder-backed Taskflow pipelines.

Reference behavior: paddlenlp/taskflow/{text_classification,ner,
information_extraction,fill_mask-style}.py — each pipeline loads a local
task model (no downloads in this environment), tokenizes, runs the encoder
head and post-processes into the reference's dict output shapes.
"""
from __future__ import annotations

from typing import Dict, List, Optional

import torch
import torch.nn.functional as F

from ..transformers import AutoTokenizer
from ..transformers.auto.modeling import (
    AutoModel,
    AutoModelForMaskedLM,
    AutoModelForSequenceClassification,
    AutoModelForTokenClassification,
)


class _EncoderTaskBase:
    auto_cls = AutoModel

    def __init__(self, model_path: str, **kwargs):
        self.tokenizer = AutoTokenizer.from_pretrained(model_path)
        self.model = self.auto_cls.from_pretrained(model_path)
        if torch.cuda.is_available():
            self.model = self.model.to("cuda:0")
        self.model.eval()
        self.device = next(self.model.parameters()).device

    def _encode(self, texts: List[str]):
        enc = self.tokenizer(texts, padding=True, return_tensors="pt")
        return {k: v.to(self.device) for k, v in enc.items()
                if k in ("input_ids", "attention_mask", "token_type_ids")}


class TextClassificationTask(_EncoderTaskBase):
    auto_cls = AutoModelForSequenceClassification

    def __init__(self, model_path: str, label_map: Optional[Dict[int, str]] = None,
                 **kwargs):
        super().__init__(model_path)
        id2label = getattr(self.model.config, "id2label", None)
        self.label_map = label_map or (
            {int(k): v for k, v in id2label.items()} if id2label else None)

    @torch.no_grad()
    def __call__(self, inputs):
        single = isinstance(inputs, str)
        texts = [inputs] if single else list(inputs)
        logits = self.model(**self._encode(texts))
        probs = logits.float().softmax(-1)
        results = []
        for t, p in zip(texts, probs):
            idx = int(p.argmax())
            label = self.label_map[idx] if self.label_map else str(idx)
            results.append({"text": t, "label": label, "score": float(p[idx])})
        return results[0] if single else results


class TokenClassificationTask(_EncoderTaskBase):
    """NER / pos_tagging: per-token labels aggregated into BIO entity spans."""

    auto_cls = AutoModelForTokenClassification

    def __init__(self, model_path: str, label_map: Optional[Dict[int, str]] = None,
                 **kwargs):
        super().__init__(model_path)
        id2label = getattr(self.model.config, "id2label", None)
        self.label_map = label_map or (
            {int(k): v for k, v in id2label.items()} if id2label else {})

    @torch.no_grad()
    def __call__(self, inputs):
        single = isinstance(inputs, str)
        texts = [inputs] if single else list(inputs)
        results = []
        for text in texts:
            encoding = self.tokenizer._tokenizer.encode(text)
            ids = torch.tensor([encoding.ids], device=self.device)
            logits = self.model(ids)
            pred = logits[0].argmax(-1).tolist()
            labels = [self.label_map.get(p, str(p)) for p in pred]
            entities = []
            cur = None
            for (tok, (s, e), lab) in zip(encoding.tokens, encoding.offsets, labels):
                if lab.startswith("B-"):
                    if cur:
                        entities.append(cur)
                    cur = {"entity": lab[2:], "start": s, "end": e,
                           "text": text[s:e]}
                elif lab.startswith("I-") and cur and cur["entity"] == lab[2:]:
                    cur["end"] = e
                    cur["text"] = text[cur["start"]:e]
                else:
                    if cur:
                        entities.append(cur)
                    cur = None
            if cur:
                entities.append(cur)
            results.append({"text": text, "entities": entities,
                            "labels": labels})
        return results[0] if single else results


class FillMaskTask(_EncoderTaskBase):
    auto_cls = AutoModelForMaskedLM

    def __init__(self, model_path: str, mask_token: str = "[MASK]",
                 top_k: int = 5, **kwargs):
        super().__init__(model_path)
        self.mask_token = mask_token
        self.mask_id = self.tokenizer._tokenizer.token_to_id(mask_token)
        assert self.mask_id is not None, f"{mask_token} not in vocab"
        self.top_k = top_k

    @torch.no_grad()
    def __call__(self, inputs):
        single = isinstance(inputs, str)
        texts = [inputs] if single else list(inputs)
        results = []
        for text in texts:
            ids = torch.tensor([self.tokenizer._tokenizer.encode(text).ids],
                               device=self.device)
            logits = self.model(ids)
            pos = (ids[0] == self.mask_id).nonzero(as_tuple=True)[0]
            preds = []
            for p in pos.tolist():
                probs = logits[0, p].float().softmax(-1)
                top = probs.topk(self.top_k)
                preds.append([
                    {"token": self.tokenizer._tokenizer.id_to_token(int(i)),
                     "score": float(s)}
                    for s, i in zip(top.values, top.indices)])
            results.append({"text": text, "predictions": preds})
        return results[0] if single else results


class FeatureExtractionTask(_EncoderTaskBase):
    """Pooled sentence embedding from the base encoder."""

    @torch.no_grad()
    def __call__(self, inputs):
        single = isinstance(inputs, str)
        texts = [inputs] if single else list(inputs)
        out = self.model(**self._encode(texts))
        pooled = out[1] if isinstance(out, tuple) else out.mean(dim=1)
        return pooled[0] if single else pooled


class TextSimilarityTask(FeatureExtractionTask):
    @torch.no_grad()
    def __call__(self, pairs):
        single = isinstance(pairs[0], str)
        batch = [pairs] if single else list(pairs)
        results = []
        for a, b in batch:
            va = super().__call__(a)
            vb = super().__call__(b)
            sim = float(F.cosine_similarity(va[None].float(), vb[None].float()))
            results.append({"text1": a, "text2": b, "similarity": sim})
        return results[0] if single else results


class InformationExtractionTask:
    """UIE pointer extraction (reference taskflow/information_extraction.py):
    schema prompts + start/end probability span decoding."""

    def __init__(self, model_path: str, schema: Optional[List[str]] = None,
                 position_prob: float = 0.5, **kwargs):
        from ..transformers import UIE

        self.tokenizer = AutoTokenizer.from_pretrained(model_path)
        self.model = UIE.from_pretrained(model_path)
        if torch.cuda.is_available():
            self.model = self.model.to("cuda:0")
        self.model.eval()
        self.device = next(self.model.parameters()).device
        self.schema = schema or []
        self.threshold = position_prob

    def set_schema(self, schema: List[str]):
        self.schema = schema if isinstance(schema, list) else [schema]

    @torch.no_grad()
    def __call__(self, inputs):
        single = isinstance(inputs, str)
        texts = [inputs] if single else list(inputs)
        results = []
        for text in texts:
            found: Dict[str, List[dict]] = {}
            for prompt in self.schema:
                tok = self.tokenizer._tokenizer
                p_enc = tok.encode(prompt)
                t_enc = tok.encode(text)
                # [prompt] [SEP-ish boundary via token_type] [text]
                ids = torch.tensor([p_enc.ids + t_enc.ids], device=self.device)
                type_ids = torch.tensor(
                    [[0] * len(p_enc.ids) + [1] * len(t_enc.ids)],
                    device=self.device)
                start_p, end_p = self.model(ids, token_type_ids=type_ids)
                off = len(p_enc.ids)
                sp = start_p[0, off:]
                ep = end_p[0, off:]
                spans = self._decode_spans(sp, ep, t_enc.offsets, text)
                if spans:
                    found[prompt] = spans
            results.append(found)
        return results[0] if single else results

    def _decode_spans(self, start_p, end_p, offsets, text):
        starts = (start_p > self.threshold).nonzero(as_tuple=True)[0].tolist()
        ends = (end_p > self.threshold).nonzero(as_tuple=True)[0].tolist()
        spans = []
        for s in starts:
            cand = [e for e in ends if e >= s]
            if not cand:
                continue
            e = cand[0]
            cs, ce = offsets[s][0], offsets[e][1]
            spans.append({
                "text": text[cs:ce], "start": cs, "end": ce,
                "probability": float(start_p[s] * end_p[e]),
            })
        return spans


class ZeroShotTextClassificationTask:
    """UTC zero-shot classification (reference taskflow
    zero_shot_text_classification): prompt = [CLS] [O-MASK] option_1 ...
    [O-MASK] option_n [SEP] text; the UTC head scores each [O-MASK]."""

    def __init__(self, model_path: str, schema: Optional[List[str]] = None,
                 omask_token: str = "[O-MASK]", single_label: bool = True,
                 pred_threshold: float = 0.5, **kwargs):
        from ..transformers import UTC

        self.tokenizer = AutoTokenizer.from_pretrained(model_path)
        self.model = UTC.from_pretrained(model_path)
        if torch.cuda.is_available():
            self.model = self.model.to("cuda:0")
        self.model.eval()
        self.device = next(self.model.parameters()).device
        self.schema = schema or []
        self.single_label = single_label
        self.pred_threshold = pred_threshold
        tok = self.tokenizer._tokenizer
        self.omask_id = tok.token_to_id(omask_token)
        self.cls_id = tok.token_to_id("[CLS]")
        self.sep_id = tok.token_to_id("[SEP]")
        assert self.omask_id is not None, f"{omask_token} not in vocab"

    def set_schema(self, schema: List[str]):
        self.schema = list(schema)

    @torch.no_grad()
    def __call__(self, inputs):
        single = isinstance(inputs, str)
        texts = [inputs] if single else list(inputs)
        tok = self.tokenizer._tokenizer
        results = []
        for text in texts:
            ids = [self.cls_id] if self.cls_id is not None else []
            omask_positions = []
            for option in self.schema:
                omask_positions.append(len(ids))
                ids.append(self.omask_id)
                ids.extend(tok.encode(option).ids)
            if self.sep_id is not None:
                ids.append(self.sep_id)
            ids.extend(tok.encode(text).ids)
            input_ids = torch.tensor([ids], device=self.device)
            om = torch.tensor([omask_positions], device=self.device)
            logits = self.model(input_ids, omask_positions=om)[0]
            if self.single_label:
                probs = logits.float().softmax(-1)
                idx = int(probs.argmax())
                preds = [{"label": self.schema[idx], "score": float(probs[idx])}]
            else:
                probs = torch.sigmoid(logits.float())
                preds = [{"label": o, "score": float(p)}
                         for o, p in zip(self.schema, probs)
                         if float(p) >= self.pred_threshold]
            results.append({"text_a": text, "predictions": preds})
        return results[0] if single else results


class TextCorrectionTask(_EncoderTaskBase):
    """Spelling/word correction via masked-LM rescoring (reference taskflow
    text_correction CSC pipeline): a token is flagged when the MLM strongly
    prefers a different token at its position."""

    auto_cls = AutoModelForMaskedLM

    def __init__(self, model_path: str, threshold: float = 0.9, **kwargs):
        super().__init__(model_path)
        self.threshold = threshold

    @torch.no_grad()
    def __call__(self, inputs):
        single = isinstance(inputs, str)
        texts = [inputs] if single else list(inputs)
        tok = self.tokenizer._tokenizer
        results = []
        for text in texts:
            encoding = tok.encode(text)
            ids = torch.tensor([encoding.ids], device=self.device)
            logits = self.model(ids)
            probs = logits[0].float().softmax(-1)
            corrections = []
            out_tokens = list(encoding.tokens)
            for i, tid in enumerate(encoding.ids):
                p_best, best = probs[i].max(-1)
                if int(best) != tid and float(p_best) >= self.threshold:
                    corrections.append({
                        "position": i,
                        "source": encoding.tokens[i],
                        "target": tok.id_to_token(int(best)),
                        "score": float(p_best),
                    })
                    out_tokens[i] = tok.id_to_token(int(best))
            results.append({"source": text, "target": " ".join(out_tokens),
                            "errors": corrections})
        return results[0] if single else results


class WordSegmentationTask(TokenClassificationTask):
    """Sequence-labeling word segmentation (reference taskflow
    word_segmentation): B/I labels aggregate characters/subtokens into
    words; returns the word list."""

    @torch.no_grad()
    def __call__(self, inputs):
        single = isinstance(inputs, str)
        texts = [inputs] if single else list(inputs)
        results = []
        for text in texts:
            encoding = self.tokenizer._tokenizer.encode(text)
            ids = torch.tensor([encoding.ids], device=self.device)
            logits = self.model(ids)
            pred = logits[0].argmax(-1).tolist()
            labels = [self.label_map.get(p, str(p)) for p in pred]
            words = []
            cur = None
            for (s, e), lab in zip(encoding.offsets, labels):
                if lab.startswith("B") or cur is None:
                    if cur is not None:
                        words.append(text[cur[0]:cur[1]])
                    cur = [s, e]
                else:  # I: extend the current word
                    cur[1] = e
            if cur is not None:
                words.append(text[cur[0]:cur[1]])
            results.append(words)
        return results[0] if single else results


class DependencyParsingTask:
    """DDParser-style biaffine dependency parsing (reference taskflow
    dependency_parsing.py): greedy head + relation decoding per token."""

    DEFAULT_RELS = ["ATT", "SBV", "VOB", "ADV", "CMP", "COO", "POB", "MT",
                    "HED", "IC", "DE", "DI", "DOB", "F", "DBL", "VV"]

    def __init__(self, model_path: str, rel_labels: Optional[List[str]] = None,
                 **kwargs):
        from .models import BiAffineParser

        self.tokenizer = AutoTokenizer.from_pretrained(model_path)
        self.rel_labels = rel_labels or self.DEFAULT_RELS
        self.model = BiAffineParser.from_pretrained(
            model_path, n_rels=len(self.rel_labels))
        if torch.cuda.is_available():
            self.model = self.model.to("cuda:0")
        self.model.eval()
        self.device = next(self.model.parameters()).device

    @torch.no_grad()
    def __call__(self, inputs):
        single = isinstance(inputs, str)
        texts = [inputs] if single else list(inputs)
        results = []
        for text in texts:
            encoding = self.tokenizer._tokenizer.encode(text)
            ids = torch.tensor([encoding.ids], device=self.device)
            heads, rels = self.model.decode(ids)
            results.append({
                "word": list(encoding.tokens),
                "head": [int(h) for h in heads[0]],
                "deprel": [self.rel_labels[int(r)] for r in rels[0]],
            })
        return results[0] if single else results


class DocumentIntelligenceTask:
    """ERNIE-Layout extractive doc QA (reference taskflow
    document_intelligence.py): input is pre-OCR'd {"doc_tokens", "doc_boxes",
    "prompt"} (no OCR engine offline); answer = argmax start/end span over
    the document segment."""

    def __init__(self, model_path: str, **kwargs):
        from ..transformers import ErnieLayoutForQuestionAnswering

        self.tokenizer = AutoTokenizer.from_pretrained(model_path)
        self.model = ErnieLayoutForQuestionAnswering.from_pretrained(model_path)
        if torch.cuda.is_available():
            self.model = self.model.to("cuda:0")
        self.model.eval()
        self.device = next(self.model.parameters()).device

    @torch.no_grad()
    def __call__(self, inputs):
        single = isinstance(inputs, dict)
        batch = [inputs] if single else list(inputs)
        tok = self.tokenizer._tokenizer
        results = []
        for ex in batch:
            prompt_ids = tok.encode(ex["prompt"]).ids
            doc_tokens = list(ex["doc_tokens"])
            doc_boxes = list(ex["doc_boxes"])
            doc_ids = []
            tok_spans = []  # subtoken -> doc token index
            for ti, word in enumerate(doc_tokens):
                sub = tok.encode(word).ids
                doc_ids.extend(sub)
                tok_spans.extend([ti] * len(sub))
            ids = torch.tensor([prompt_ids + doc_ids], device=self.device)
            zero_box = [0, 0, 0, 0]
            boxes = [zero_box] * len(prompt_ids) + \
                [list(doc_boxes[t]) for t in tok_spans]
            bbox = torch.tensor([boxes], device=self.device)
            start_logits, end_logits = self.model(ids, bbox)
            off = len(prompt_ids)
            s = int(start_logits[0, off:].argmax())
            e_cands = end_logits[0, off + s:]
            e = s + int(e_cands.argmax())
            ans_tokens = sorted(set(tok_spans[s:e + 1]))
            results.append({
                "prompt": ex["prompt"],
                "result": [{
                    "value": " ".join(doc_tokens[t] for t in ans_tokens),
                    "start": s, "end": e,
                }],
            })
        return results[0] if single else results


class KnowledgeMiningTask(TokenClassificationTask):
    """WordTag-style knowledge mining (reference taskflow
    knowledge_mining.py): token-classification tags per word plus optional
    term linking against a USER-SUPPLIED term dictionary
    ({surface_form: termid}) — the reference's TermTree KB is a downloadable
    artifact that does not exist offline, so linking is bring-your-own."""

    def __init__(self, model_path: str, label_map=None, term_dict=None,
                 **kwargs):
        super().__init__(model_path, label_map=label_map)
        self.term_dict = term_dict or {}

    @torch.no_grad()
    def __call__(self, inputs):
        single = isinstance(inputs, str)
        texts = [inputs] if single else list(inputs)
        results = []
        for text in texts:
            base = super().__call__(text)
            items = []
            for ent in base["entities"]:
                item = {"item": ent["text"], "offset": ent["start"],
                        "wordtag_label": ent["entity"]}
                termid = self.term_dict.get(ent["text"])
                if termid is not None:
                    item["termid"] = termid
                items.append(item)
            results.append({"text": text, "items": items})
        return results[0] if single else results
