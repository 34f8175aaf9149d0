"""Generic task heads synthesized over any encoder/decoder base model.

Reference behavior: nearly every reference family ships
ForSequenceClassification / ForTokenClassification / ForQuestionAnswering /
ForMultipleChoice heads with the same ~20-line bodies.  Here
one factory builds them for any family whose base model returns either
(sequence_output, pooled_output) or just sequence_output, so every
registered family answers AutoModelFor* without per-family boilerplate.
Families with hand-written heads keep them (the registry prefers explicit
entries).
"""
from __future__ import annotations

import inspect
from typing import Optional

import torch
import torch.nn as nn
import torch.nn.functional as F

_CACHE = {}


def _call_base(base, kwargs):
    """Call the base model passing only the kwargs its forward accepts."""
    sig = getattr(base, "_fwd_params", None)
    if sig is None:
        sig = set(inspect.signature(type(base).forward).parameters)
        base._fwd_params = sig
    out = base(**{k: v for k, v in kwargs.items() if k in sig and v is not None})
    if isinstance(out, tuple):
        seq, pooled = out[0], (out[1] if len(out) > 1 else None)
    else:
        seq, pooled = out, None
    if pooled is None or pooled.dim() != 2:
        pooled = seq[:, 0]
    return seq, pooled


def _find_pretrained_base(base_cls):
    for klass in base_cls.__mro__[1:]:
        if getattr(klass, "base_model_prefix", None) is not None and \
                hasattr(klass, "from_pretrained"):
            return klass
    raise TypeError(f"{base_cls} has no PretrainedModel ancestor")


def synthesize_head(base_cls, kind: str):
    """Build (and cache) a task-head class over `base_cls`."""
    key = (base_cls, kind)
    if key in _CACHE:
        return _CACHE[key]
    pretrained = _find_pretrained_base(base_cls)
    prefix = getattr(pretrained, "base_model_prefix", "model") or "model"

    def _dropout(config):
        p = getattr(config, "classifier_dropout", None)
        if p is None:
            p = getattr(config, "hidden_dropout_prob", 0.1)
        return nn.Dropout(p)

    if kind == "sequence_classification":
        class Head(pretrained):
            def __init__(self, config):
                super().__init__(config)
                setattr(self, prefix, base_cls(config))
                self.dropout = _dropout(config)
                self.classifier = nn.Linear(
                    config.hidden_size, getattr(config, "num_labels", 2))

            def forward(self, input_ids, token_type_ids=None, position_ids=None,
                        attention_mask=None, labels=None):
                _, pooled = _call_base(getattr(self, prefix), dict(
                    input_ids=input_ids, token_type_ids=token_type_ids,
                    position_ids=position_ids, attention_mask=attention_mask))
                logits = self.classifier(self.dropout(pooled))
                if labels is not None:
                    if logits.shape[-1] == 1:
                        loss = F.mse_loss(logits.squeeze(-1), labels.float())
                    else:
                        loss = F.cross_entropy(logits, labels.view(-1))
                    return loss, logits
                return logits

    elif kind == "token_classification":
        class Head(pretrained):
            def __init__(self, config):
                super().__init__(config)
                setattr(self, prefix, base_cls(config))
                self.dropout = _dropout(config)
                self.classifier = nn.Linear(
                    config.hidden_size, getattr(config, "num_labels", 2))

            def forward(self, input_ids, token_type_ids=None, position_ids=None,
                        attention_mask=None, labels=None):
                seq, _ = _call_base(getattr(self, prefix), dict(
                    input_ids=input_ids, token_type_ids=token_type_ids,
                    position_ids=position_ids, attention_mask=attention_mask))
                logits = self.classifier(self.dropout(seq))
                if labels is not None:
                    loss = F.cross_entropy(logits.view(-1, logits.shape[-1]),
                                           labels.view(-1), ignore_index=-100)
                    return loss, logits
                return logits

    elif kind == "question_answering":
        class Head(pretrained):
            def __init__(self, config):
                super().__init__(config)
                setattr(self, prefix, base_cls(config))
                self.qa_outputs = nn.Linear(config.hidden_size, 2)

            def forward(self, input_ids, token_type_ids=None, position_ids=None,
                        attention_mask=None, start_positions=None,
                        end_positions=None):
                seq, _ = _call_base(getattr(self, prefix), dict(
                    input_ids=input_ids, token_type_ids=token_type_ids,
                    position_ids=position_ids, attention_mask=attention_mask))
                start, end = self.qa_outputs(seq).chunk(2, dim=-1)
                start, end = start.squeeze(-1), end.squeeze(-1)
                if start_positions is not None and end_positions is not None:
                    loss = (F.cross_entropy(start, start_positions.view(-1)) +
                            F.cross_entropy(end, end_positions.view(-1))) / 2
                    return loss, start, end
                return start, end

    elif kind == "multiple_choice":
        class Head(pretrained):
            def __init__(self, config):
                super().__init__(config)
                setattr(self, prefix, base_cls(config))
                self.dropout = _dropout(config)
                self.classifier = nn.Linear(config.hidden_size, 1)

            def forward(self, input_ids, token_type_ids=None, position_ids=None,
                        attention_mask=None, labels=None):
                # [B, num_choices, S] -> flatten, score each choice
                B, C, S = input_ids.shape
                flat = dict(
                    input_ids=input_ids.reshape(B * C, S),
                    token_type_ids=(token_type_ids.reshape(B * C, S)
                                    if token_type_ids is not None else None),
                    position_ids=(position_ids.reshape(B * C, S)
                                  if position_ids is not None else None),
                    attention_mask=(attention_mask.reshape(B * C, S)
                                    if attention_mask is not None else None))
                _, pooled = _call_base(getattr(self, prefix), flat)
                logits = self.classifier(self.dropout(pooled)).reshape(B, C)
                if labels is not None:
                    return F.cross_entropy(logits, labels.view(-1)), logits
                return logits

    else:
        raise ValueError(f"unknown task-head kind: {kind}")

    Head.__name__ = f"{base_cls.__name__.replace('Model', '')}For" + {
        "sequence_classification": "SequenceClassification",
        "token_classification": "TokenClassification",
        "question_answering": "QuestionAnswering",
        "multiple_choice": "MultipleChoice",
    }[kind]
    Head.__qualname__ = Head.__name__
    _CACHE[key] = Head
    return Head
