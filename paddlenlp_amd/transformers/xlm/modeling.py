"""XLM family (reference: paddlenlp/transformers/xlm/modeling.py).

Cross-lingual LM encoder: BERT-ish post-LN tower with per-token LANGUAGE
embeddings (n_langs > 1) added next to the positional ones, optional fixed
sinusoidal positions, and a vocab-projection LM head tied to the input
embeddings.
"""
from __future__ import annotations

import torch
import torch.nn as nn
import torch.nn.functional as F

from ..bert.configuration import BertConfig
from ..encoder import TransformerEncoder, init_encoder_weights
from ..model_utils import PretrainedModel
from ..pegasus.modeling import sinusoidal_positions

__all__ = ["XLMConfig", "XLMModel", "XLMWithLMHeadModel",
           "XLMForSequenceClassification"]


class XLMConfig(BertConfig):
    model_type = "xlm"

    def __init__(self, n_langs=1, use_lang_embeddings=True,
                 use_sinusoidal_embeddings=False, **kwargs):
        super().__init__(**kwargs)
        self.n_langs = n_langs
        self.use_lang_embeddings = use_lang_embeddings
        self.use_sinusoidal_embeddings = use_sinusoidal_embeddings


class XLMPretrainedModel(PretrainedModel):
    config_class = XLMConfig
    base_model_prefix = "xlm"

    def _init_weights(self, module):
        init_encoder_weights(module, self.config.initializer_range)


class XLMModel(XLMPretrainedModel):
    def __init__(self, config: XLMConfig):
        super().__init__(config)
        h = config.hidden_size
        self.word_embeddings = nn.Embedding(config.vocab_size, h,
                                            padding_idx=config.pad_token_id)
        if config.use_sinusoidal_embeddings:
            self.register_buffer("position_table", torch.empty(0),
                                 persistent=False)
            self.position_embeddings = None
        else:
            self.position_embeddings = nn.Embedding(
                config.max_position_embeddings, h)
        if config.n_langs > 1 and config.use_lang_embeddings:
            self.lang_embeddings = nn.Embedding(config.n_langs, h)
        else:
            self.lang_embeddings = None
        self.embed_norm = nn.LayerNorm(h, eps=config.layer_norm_eps)
        self.embed_dropout = nn.Dropout(config.hidden_dropout_prob)
        self.encoder = TransformerEncoder(config)

    def get_input_embeddings(self):
        return self.word_embeddings

    def _positions(self, S, device, dtype):
        if self.position_embeddings is not None:
            return self.position_embeddings(torch.arange(S, device=device))
        if self.position_table.numel() == 0 or self.position_table.device != device:
            self.position_table = sinusoidal_positions(
                self.config.max_position_embeddings,
                self.config.hidden_size).to(device=device, dtype=dtype)
        return self.position_table[:S]

    def forward(self, input_ids, langs=None, attention_mask=None):
        S = input_ids.shape[1]
        x = self.word_embeddings(input_ids)
        x = x + self._positions(S, input_ids.device, x.dtype)
        if langs is not None and self.lang_embeddings is not None:
            x = x + self.lang_embeddings(langs)
        x = self.embed_dropout(self.embed_norm(x))
        return self.encoder(x, attention_mask)


class XLMWithLMHeadModel(XLMPretrainedModel):
    _tied_weights_keys = ["lm_head.weight"]

    def __init__(self, config: XLMConfig):
        super().__init__(config)
        self.xlm = XLMModel(config)
        self.lm_head = nn.Linear(config.hidden_size, config.vocab_size)
        self.lm_head.weight = self.xlm.word_embeddings.weight

    def tie_weights(self):
        self.lm_head.weight = self.xlm.word_embeddings.weight

    def get_input_embeddings(self):
        return self.xlm.word_embeddings

    def forward(self, input_ids, langs=None, attention_mask=None, labels=None):
        seq = self.xlm(input_ids, langs, attention_mask)
        logits = self.lm_head(seq)
        if labels is not None:
            loss = F.cross_entropy(logits.view(-1, logits.shape[-1]),
                                   labels.view(-1), ignore_index=-100)
            return loss, logits
        return logits


class XLMForSequenceClassification(XLMPretrainedModel):
    def __init__(self, config: XLMConfig):
        super().__init__(config)
        self.xlm = XLMModel(config)
        self.dropout = nn.Dropout(config.hidden_dropout_prob)
        self.classifier = nn.Linear(config.hidden_size, config.num_labels)

    def forward(self, input_ids, langs=None, attention_mask=None, labels=None):
        seq = self.xlm(input_ids, langs, attention_mask)
        logits = self.classifier(self.dropout(seq[:, 0]))  # first-token pool
        if labels is not None:
            return F.cross_entropy(logits, labels.view(-1)), logits
        return logits
