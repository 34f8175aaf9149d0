"""MegatronBERT family (reference: paddlenlp/transformers/megatronbert/).

BERT re-trained with Megatron-LM's PRE-LN residual order: each sublayer
computes x + f(LN(x)) (the reference's MegatronBertAttention:212 applies
layer_norm before attention), with one extra LayerNorm after the stack.
Built on the pegasus pre-LN encoder layer (identical math) so the tower
stays on the shared flash/SDPA seam.
"""
from __future__ import annotations

import torch
import torch.nn as nn
import torch.nn.functional as F

from ..bert.configuration import BertConfig
from ..encoder import EncoderEmbeddings, EncoderPooler, init_encoder_weights
from ..model_utils import PretrainedModel
from ..pegasus.modeling import _PegasusEncoderLayer

__all__ = ["MegatronBertConfig", "MegatronBertModel",
           "MegatronBertForSequenceClassification",
           "MegatronBertForQuestionAnswering"]


class MegatronBertConfig(BertConfig):
    model_type = "megatronbert"

    def __init__(self, vocab_size=29056, hidden_size=1024,
                 num_hidden_layers=24, num_attention_heads=16,
                 intermediate_size=4096, **kwargs):
        super().__init__(vocab_size=vocab_size, hidden_size=hidden_size,
                         num_hidden_layers=num_hidden_layers,
                         num_attention_heads=num_attention_heads,
                         intermediate_size=intermediate_size, **kwargs)

    # pre-LN layer classes read seq2seq-style names
    @property
    def d_model(self):
        return self.hidden_size

    @property
    def encoder_attention_heads(self):
        return self.num_attention_heads

    @property
    def encoder_ffn_dim(self):
        return self.intermediate_size

    @property
    def activation_function(self):
        return self.hidden_act


class MegatronBertPretrainedModel(PretrainedModel):
    config_class = MegatronBertConfig
    base_model_prefix = "megatronbert"

    def _init_weights(self, module):
        init_encoder_weights(module, self.config.initializer_range)


class MegatronBertModel(MegatronBertPretrainedModel):
    def __init__(self, config: MegatronBertConfig):
        super().__init__(config)
        self.embeddings = EncoderEmbeddings(config)
        self.layers = nn.ModuleList(
            [_PegasusEncoderLayer(config)
             for _ in range(config.num_hidden_layers)])
        self.ln = nn.LayerNorm(config.hidden_size, eps=config.layer_norm_eps)
        self.pooler = EncoderPooler(config)

    def get_input_embeddings(self):
        return self.embeddings.word_embeddings

    def forward(self, input_ids, token_type_ids=None, attention_mask=None):
        x = self.embeddings(input_ids, token_type_ids)
        for layer in self.layers:
            x = layer(x)
        x = self.ln(x)
        return x, self.pooler(x)


class MegatronBertForSequenceClassification(MegatronBertPretrainedModel):
    def __init__(self, config: MegatronBertConfig):
        super().__init__(config)
        self.megatronbert = MegatronBertModel(config)
        self.dropout = nn.Dropout(config.hidden_dropout_prob)
        self.classifier = nn.Linear(config.hidden_size, config.num_labels)

    def forward(self, input_ids, token_type_ids=None, attention_mask=None,
                labels=None):
        _, pooled = self.megatronbert(input_ids, token_type_ids,
                                      attention_mask)
        logits = self.classifier(self.dropout(pooled))
        if labels is not None:
            return F.cross_entropy(logits, labels.view(-1)), logits
        return logits


class MegatronBertForQuestionAnswering(MegatronBertPretrainedModel):
    def __init__(self, config: MegatronBertConfig):
        super().__init__(config)
        self.megatronbert = MegatronBertModel(config)
        self.classifier = nn.Linear(config.hidden_size, 2)

    def forward(self, input_ids, token_type_ids=None, attention_mask=None):
        seq, _ = self.megatronbert(input_ids, token_type_ids, attention_mask)
        start, end = self.classifier(seq).chunk(2, dim=-1)
        return start.squeeze(-1), end.squeeze(-1)
