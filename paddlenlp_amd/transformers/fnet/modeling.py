"""FNet family (reference: paddlenlp/transformers/fnet/modeling.py).

Attention-free encoder: each layer mixes tokens with the REAL part of a 2-D
FFT over (sequence, hidden) instead of self-attention, then the usual
post-LN FFN.  Parameter-free mixing means there is no padding mask — the
reference behaves the same way.  Embeddings carry an extra projection
(embedding hidden may differ from model hidden in checkpoints).
"""
from __future__ import annotations

import torch
import torch.nn as nn
import torch.nn.functional as F

from ..bert.configuration import BertConfig
from ..encoder import ACT2FN, EncoderPooler, LMPredictionHead, init_encoder_weights
from ..model_utils import PretrainedModel

__all__ = ["FNetConfig", "FNetModel", "FNetForSequenceClassification",
           "FNetForMaskedLM"]


class FNetConfig(BertConfig):
    model_type = "fnet"

    def __init__(self, pad_token_id=3, type_vocab_size=4, **kwargs):
        kwargs["type_vocab_size"] = type_vocab_size
        super().__init__(pad_token_id=pad_token_id, **kwargs)


class FNetLayer(nn.Module):
    def __init__(self, config: FNetConfig):
        super().__init__()
        h = config.hidden_size
        self.fourier_norm = nn.LayerNorm(h, eps=config.layer_norm_eps)
        self.fc_in = nn.Linear(h, config.intermediate_size)
        self.fc_out = nn.Linear(config.intermediate_size, h)
        self.mlp_norm = nn.LayerNorm(h, eps=config.layer_norm_eps)
        self.act = ACT2FN[config.hidden_act]
        self.dropout = nn.Dropout(config.hidden_dropout_prob)

    def forward(self, x):
        # token mixing: real part of FFT over hidden then sequence dims
        mixed = torch.fft.fftn(x.float(), dim=(1, 2)).real.to(x.dtype)
        x = self.fourier_norm(x + mixed)
        mlp = self.fc_out(self.act(self.fc_in(x)))
        return self.mlp_norm(x + self.dropout(mlp))


class FNetPretrainedModel(PretrainedModel):
    config_class = FNetConfig
    base_model_prefix = "fnet"

    def _init_weights(self, module):
        init_encoder_weights(module, self.config.initializer_range)


class FNetModel(FNetPretrainedModel):
    def __init__(self, config: FNetConfig):
        super().__init__(config)
        self.word_embeddings = nn.Embedding(config.vocab_size,
                                            config.hidden_size,
                                            padding_idx=config.pad_token_id)
        self.position_embeddings = nn.Embedding(config.max_position_embeddings,
                                                config.hidden_size)
        self.token_type_embeddings = nn.Embedding(config.type_vocab_size,
                                                  config.hidden_size)
        self.embed_norm = nn.LayerNorm(config.hidden_size,
                                       eps=config.layer_norm_eps)
        self.embed_proj = nn.Linear(config.hidden_size, config.hidden_size)
        self.embed_dropout = nn.Dropout(config.hidden_dropout_prob)
        self.layers = nn.ModuleList(
            [FNetLayer(config) for _ in range(config.num_hidden_layers)])
        self.pooler = EncoderPooler(config)

    def get_input_embeddings(self):
        return self.word_embeddings

    def forward(self, input_ids, token_type_ids=None):
        S = input_ids.shape[1]
        if token_type_ids is None:
            token_type_ids = torch.zeros_like(input_ids)
        pos = torch.arange(S, device=input_ids.device)
        x = (self.word_embeddings(input_ids)
             + self.position_embeddings(pos)
             + self.token_type_embeddings(token_type_ids))
        x = self.embed_dropout(self.embed_proj(self.embed_norm(x)))
        for layer in self.layers:
            x = layer(x)
        return x, self.pooler(x)


class FNetForSequenceClassification(FNetPretrainedModel):
    def __init__(self, config: FNetConfig):
        super().__init__(config)
        self.fnet = FNetModel(config)
        self.dropout = nn.Dropout(config.hidden_dropout_prob)
        self.classifier = nn.Linear(config.hidden_size, config.num_labels)

    def forward(self, input_ids, token_type_ids=None, labels=None):
        _, pooled = self.fnet(input_ids, token_type_ids)
        logits = self.classifier(self.dropout(pooled))
        if labels is not None:
            return F.cross_entropy(logits, labels.view(-1)), logits
        return logits


class FNetForMaskedLM(FNetPretrainedModel):
    def __init__(self, config: FNetConfig):
        super().__init__(config)
        self.fnet = FNetModel(config)
        self.lm_head = LMPredictionHead(
            config, embedding_weights=self.fnet.word_embeddings.weight)

    def forward(self, input_ids, token_type_ids=None, labels=None):
        seq, _ = self.fnet(input_ids, token_type_ids)
        logits = self.lm_head(seq)
        if labels is not None:
            loss = F.cross_entropy(logits.view(-1, logits.shape[-1]),
                                   labels.view(-1), ignore_index=-100)
            return loss, logits
        return logits
