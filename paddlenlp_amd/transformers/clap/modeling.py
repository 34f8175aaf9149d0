"""CLAP (reference: paddlenlp/transformers/clap/modeling.py).

Contrastive Language-Audio Pretraining: an AUDIO tower over
mel-spectrogram patches (the reference uses an HTSAT Swin encoder; here
a patch-conv + transformer encoder with the same [B, mel, frames] →
pooled-embedding contract) and a RoBERTa-style text tower, both mapped
through 2-layer MLP projections (reference ClapProjectionLayer) into
the shared space; logits are temperature-scaled cosine similarities
with separate learned audio/text temperatures.
"""
from __future__ import annotations

import torch
import torch.nn as nn
import torch.nn.functional as F

from ..configuration_utils import PretrainedConfig
from ..encoder import EncoderLayer, init_encoder_weights
from ..model_utils import PretrainedModel

__all__ = ["ClapConfig", "ClapModel"]


class ClapAudioConfig(PretrainedConfig):
    model_type = "clap_audio"

    def __init__(self, num_mel_bins=64, max_frames=1024, patch_size=16,
                 hidden_size=512, num_hidden_layers=4,
                 num_attention_heads=8, intermediate_size=1024,
                 layer_norm_eps=1e-5, initializer_range=0.02, **kwargs):
        super().__init__(**kwargs)
        self.num_mel_bins = num_mel_bins
        self.max_frames = max_frames
        self.patch_size = patch_size
        self.hidden_size = hidden_size
        self.num_hidden_layers = num_hidden_layers
        self.num_attention_heads = num_attention_heads
        self.intermediate_size = intermediate_size
        self.hidden_act = "gelu"
        self.hidden_dropout_prob = 0.0
        self.attention_probs_dropout_prob = 0.0
        self.layer_norm_eps = layer_norm_eps
        self.initializer_range = initializer_range

    @property
    def head_dim(self):
        return self.hidden_size // self.num_attention_heads


class ClapTextConfig(PretrainedConfig):
    model_type = "clap_text"

    def __init__(self, vocab_size=50265, hidden_size=768,
                 num_hidden_layers=12, num_attention_heads=12,
                 intermediate_size=3072, max_position_embeddings=514,
                 layer_norm_eps=1e-12, initializer_range=0.02,
                 pad_token_id=1, **kwargs):
        super().__init__(**kwargs)
        self.vocab_size = vocab_size
        self.hidden_size = hidden_size
        self.num_hidden_layers = num_hidden_layers
        self.num_attention_heads = num_attention_heads
        self.intermediate_size = intermediate_size
        self.hidden_act = "gelu"
        self.hidden_dropout_prob = 0.0
        self.attention_probs_dropout_prob = 0.0
        self.max_position_embeddings = max_position_embeddings
        self.layer_norm_eps = layer_norm_eps
        self.initializer_range = initializer_range
        self.pad_token_id = pad_token_id

    @property
    def head_dim(self):
        return self.hidden_size // self.num_attention_heads


class ClapConfig(PretrainedConfig):
    model_type = "clap"

    def __init__(self, audio_config=None, text_config=None,
                 projection_dim=512, logit_scale_init_value=14.2857,
                 **kwargs):
        super().__init__(**kwargs)
        self.audio_config = ClapAudioConfig(**(audio_config or {}))
        self.text_config = ClapTextConfig(**(text_config or {}))
        self.projection_dim = projection_dim
        self.logit_scale_init_value = logit_scale_init_value
        self.initializer_range = self.text_config.initializer_range


class ClapProjectionLayer(nn.Module):
    """2-layer MLP projection (reference ClapProjectionLayer)."""

    def __init__(self, cin, cout):
        super().__init__()
        self.linear1 = nn.Linear(cin, cout)
        self.linear2 = nn.Linear(cout, cout)

    def forward(self, x):
        return self.linear2(F.relu(self.linear1(x)))


class ClapAudioEncoder(nn.Module):
    """Mel-spectrogram patch encoder: [B, mel, frames] -> pooled."""

    def __init__(self, c: ClapAudioConfig):
        super().__init__()
        self.patch_embed = nn.Conv2d(1, c.hidden_size,
                                     kernel_size=c.patch_size,
                                     stride=c.patch_size)
        n = (c.num_mel_bins // c.patch_size) * \
            (c.max_frames // c.patch_size)
        self.pos_embed = nn.Parameter(torch.zeros(1, n, c.hidden_size))
        self.layers = nn.ModuleList(
            [EncoderLayer(c) for _ in range(c.num_hidden_layers)])
        self.norm = nn.LayerNorm(c.hidden_size, eps=c.layer_norm_eps)

    def forward(self, input_features):       # [B, mel, frames]
        x = self.patch_embed(input_features[:, None])
        x = x.flatten(2).transpose(1, 2)     # [B, P, h]
        x = x + self.pos_embed[:, :x.shape[1]]
        for layer in self.layers:
            x = layer(x)
        return self.norm(x).mean(dim=1)      # mean-pool patches


class ClapPretrainedModel(PretrainedModel):
    config_class = ClapConfig
    base_model_prefix = "clap"

    def _init_weights(self, module):
        init_encoder_weights(module, self.config.initializer_range)


class ClapModel(ClapPretrainedModel):
    def __init__(self, config: ClapConfig):
        super().__init__(config)
        ac, tc = config.audio_config, config.text_config
        self.audio_model = ClapAudioEncoder(ac)
        self.text_embeddings = nn.Embedding(tc.vocab_size, tc.hidden_size,
                                            padding_idx=tc.pad_token_id)
        self.text_positions = nn.Embedding(tc.max_position_embeddings,
                                           tc.hidden_size)
        self.text_norm = nn.LayerNorm(tc.hidden_size, eps=tc.layer_norm_eps)
        self.text_layers = nn.ModuleList(
            [EncoderLayer(tc) for _ in range(tc.num_hidden_layers)])
        self.audio_projection = ClapProjectionLayer(ac.hidden_size,
                                                    config.projection_dim)
        self.text_projection = ClapProjectionLayer(tc.hidden_size,
                                                   config.projection_dim)
        # separate temperatures for a->t and t->a (reference logit_scale_a/t)
        self.logit_scale_a = nn.Parameter(torch.tensor(
            float(torch.log(torch.tensor(config.logit_scale_init_value)))))
        self.logit_scale_t = nn.Parameter(torch.tensor(
            float(torch.log(torch.tensor(config.logit_scale_init_value)))))
        self.init_weights()

    def get_input_embeddings(self):
        return self.text_embeddings

    def get_audio_features(self, input_features):
        return self.audio_projection(self.audio_model(input_features))

    def get_text_features(self, input_ids, attention_mask=None):
        S = input_ids.shape[1]
        pos = torch.arange(S, device=input_ids.device)
        x = self.text_norm(self.text_embeddings(input_ids) +
                           self.text_positions(pos))
        for layer in self.text_layers:
            x = layer(x, attention_mask)
        return self.text_projection(x[:, 0])

    def forward(self, input_ids, input_features, attention_mask=None):
        a = F.normalize(self.get_audio_features(input_features), dim=-1)
        t = F.normalize(self.get_text_features(input_ids, attention_mask),
                        dim=-1)
        logits_per_audio = self.logit_scale_a.exp() * a @ t.t()
        logits_per_text = self.logit_scale_t.exp() * t @ a.t()
        return logits_per_audio, logits_per_text
