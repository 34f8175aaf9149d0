"""BLIP model family (reference: paddlenlp/transformers/blip/modeling.py).

Vision-language pretraining trio over one ViT (reused from CLIP) and a
BERT-style text tower WITH cross-attention into the image features:
- ITC: contrastive image/text features (CLS-pooled, projected),
- ITM: image-text matching head over the cross-attended CLS state,
- captioning: causal text decoder with cross-attention (BlipForConditionalGeneration).
"""
from __future__ import annotations

import torch
import torch.nn as nn
import torch.nn.functional as F

from ..clip.modeling import CLIPVisionTransformer
from ..model_utils import PretrainedModel
from .configuration import BlipConfig, BlipTextConfig, BlipVisionConfig

__all__ = ["BlipModel", "BlipForImageTextRetrieval",
           "BlipForConditionalGeneration"]


class _BlipTextLayer(nn.Module):
    """Post-LN BERT layer + optional cross-attention into image states."""

    def __init__(self, c: BlipTextConfig, with_cross: bool):
        super().__init__()
        h = c.hidden_size
        self.num_heads = c.num_attention_heads
        self.head_dim = h // c.num_attention_heads
        self.self_qkv = nn.Linear(h, 3 * h)
        self.self_out = nn.Linear(h, h)
        self.self_norm = nn.LayerNorm(h, eps=c.layer_norm_eps)
        self.with_cross = with_cross
        if with_cross:
            self.cross_q = nn.Linear(h, h)
            self.cross_kv = nn.Linear(h, 2 * h)
            self.cross_out = nn.Linear(h, h)
            self.cross_norm = nn.LayerNorm(h, eps=c.layer_norm_eps)
        self.fc_in = nn.Linear(h, c.intermediate_size)
        self.fc_out = nn.Linear(c.intermediate_size, h)
        self.mlp_norm = nn.LayerNorm(h, eps=c.layer_norm_eps)

    def _heads(self, t, B, S):
        return t.view(B, S, self.num_heads, self.head_dim).transpose(1, 2)

    def forward(self, x, image_states=None, causal=False):
        B, S, H = x.shape
        q, k, v = self.self_qkv(x).chunk(3, dim=-1)
        attn = F.scaled_dot_product_attention(
            self._heads(q, B, S), self._heads(k, B, S), self._heads(v, B, S),
            is_causal=causal and S > 1)
        x = self.self_norm(x + self.self_out(
            attn.transpose(1, 2).reshape(B, S, H)))
        if self.with_cross and image_states is not None:
            Si = image_states.shape[1]
            cq = self._heads(self.cross_q(x), B, S)
            ck, cv = self.cross_kv(image_states).chunk(2, dim=-1)
            cross = F.scaled_dot_product_attention(
                cq, self._heads(ck, B, Si), self._heads(cv, B, Si))
            x = self.cross_norm(x + self.cross_out(
                cross.transpose(1, 2).reshape(B, S, H)))
        x = self.mlp_norm(x + self.fc_out(F.gelu(self.fc_in(x))))
        return x


class BlipTextEncoder(nn.Module):
    def __init__(self, c: BlipTextConfig, with_cross: bool):
        super().__init__()
        self.embeddings = nn.Embedding(c.vocab_size, c.hidden_size,
                                       padding_idx=c.pad_token_id)
        self.position_embeddings = nn.Embedding(
            c.max_position_embeddings, c.hidden_size)
        self.ln = nn.LayerNorm(c.hidden_size, eps=c.layer_norm_eps)
        self.layers = nn.ModuleList(
            [_BlipTextLayer(c, with_cross) for _ in range(c.num_hidden_layers)])

    def forward(self, input_ids, image_states=None, causal=False):
        pos = torch.arange(input_ids.shape[1], device=input_ids.device)
        x = self.ln(self.embeddings(input_ids) + self.position_embeddings(pos))
        for layer in self.layers:
            x = layer(x, image_states, causal)
        return x


class BlipPretrainedModel(PretrainedModel):
    config_class = BlipConfig
    base_model_prefix = "blip"

    def _init_weights(self, module):
        if isinstance(module, (nn.Linear, nn.Conv2d)):
            module.weight.data.normal_(mean=0.0, std=0.02)
            if module.bias is not None:
                module.bias.data.zero_()
        elif isinstance(module, nn.Embedding):
            module.weight.data.normal_(mean=0.0, std=0.02)
        elif isinstance(module, nn.LayerNorm):
            module.weight.data.fill_(1.0)
            module.bias.data.zero_()


class BlipModel(BlipPretrainedModel):
    """ITC dual towers (text tower WITHOUT cross-attention, CLIP-style)."""

    def __init__(self, config: BlipConfig):
        super().__init__(config)
        self.vision_model = CLIPVisionTransformer(config.vision_config)
        self.text_model = BlipTextEncoder(config.text_config, with_cross=False)
        self.visual_projection = nn.Linear(
            config.vision_config.hidden_size, config.projection_dim, bias=False)
        self.text_projection = nn.Linear(
            config.text_config.hidden_size, config.projection_dim, bias=False)
        self.logit_scale = nn.Parameter(
            torch.tensor(float(config.logit_scale_init_value)))

    def get_image_features(self, pixel_values):
        _, pooled = self.vision_model(pixel_values)
        return self.visual_projection(pooled)

    def get_text_features(self, input_ids):
        x = self.text_model(input_ids)
        return self.text_projection(x[:, 0])  # CLS pooling

    def forward(self, input_ids, pixel_values, return_loss: bool = False):
        tf = F.normalize(self.get_text_features(input_ids), dim=-1)
        vf = F.normalize(self.get_image_features(pixel_values), dim=-1)
        scale = self.logit_scale.exp()
        logits_per_text = scale * tf @ vf.t()
        if return_loss:
            labels = torch.arange(tf.shape[0], device=tf.device)
            loss = 0.5 * (F.cross_entropy(logits_per_text, labels)
                          + F.cross_entropy(logits_per_text.t(), labels))
            return loss, logits_per_text.t(), logits_per_text
        return logits_per_text.t(), logits_per_text


class BlipForImageTextRetrieval(BlipPretrainedModel):
    """ITM: binary match head over the cross-attended CLS state."""

    def __init__(self, config: BlipConfig):
        super().__init__(config)
        self.vision_model = CLIPVisionTransformer(config.vision_config)
        self.text_encoder = BlipTextEncoder(config.text_config, with_cross=True)
        self.itm_head = nn.Linear(config.text_config.hidden_size, 2)

    def forward(self, input_ids, pixel_values, labels=None):
        image_states, _ = self.vision_model(pixel_values)
        x = self.text_encoder(input_ids, image_states)
        logits = self.itm_head(x[:, 0])
        if labels is not None:
            loss = F.cross_entropy(logits, labels.view(-1))
            return loss, logits
        return logits


class BlipForConditionalGeneration(BlipPretrainedModel):
    """Image captioning: causal text decoder with cross-attention."""

    def __init__(self, config: BlipConfig):
        super().__init__(config)
        self.vision_model = CLIPVisionTransformer(config.vision_config)
        self.text_decoder = BlipTextEncoder(config.text_config, with_cross=True)
        self.lm_head = nn.Linear(config.text_config.hidden_size,
                                 config.text_config.vocab_size, bias=False)

    def forward(self, pixel_values, input_ids, labels=None):
        image_states, _ = self.vision_model(pixel_values)
        x = self.text_decoder(input_ids, image_states, causal=True)
        logits = self.lm_head(x)
        if labels is not None:
            loss = F.cross_entropy(
                logits.view(-1, logits.shape[-1]), labels.view(-1),
                ignore_index=-100)
            return loss, logits
        return logits

    @torch.no_grad()
    def generate(self, pixel_values, max_new_tokens: int = 16,
                 bos_token_id=None, eos_token_id=None):
        c = self.config.text_config
        bos = bos_token_id if bos_token_id is not None else c.bos_token_id
        eos = eos_token_id if eos_token_id is not None else c.eos_token_id
        B = pixel_values.shape[0]
        device = pixel_values.device
        image_states, _ = self.vision_model(pixel_values)
        ids = torch.full((B, 1), bos, dtype=torch.long, device=device)
        for _ in range(max_new_tokens):
            x = self.text_decoder(ids, image_states, causal=True)
            token = self.lm_head(x[:, -1]).argmax(-1, keepdim=True)
            ids = torch.cat([ids, token], dim=1)
            if (token == eos).all():
                break
        return ids[:, 1:]
