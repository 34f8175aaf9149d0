"""CLIP model family (reference: paddlenlp/transformers/clip/modeling.py).

Dual-tower contrastive model: a ViT vision encoder (conv patch embedding +
class token + pre-LN transformer) and a causal text encoder pooled at the
EOS position, both projected into a shared space; logits are the
temperature-scaled cosine similarities.  Pre-LN layers here (CLIP uses
pre-LN, unlike the BERT-line post-LN core), attention through SDPA.
"""
from __future__ import annotations

import torch
import torch.nn as nn
import torch.nn.functional as F

from ..model_utils import PretrainedModel
from .configuration import CLIPConfig, CLIPTextConfig, CLIPVisionConfig

__all__ = ["CLIPModel", "CLIPTextModel", "CLIPVisionModel"]


class _PreLNLayer(nn.Module):
    def __init__(self, hidden, heads, intermediate, eps):
        super().__init__()
        self.ln_1 = nn.LayerNorm(hidden, eps=eps)
        self.num_heads = heads
        self.head_dim = hidden // heads
        self.qkv = nn.Linear(hidden, 3 * hidden)
        self.out = nn.Linear(hidden, hidden)
        self.ln_2 = nn.LayerNorm(hidden, eps=eps)
        self.fc1 = nn.Linear(hidden, intermediate)
        self.fc2 = nn.Linear(intermediate, hidden)

    def _attn(self, x, causal):
        B, S, H = x.shape
        q, k, v = self.qkv(x).chunk(3, dim=-1)
        shape = (B, S, self.num_heads, self.head_dim)
        q = q.view(shape).transpose(1, 2)
        k = k.view(shape).transpose(1, 2)
        v = v.view(shape).transpose(1, 2)
        o = F.scaled_dot_product_attention(q, k, v, is_causal=causal)
        return self.out(o.transpose(1, 2).reshape(B, S, H))

    def forward(self, x, causal=False):
        x = x + self._attn(self.ln_1(x), causal)
        x = x + self.fc2(F.gelu(self.fc1(self.ln_2(x)), approximate="tanh"))
        return x


class CLIPVisionTransformer(nn.Module):
    def __init__(self, c: CLIPVisionConfig):
        super().__init__()
        self.patch_embedding = nn.Conv2d(
            c.num_channels, c.hidden_size, kernel_size=c.patch_size,
            stride=c.patch_size, bias=False)
        n_patches = (c.image_size // c.patch_size) ** 2
        self.class_embedding = nn.Parameter(torch.zeros(c.hidden_size))
        self.position_embedding = nn.Embedding(n_patches + 1, c.hidden_size)
        self.pre_layernorm = nn.LayerNorm(c.hidden_size, eps=c.layer_norm_eps)
        self.layers = nn.ModuleList(
            [_PreLNLayer(c.hidden_size, c.num_attention_heads,
                         c.intermediate_size, c.layer_norm_eps)
             for _ in range(c.num_hidden_layers)])
        self.post_layernorm = nn.LayerNorm(c.hidden_size, eps=c.layer_norm_eps)

    def forward(self, pixel_values):  # [B, C, H, W]
        B = pixel_values.shape[0]
        patches = self.patch_embedding(pixel_values)         # [B, h, gh, gw]
        patches = patches.flatten(2).transpose(1, 2)         # [B, P, h]
        cls = self.class_embedding[None, None].expand(B, 1, -1)
        x = torch.cat([cls, patches], dim=1)
        pos = torch.arange(x.shape[1], device=x.device)
        x = self.pre_layernorm(x + self.position_embedding(pos))
        for layer in self.layers:
            x = layer(x, causal=False)
        pooled = self.post_layernorm(x[:, 0])                # class token
        return x, pooled


class CLIPTextTransformer(nn.Module):
    def __init__(self, c: CLIPTextConfig):
        super().__init__()
        self.token_embedding = nn.Embedding(c.vocab_size, c.hidden_size)
        self.position_embedding = nn.Embedding(
            c.max_position_embeddings, c.hidden_size)
        self.layers = nn.ModuleList(
            [_PreLNLayer(c.hidden_size, c.num_attention_heads,
                         c.intermediate_size, c.layer_norm_eps)
             for _ in range(c.num_hidden_layers)])
        self.final_layer_norm = nn.LayerNorm(c.hidden_size, eps=c.layer_norm_eps)
        self.eos_token_id = c.eos_token_id

    def forward(self, input_ids):
        pos = torch.arange(input_ids.shape[1], device=input_ids.device)
        x = self.token_embedding(input_ids) + self.position_embedding(pos)
        for layer in self.layers:
            x = layer(x, causal=True)  # CLIP text tower is causal
        x = self.final_layer_norm(x)
        # pool at the (first) EOS position; fall back to the last token
        is_eos = (input_ids == self.eos_token_id).int()
        has_eos = is_eos.any(-1)
        eos_pos = torch.where(has_eos, is_eos.argmax(-1),
                              torch.full_like(has_eos.int(), input_ids.shape[1] - 1).long())
        pooled = x[torch.arange(x.shape[0], device=x.device), eos_pos]
        return x, pooled


class CLIPPretrainedModel(PretrainedModel):
    config_class = CLIPConfig
    base_model_prefix = "clip"

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


class CLIPTextModel(CLIPPretrainedModel):
    config_class = CLIPTextConfig
    base_model_prefix = "clip_text"

    def __init__(self, config: CLIPTextConfig):
        super().__init__(config)
        self.text_model = CLIPTextTransformer(config)

    def forward(self, input_ids):
        return self.text_model(input_ids)


class CLIPVisionModel(CLIPPretrainedModel):
    config_class = CLIPVisionConfig
    base_model_prefix = "clip_vision"

    def __init__(self, config: CLIPVisionConfig):
        super().__init__(config)
        self.vision_model = CLIPVisionTransformer(config)

    def forward(self, pixel_values):
        return self.vision_model(pixel_values)


class CLIPModel(CLIPPretrainedModel):
    def __init__(self, config: CLIPConfig):
        super().__init__(config)
        self.text_model = CLIPTextTransformer(config.text_config)
        self.vision_model = CLIPVisionTransformer(config.vision_config)
        self.text_projection = nn.Linear(
            config.text_config.hidden_size, config.projection_dim, bias=False)
        self.visual_projection = nn.Linear(
            config.vision_config.hidden_size, config.projection_dim, bias=False)
        self.logit_scale = nn.Parameter(
            torch.tensor(float(config.logit_scale_init_value)))

    def get_text_features(self, input_ids):
        _, pooled = self.text_model(input_ids)
        return self.text_projection(pooled)

    def get_image_features(self, pixel_values):
        _, pooled = self.vision_model(pixel_values)
        return self.visual_projection(pooled)

    def forward(self, input_ids, pixel_values, return_loss: bool = False):
        tf = F.normalize(self.get_text_features(input_ids), dim=-1)
        vf = F.normalize(self.get_image_features(pixel_values), dim=-1)
        scale = self.logit_scale.exp()
        logits_per_text = scale * tf @ vf.t()
        logits_per_image = logits_per_text.t()
        if return_loss:
            labels = torch.arange(tf.shape[0], device=tf.device)
            loss = 0.5 * (F.cross_entropy(logits_per_text, labels)
                          + F.cross_entropy(logits_per_image, labels))
            return loss, logits_per_image, logits_per_text
        return logits_per_image, logits_per_text
