"""LayoutLMv2 (reference: paddlenlp/transformers/layoutlmv2/modeling.py).

Document understanding with THREE modalities: text + 2-D layout
(four-corner + width/height bbox embeddings, reference
_cal_spatial_position_embeddings :111-134), a VISUAL stream of image
patches appended to the text sequence (the reference pulls a
ResNet-FPN backbone from layoutxlm; here a conv patch embedder fills
the same [B, grid*grid, H] contract), and RELATIVE attention biases —
1-D position buckets and 2-D spatial buckets projected per head
(:371-377).  LayoutXLM is the multilingual same-architecture variant.
"""
from __future__ import annotations

import torch
import torch.nn as nn
import torch.nn.functional as F

from ..configuration_utils import PretrainedConfig
from ..encoder import ACT2FN, init_encoder_weights
from ..model_utils import PretrainedModel

__all__ = ["LayoutLMv2Config", "LayoutLMv2Model",
           "LayoutLMv2ForTokenClassification"]


class LayoutLMv2Config(PretrainedConfig):
    model_type = "layoutlmv2"

    def __init__(self, vocab_size=30522, hidden_size=768,
                 num_hidden_layers=12, num_attention_heads=12,
                 intermediate_size=3072, hidden_act="gelu",
                 hidden_dropout_prob=0.1, max_position_embeddings=512,
                 max_2d_position_embeddings=1024, coordinate_size=128,
                 shape_size=128, image_feature_pool_shape=(7, 7, 256),
                 has_relative_attention_bias=True, rel_pos_bins=32,
                 max_rel_pos=128, has_spatial_attention_bias=True,
                 rel_2d_pos_bins=64, max_rel_2d_pos=256,
                 type_vocab_size=2, initializer_range=0.02,
                 layer_norm_eps=1e-12, pad_token_id=0, num_labels=2,
                 **kwargs):
        super().__init__(**kwargs)
        self.vocab_size = vocab_size
        self.hidden_size = hidden_size
        self.num_hidden_layers = num_hidden_layers
        self.num_attention_heads = num_attention_heads
        self.intermediate_size = intermediate_size
        self.hidden_act = hidden_act
        self.hidden_dropout_prob = hidden_dropout_prob
        self.max_position_embeddings = max_position_embeddings
        self.max_2d_position_embeddings = max_2d_position_embeddings
        self.coordinate_size = coordinate_size
        self.shape_size = shape_size
        self.image_feature_pool_shape = list(image_feature_pool_shape)
        self.has_relative_attention_bias = has_relative_attention_bias
        self.rel_pos_bins = rel_pos_bins
        self.max_rel_pos = max_rel_pos
        self.has_spatial_attention_bias = has_spatial_attention_bias
        self.rel_2d_pos_bins = rel_2d_pos_bins
        self.max_rel_2d_pos = max_rel_2d_pos
        self.type_vocab_size = type_vocab_size
        self.initializer_range = initializer_range
        self.layer_norm_eps = layer_norm_eps
        self.pad_token_id = pad_token_id
        self.num_labels = num_labels

    @property
    def head_dim(self):
        return self.hidden_size // self.num_attention_heads


def relative_position_bucket(rel, num_buckets, max_distance):
    """Bidirectional T5-style buckets (reference relative_position_bucket)."""
    ret = (rel > 0).long() * (num_buckets // 2)
    n = rel.abs()
    half = num_buckets // 2
    exact = half // 2
    is_small = n < exact
    large = (exact + (
        torch.log(n.float() / exact + 1e-6)
        / torch.log(torch.tensor(max_distance / exact)) * (half - exact)
    ).long()).clamp(min=0, max=half - 1)
    return ret + torch.where(is_small, n, large)


class _BiasedLayer(nn.Module):
    def __init__(self, config: LayoutLMv2Config):
        super().__init__()
        h = config.hidden_size
        self.nh, self.dh = config.num_attention_heads, config.head_dim
        self.qkv = nn.Linear(h, 3 * h)
        self.out = nn.Linear(h, h)
        self.attn_norm = nn.LayerNorm(h, eps=config.layer_norm_eps)
        self.fc_in = nn.Linear(h, config.intermediate_size)
        self.fc_out = nn.Linear(config.intermediate_size, h)
        self.mlp_norm = nn.LayerNorm(h, eps=config.layer_norm_eps)
        self.act = ACT2FN[config.hidden_act]

    def forward(self, x, bias):
        B, S, H = x.shape
        q, k, v = self.qkv(x).chunk(3, dim=-1)
        shp = (B, S, self.nh, self.dh)
        o = F.scaled_dot_product_attention(
            q.view(shp).transpose(1, 2), k.view(shp).transpose(1, 2),
            v.view(shp).transpose(1, 2), attn_mask=bias)
        x = self.attn_norm(x + self.out(o.transpose(1, 2).reshape(B, S, H)))
        return self.mlp_norm(x + self.fc_out(self.act(self.fc_in(x))))


class LayoutLMv2PretrainedModel(PretrainedModel):
    config_class = LayoutLMv2Config
    base_model_prefix = "layoutlmv2"

    def _init_weights(self, module):
        init_encoder_weights(module, self.config.initializer_range)


class LayoutLMv2Model(LayoutLMv2PretrainedModel):
    def __init__(self, config: LayoutLMv2Config):
        super().__init__(config)
        h = config.hidden_size
        self.embeddings = nn.Embedding(config.vocab_size, h,
                                       padding_idx=config.pad_token_id)
        self.position_embeddings = nn.Embedding(
            config.max_position_embeddings, h)
        self.token_type_embeddings = nn.Embedding(config.type_vocab_size, h)
        c, s = config.coordinate_size, config.shape_size
        self.x_embeddings = nn.Embedding(
            config.max_2d_position_embeddings, c)
        self.y_embeddings = nn.Embedding(
            config.max_2d_position_embeddings, c)
        self.h_embeddings = nn.Embedding(
            config.max_2d_position_embeddings, s)
        self.w_embeddings = nn.Embedding(
            config.max_2d_position_embeddings, s)
        self.spatial_proj = nn.Linear(4 * c + 2 * s, h)
        # visual stream: conv patch embedder standing in for the
        # reference's ResNet-FPN pooled features (same output contract)
        g = config.image_feature_pool_shape[0]
        self.visual_proj = nn.Conv2d(3, h, kernel_size=1)
        self.visual_pool = nn.AdaptiveAvgPool2d((g, g))
        self.visual_norm = nn.LayerNorm(h, eps=config.layer_norm_eps)
        self.embed_norm = nn.LayerNorm(h, eps=config.layer_norm_eps)
        self.layers = nn.ModuleList(
            [_BiasedLayer(config)
             for _ in range(config.num_hidden_layers)])
        if config.has_relative_attention_bias:
            self.rel_pos_bias = nn.Linear(config.rel_pos_bins,
                                          config.num_attention_heads,
                                          bias=False)
        if config.has_spatial_attention_bias:
            self.rel_pos_x_bias = nn.Linear(config.rel_2d_pos_bins,
                                            config.num_attention_heads,
                                            bias=False)
            self.rel_pos_y_bias = nn.Linear(config.rel_2d_pos_bins,
                                            config.num_attention_heads,
                                            bias=False)
        self.init_weights()

    def get_input_embeddings(self):
        return self.embeddings

    def _spatial(self, bbox):
        bbox = bbox.clamp(0, self.config.max_2d_position_embeddings - 1)
        x0 = self.x_embeddings(bbox[:, :, 0])
        y0 = self.y_embeddings(bbox[:, :, 1])
        x1 = self.x_embeddings(bbox[:, :, 2])
        y1 = self.y_embeddings(bbox[:, :, 3])
        w = self.w_embeddings((bbox[:, :, 2] - bbox[:, :, 0]).clamp(min=0))
        hh = self.h_embeddings((bbox[:, :, 3] - bbox[:, :, 1]).clamp(min=0))
        return self.spatial_proj(torch.cat([x0, y0, x1, y1, w, hh], dim=-1))

    def _rel_bias(self, pos, bbox):
        cfg = self.config
        bias = None
        if cfg.has_relative_attention_bias:
            rel = pos[:, None, :] - pos[:, :, None]
            b = relative_position_bucket(rel, cfg.rel_pos_bins,
                                         cfg.max_rel_pos)
            onehot = F.one_hot(b, cfg.rel_pos_bins).float()
            bias = self.rel_pos_bias(onehot).permute(0, 3, 1, 2)
        if cfg.has_spatial_attention_bias:
            cx = bbox[:, :, 0]
            cy = bbox[:, :, 3]
            for coord, proj in ((cx, self.rel_pos_x_bias),
                                (cy, self.rel_pos_y_bias)):
                rel = coord[:, None, :] - coord[:, :, None]
                b = relative_position_bucket(rel, cfg.rel_2d_pos_bins,
                                             cfg.max_rel_2d_pos)
                onehot = F.one_hot(b, cfg.rel_2d_pos_bins).float()
                extra = proj(onehot).permute(0, 3, 1, 2)
                bias = extra if bias is None else bias + extra
        return bias

    def forward(self, input_ids, bbox=None, image=None,
                token_type_ids=None):
        B, S = input_ids.shape
        device = input_ids.device
        if bbox is None:
            bbox = torch.zeros(B, S, 4, dtype=torch.long, device=device)
        pos = torch.arange(S, device=device)
        x = self.embeddings(input_ids) + self.position_embeddings(pos) \
            + self._spatial(bbox)
        if token_type_ids is not None:
            x = x + self.token_type_embeddings(token_type_ids)
        x = self.embed_norm(x)

        g = self.config.image_feature_pool_shape[0]
        n_vis = g * g
        if image is not None:
            vis = self.visual_pool(self.visual_proj(image))
            vis = self.visual_norm(vis.flatten(2).transpose(1, 2))
        else:
            vis = x.new_zeros(B, n_vis, x.shape[-1])
        # visual tokens carry grid bboxes on the same 0..1023 canvas
        gy, gx = torch.meshgrid(torch.arange(g, device=device),
                                torch.arange(g, device=device),
                                indexing="ij")
        step = self.config.max_2d_position_embeddings // g
        vis_bbox = torch.stack(
            [gx * step, gy * step,
             ((gx + 1) * step).clamp(max=self.config.max_2d_position_embeddings - 1),
             ((gy + 1) * step).clamp(max=self.config.max_2d_position_embeddings - 1)],
            dim=-1).reshape(1, n_vis, 4).expand(B, n_vis, 4)
        full = torch.cat([x, vis], dim=1)
        full_pos = torch.cat(
            [pos.expand(B, S),
             torch.arange(n_vis, device=device).expand(B, n_vis)], dim=1)
        full_bbox = torch.cat([bbox, vis_bbox], dim=1)
        bias = self._rel_bias(full_pos, full_bbox)
        for layer in self.layers:
            full = layer(full, bias)
        return full[:, :S], full[:, S:]


class LayoutLMv2ForTokenClassification(LayoutLMv2PretrainedModel):
    def __init__(self, config: LayoutLMv2Config):
        super().__init__(config)
        self.layoutlmv2 = LayoutLMv2Model(config)
        self.dropout = nn.Dropout(config.hidden_dropout_prob)
        self.classifier = nn.Linear(config.hidden_size, config.num_labels)
        self.init_weights()

    def forward(self, input_ids, bbox=None, image=None,
                token_type_ids=None, labels=None):
        text, _ = self.layoutlmv2(input_ids, bbox, image, token_type_ids)
        logits = self.classifier(self.dropout(text))
        if labels is not None:
            loss = F.cross_entropy(
                logits.reshape(-1, self.config.num_labels),
                labels.reshape(-1), ignore_index=-100)
            return loss, logits
        return logits
