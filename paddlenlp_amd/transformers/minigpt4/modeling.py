"""MiniGPT-4 (reference: paddlenlp/transformers/minigpt4/modeling.py).

BLIP-2-style bridge into a Llama chat LM: ViT vision encoder + Q-Former
(reference duplicates the BLIP-2 stack :183-700; here it IS the shared
Blip2 bridge) projected to the Llama hidden size and prepended as soft
prompts; generation runs through the framework LlamaForCausalLM.
"""
from __future__ import annotations

import torch
import torch.nn as nn
import torch.nn.functional as F

from ..blip_2.modeling import Blip2Config, Blip2Model
from ..configuration_utils import PretrainedConfig
from ..llama import LlamaConfig, LlamaForCausalLM
from ..model_utils import PretrainedModel

__all__ = ["MiniGPT4Config", "MiniGPT4ForConditionalGeneration"]


class MiniGPT4Config(PretrainedConfig):
    model_type = "minigpt4"

    def __init__(self, vision_config=None, qformer_config=None,
                 text_config=None, num_query_tokens=32, **kwargs):
        super().__init__(**kwargs)
        self.text_config = LlamaConfig(**(text_config or {}))
        self.bridge_config = Blip2Config(
            vision_config=vision_config, qformer_config=qformer_config,
            num_query_tokens=num_query_tokens,
            lm_hidden_size=self.text_config.hidden_size)
        self.num_query_tokens = num_query_tokens
        self.initializer_range = self.text_config.initializer_range


class MiniGPT4ForConditionalGeneration(PretrainedModel):
    config_class = MiniGPT4Config
    base_model_prefix = "minigpt4"

    def _init_weights(self, module):
        pass  # sub-models initialize themselves

    def __init__(self, config: MiniGPT4Config):
        super().__init__(config)
        self.bridge = Blip2Model(config.bridge_config)
        self.language_model = LlamaForCausalLM(config.text_config)

    def forward(self, pixel_values, input_ids, labels=None):
        prompt = self.bridge(pixel_values)            # [B, Q, H]
        embed = self.language_model.get_input_embeddings()(input_ids)
        inputs_embeds = torch.cat([prompt, embed], dim=1)
        logits = self.language_model(inputs_embeds=inputs_embeds)
        if labels is not None:
            # loss only over the text region (visual prompt skipped)
            text_logits = logits[:, prompt.shape[1]:]
            loss = F.cross_entropy(
                text_logits[:, :-1].reshape(-1, logits.shape[-1]),
                labels[:, 1:].reshape(-1), ignore_index=-100)
            return loss, logits
        return logits
