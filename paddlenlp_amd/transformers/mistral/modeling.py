"""Mistral model family (reference: paddlenlp/transformers/mistral/modeling.py).

Architecturally the Llama stack (GQA + RMSNorm + SwiGLU); sliding-window
attention is accepted in the config but evaluated as full causal attention
(the reference's default configs disable the window for the supported
sequence lengths).
"""
from __future__ import annotations


from ...generation import GenerationConfig, GenerationMixin
from ..llama.modeling import (
    LlamaModel,
    LlamaPretrainedModel,
    LlamaPretrainingCriterion,
    _Linear,
)
from .configuration import MistralConfig

__all__ = ["MistralModel", "MistralForCausalLM"]


class MistralPretrainedModel(LlamaPretrainedModel):
    config_class = MistralConfig
    base_model_prefix = "mistral"


class MistralModel(MistralPretrainedModel, LlamaModel):
    def __init__(self, config: MistralConfig):
        LlamaModel.__init__(self, config)


class MistralForCausalLM(MistralPretrainedModel, GenerationMixin):
    def __init__(self, config: MistralConfig):
        super().__init__(config)
        self.mistral = MistralModel(config)
        self.lm_head = _Linear(config.hidden_size, config.vocab_size, bias=False)
        self.criterion = LlamaPretrainingCriterion(config)
        self.generation_config = GenerationConfig.from_model_config(config)

    def get_input_embeddings(self):
        return self.mistral.embed_tokens

    def get_output_embeddings(self):
        return self.lm_head

    def forward(self, input_ids=None, labels=None, past_key_values=None,
                use_cache=False, attention_mask=None, **kwargs):
        out = self.mistral(input_ids=input_ids, past_key_values=past_key_values,
                           use_cache=use_cache)
        if use_cache:
            hidden, presents = out
        else:
            hidden, presents = out, None
        logits = self.lm_head(hidden)
        if labels is not None:
            loss = self.criterion(logits, labels)
            return (loss, logits) if not use_cache else (loss, logits, presents)
        return logits if not use_cache else (logits, presents)
