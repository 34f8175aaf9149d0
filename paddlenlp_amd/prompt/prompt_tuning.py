"""Soft prompt tuning (reference: paddlenlp/prompt — template/verbalizer
framework; this is the soft-prompt core used by causal LMs)."""
from __future__ import annotations

import json
import os
from dataclasses import asdict, dataclass

import torch
import torch.nn as nn


@dataclass
class PromptTuningConfig:
    num_prompt_tokens: int = 16
    init_from_vocab: bool = True


class PromptTuningModel(nn.Module):
    """Prepends trainable soft-prompt embeddings to the input."""

    def __init__(self, model, config: PromptTuningConfig):
        super().__init__()
        self.model = model
        self.prompt_config = config
        emb = model.get_input_embeddings()
        hidden = emb.weight.shape[1]
        if config.init_from_vocab:
            idx = torch.randint(0, emb.weight.shape[0], (config.num_prompt_tokens,))
            init = emb.weight[idx].detach().clone()
        else:
            init = torch.randn(config.num_prompt_tokens, hidden) * 0.02
        self.soft_prompt = nn.Parameter(init)
        for p in self.model.parameters():
            p.requires_grad_(False)

    @property
    def config(self):
        return self.model.config

    def forward(self, input_ids=None, labels=None, **kwargs):
        emb = self.model.get_input_embeddings()(input_ids)
        B = emb.shape[0]
        prompt = self.soft_prompt[None].expand(B, -1, -1).to(emb.dtype)
        inputs_embeds = torch.cat([prompt, emb], dim=1)
        if labels is not None:
            pad = labels.new_full((B, self.prompt_config.num_prompt_tokens), -100)
            labels = torch.cat([pad, labels], dim=1)
        return self.model(inputs_embeds=inputs_embeds, labels=labels, **kwargs)

    def save_pretrained(self, path):
        os.makedirs(path, exist_ok=True)
        with open(os.path.join(path, "prompt_config.json"), "w") as f:
            json.dump(asdict(self.prompt_config), f)
        torch.save({"soft_prompt": self.soft_prompt.detach().cpu()},
                   os.path.join(path, "prompt_state.pt"))
