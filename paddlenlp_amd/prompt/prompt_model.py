"""PromptModelForSequenceClassification: template + MLM + verbalizer.

Reference behavior: paddlenlp/prompt/prompt_model.py — renders the template
per example, injects learnable soft-token embeddings, reads the MLM logits
at the mask position and maps them to label logits via the verbalizer.
"""
from __future__ import annotations

from typing import Dict, List

import torch
import torch.nn as nn
import torch.nn.functional as F

from .template import SOFT_PLACEHOLDER_ID, Template
from .verbalizer import ManualVerbalizer, SoftVerbalizer


class PromptModelForSequenceClassification(nn.Module):
    def __init__(self, mlm_model, template: Template, verbalizer,
                 freeze_plm: bool = False):
        super().__init__()
        self.plm = mlm_model
        self.template = template
        self.verbalizer = verbalizer
        embed = mlm_model.get_input_embeddings()
        self.embed = embed
        if template.num_soft_tokens:
            # soft prompts initialized from random vocab rows
            init = embed.weight[
                torch.randint(0, embed.num_embeddings,
                              (template.num_soft_tokens,))].detach().clone()
            self.soft_embeddings = nn.Parameter(init)
        else:
            self.soft_embeddings = None
        if freeze_plm:
            for p in self.plm.parameters():
                p.requires_grad_(False)

    def _encode_batch(self, examples: List[Dict[str, str]], device):
        rendered = [self.template.render(e) for e in examples]
        maxlen = max(len(r["input_ids"]) for r in rendered)
        pad_id = 0
        ids = torch.full((len(rendered), maxlen), pad_id, dtype=torch.long,
                         device=device)
        mask_pos = torch.zeros(len(rendered), dtype=torch.long, device=device)
        soft_map = []  # (batch_idx, seq_pos, soft_idx)
        for b, r in enumerate(rendered):
            for j, t in enumerate(r["input_ids"]):
                ids[b, j] = t if t != SOFT_PLACEHOLDER_ID else 0
            mask_pos[b] = r["mask_position"]
            for si, pos in enumerate(r["soft_positions"]):
                soft_map.append((b, pos, si))
        return ids, mask_pos, soft_map

    def forward(self, examples: List[Dict[str, str]], labels=None):
        device = next(self.plm.parameters()).device
        ids, mask_pos, soft_map = self._encode_batch(examples, device)
        if self.soft_embeddings is not None and soft_map:
            # inject soft prompts through an embedding override: build
            # inputs_embeds-equivalent by a post-embedding swap hook
            embeds = self.embed(ids)
            for b, pos, si in soft_map:
                embeds[b, pos] = self.soft_embeddings[si].to(embeds.dtype)
            logits = self._forward_with_embeds(ids, embeds)
        else:
            logits = self._mlm_logits(ids)
        batch_idx = torch.arange(ids.shape[0], device=device)
        if isinstance(self.verbalizer, SoftVerbalizer):
            raise NotImplementedError(
                "SoftVerbalizer needs hidden states; use it with "
                "process_hidden over the backbone directly")
        mask_logits = logits[batch_idx, mask_pos].float()
        label_logits = self.verbalizer.process_logits(mask_logits)
        if labels is not None:
            loss = F.cross_entropy(label_logits, labels.view(-1).to(device))
            return loss, label_logits
        return label_logits

    def _mlm_logits(self, ids):
        out = self.plm(ids)
        return out[1] if isinstance(out, tuple) else out

    def _forward_with_embeds(self, ids, embeds):
        """Run the MLM with swapped input embeddings via a temporary
        forward hook on the embedding module."""
        handle_out = {}

        def hook(module, args, output):
            return embeds

        h = self.embed.register_forward_hook(hook)
        try:
            logits = self._mlm_logits(ids)
        finally:
            h.remove()
        return logits

    def predict(self, examples):
        with torch.no_grad():
            logits = self.forward(examples)
        idx = logits.argmax(-1)
        return [self.verbalizer.labels[int(i)] for i in idx]
