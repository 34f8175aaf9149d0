"""Prompt templates: hard text, input fields, soft (learnable) tokens, mask.

Reference behavior: paddlenlp/prompt/template.py (ManualTemplate /
SoftTemplate part lists: {'text': field} {'hard': '...'} {'soft': n}
{'mask'}).  render() produces token ids plus the mask position and the
positions of soft tokens (the model injects learnable embeddings there).
"""
from __future__ import annotations

from typing import Dict, List

SOFT_PLACEHOLDER_ID = -1  # replaced by learnable embeddings in the model


class Template:
    def __init__(self, parts: List[dict], tokenizer,
                 mask_token: str = "[MASK]"):
        self.parts = parts
        self.tokenizer = tokenizer
        self.mask_id = tokenizer._tokenizer.token_to_id(mask_token)
        assert self.mask_id is not None, f"{mask_token} not in vocab"
        self.num_soft_tokens = sum(
            int(p.get("soft") or 0) for p in parts if "soft" in p)

    def render(self, example: Dict[str, str]) -> dict:
        """-> {input_ids, mask_position, soft_positions} (unpadded)."""
        ids: List[int] = []
        soft_positions: List[int] = []
        mask_position = None
        soft_index = 0
        for part in self.parts:
            if "text" in part:
                ids.extend(self.tokenizer._tokenizer.encode(
                    str(example[part["text"]])).ids)
            elif "hard" in part:
                ids.extend(self.tokenizer._tokenizer.encode(part["hard"]).ids)
            elif "soft" in part:
                n = int(part.get("soft") or 1)
                for _ in range(n):
                    soft_positions.append(len(ids))
                    ids.append(SOFT_PLACEHOLDER_ID)
                    soft_index += 1
            elif "mask" in part:
                mask_position = len(ids)
                ids.append(self.mask_id)
            else:
                raise ValueError(f"unknown template part {part}")
        assert mask_position is not None, "template needs a {'mask'} part"
        return {"input_ids": ids, "mask_position": mask_position,
                "soft_positions": soft_positions}


class ManualTemplate(Template):
    """Hard-text-only template (no soft tokens)."""

    def __init__(self, parts, tokenizer, **kwargs):
        assert not any("soft" in p for p in parts), \
            "ManualTemplate takes no soft parts; use SoftTemplate"
        super().__init__(parts, tokenizer, **kwargs)


class SoftTemplate(Template):
    pass
