"""GenerationConfig (reference: paddlenlp/generation/configuration_utils.py)."""
from __future__ import annotations

import json
import os
from dataclasses import asdict, dataclass
from typing import List, Optional, Union

from ..utils.env import GENERATION_CONFIG_NAME


@dataclass
class GenerationConfig:
    max_new_tokens: int = 64
    min_new_tokens: int = 0
    max_length: Optional[int] = None
    do_sample: bool = False
    num_beams: int = 1
    num_beam_groups: int = 1
    diversity_penalty: float = 0.0
    temperature: float = 1.0
    top_k: int = 50
    top_p: float = 1.0
    repetition_penalty: float = 1.0
    length_penalty: float = 1.0
    early_stopping: bool = False
    num_return_sequences: int = 1
    use_cache: bool = True
    no_repeat_ngram_size: int = 0
    forced_bos_token_id: Optional[int] = None
    forced_eos_token_id: Optional[int] = None
    bad_words_ids: Optional[List[List[int]]] = None
    sequence_bias: Optional[dict] = None
    prefix_allowed_tokens_fn: Optional[object] = None
    bos_token_id: Optional[int] = None
    eos_token_id: Optional[Union[int, List[int]]] = None
    pad_token_id: Optional[int] = None

    def eos_ids(self) -> List[int]:
        if self.eos_token_id is None:
            return []
        if isinstance(self.eos_token_id, int):
            return [self.eos_token_id]
        return list(self.eos_token_id)

    def save_pretrained(self, save_directory: str):
        os.makedirs(save_directory, exist_ok=True)
        payload = {k: v for k, v in asdict(self).items()
                   if not callable(v) and k != "prefix_allowed_tokens_fn"}
        with open(os.path.join(save_directory, GENERATION_CONFIG_NAME), "w") as f:
            json.dump(payload, f, indent=2)

    @classmethod
    def from_pretrained(cls, path: str):
        f = os.path.join(path, GENERATION_CONFIG_NAME)
        if not os.path.isfile(f):
            return cls()
        with open(f) as fh:
            data = json.load(fh)
        known = {k: v for k, v in data.items() if k in cls.__dataclass_fields__}
        return cls(**known)

    @classmethod
    def from_model_config(cls, config):
        return cls(
            bos_token_id=getattr(config, "bos_token_id", None),
            eos_token_id=getattr(config, "eos_token_id", None),
            pad_token_id=getattr(config, "pad_token_id", None),
        )
