"""Tokenizer layer: wraps the HF `tokenizers` Rust library.

Reference behavior: paddlenlp/transformers/tokenizer_utils_fast.py — in the
reference, "fast tokenizers" are also a thin wrapper over the same Rust
library (tokenizer_utils_fast.py:27-29), so this is a dependency both
frameworks share rather than ported code.  The Python-side API surface
(encode/decode/batch, padding, chat templates, save/load) mirrors
PretrainedTokenizer (tokenizer_utils_base.py).
"""
from __future__ import annotations

import json
import os
from typing import Dict, List, Optional, Union

from ..utils.env import TOKENIZER_CONFIG_NAME

__all__ = ["PretrainedTokenizer"]


class PretrainedTokenizer:
    """Tokenizer backed by a tokenizers.Tokenizer (tokenizer.json)."""

    def __init__(
        self,
        tokenizer=None,
        bos_token: Optional[str] = None,
        eos_token: Optional[str] = None,
        pad_token: Optional[str] = None,
        unk_token: Optional[str] = None,
        chat_template: Optional[str] = None,
        model_max_length: int = 1 << 30,
        **kwargs,
    ):
        self._tokenizer = tokenizer
        self.bos_token = bos_token
        self.eos_token = eos_token
        self.pad_token = pad_token or eos_token
        self.unk_token = unk_token
        self.chat_template = chat_template
        self.model_max_length = model_max_length
        self.init_kwargs = kwargs

    # ---- special-token ids ----
    def _token_id(self, token):
        if token is None or self._tokenizer is None:
            return None
        tid = self._tokenizer.token_to_id(token)
        return tid

    @property
    def bos_token_id(self):
        return self._token_id(self.bos_token)

    @property
    def eos_token_id(self):
        return self._token_id(self.eos_token)

    @property
    def pad_token_id(self):
        return self._token_id(self.pad_token)

    @property
    def vocab_size(self):
        return self._tokenizer.get_vocab_size() if self._tokenizer else 0

    def __len__(self):
        return self.vocab_size

    # ---- encode / decode ----
    def encode(self, text: str, add_special_tokens: bool = True) -> List[int]:
        enc = self._tokenizer.encode(text, add_special_tokens=add_special_tokens)
        return enc.ids

    def __call__(
        self,
        text: Union[str, List[str]],
        add_special_tokens: bool = True,
        padding: bool = False,
        truncation: bool = False,
        max_length: Optional[int] = None,
        return_tensors: Optional[str] = None,
    ) -> Dict:
        single = isinstance(text, str)
        texts = [text] if single else list(text)
        encodings = self._tokenizer.encode_batch(texts, add_special_tokens=add_special_tokens)
        ids = [e.ids for e in encodings]
        if truncation and max_length:
            ids = [i[:max_length] for i in ids]
        if padding:
            pad_id = self.pad_token_id or 0
            longest = max_length if (padding == "max_length" and max_length) else max(len(i) for i in ids)
            attn = [[0] * (longest - len(i)) + [1] * len(i) for i in ids]
            ids = [[pad_id] * (longest - len(i)) + i for i in ids]  # left-pad (decoder-only)
        else:
            attn = [[1] * len(i) for i in ids]
        out = {"input_ids": ids, "attention_mask": attn}
        if single and not return_tensors:
            out = {k: v[0] for k, v in out.items()}
        if return_tensors == "pt":
            import torch
            out = {k: torch.tensor(v, dtype=torch.int64) for k, v in out.items()}
        return out

    def decode(self, token_ids, skip_special_tokens: bool = True) -> str:
        if hasattr(token_ids, "tolist"):
            token_ids = token_ids.tolist()
        return self._tokenizer.decode(token_ids, skip_special_tokens=skip_special_tokens)

    def batch_decode(self, sequences, skip_special_tokens: bool = True) -> List[str]:
        return [self.decode(s, skip_special_tokens) for s in sequences]

    def convert_tokens_to_ids(self, tokens):
        if isinstance(tokens, str):
            return self._tokenizer.token_to_id(tokens)
        return [self._tokenizer.token_to_id(t) for t in tokens]

    def convert_ids_to_tokens(self, ids):
        if isinstance(ids, int):
            return self._tokenizer.id_to_token(ids)
        return [self._tokenizer.id_to_token(i) for i in ids]

    # ---- chat template (jinja-free minimal renderer for common templates) ----
    def apply_chat_template(self, conversation, tokenize=True, add_generation_prompt=True):
        if self.chat_template:
            try:
                import jinja2

                tmpl = jinja2.Template(self.chat_template)
                text = tmpl.render(
                    messages=conversation, add_generation_prompt=add_generation_prompt,
                    bos_token=self.bos_token or "", eos_token=self.eos_token or "",
                )
            except ImportError:
                text = self._default_chat_format(conversation, add_generation_prompt)
        else:
            text = self._default_chat_format(conversation, add_generation_prompt)
        return self.encode(text) if tokenize else text

    def _default_chat_format(self, conversation, add_generation_prompt):
        parts = []
        for msg in conversation:
            parts.append(f"<|{msg['role']}|>\n{msg['content']}\n")
        if add_generation_prompt:
            parts.append("<|assistant|>\n")
        return "".join(parts)

    # ---- save / load ----
    def save_pretrained(self, save_directory: str):
        os.makedirs(save_directory, exist_ok=True)
        if self._tokenizer is not None:
            self._tokenizer.save(os.path.join(save_directory, "tokenizer.json"))
        cfg = {
            "tokenizer_class": type(self).__name__,
            "bos_token": self.bos_token,
            "eos_token": self.eos_token,
            "pad_token": self.pad_token,
            "unk_token": self.unk_token,
            "chat_template": self.chat_template,
            "model_max_length": self.model_max_length,
        }
        with open(os.path.join(save_directory, TOKENIZER_CONFIG_NAME), "w") as f:
            json.dump(cfg, f, indent=2)

    @classmethod
    def from_pretrained(cls, path: str, **kwargs):
        from tokenizers import Tokenizer

        tok_file = os.path.join(path, "tokenizer.json")
        if not os.path.isfile(tok_file):
            raise FileNotFoundError(f"tokenizer.json not found in {path}")
        tokenizer = Tokenizer.from_file(tok_file)
        cfg_file = os.path.join(path, TOKENIZER_CONFIG_NAME)
        cfg = {}
        if os.path.isfile(cfg_file):
            with open(cfg_file) as f:
                cfg = json.load(f)
        cfg.pop("tokenizer_class", None)
        # HF-style token dicts -> plain strings
        for key in ("bos_token", "eos_token", "pad_token", "unk_token"):
            v = cfg.get(key)
            if isinstance(v, dict):
                cfg[key] = v.get("content")
        known = {"bos_token", "eos_token", "pad_token", "unk_token", "chat_template", "model_max_length"}
        init = {k: v for k, v in cfg.items() if k in known}
        cfg.update(kwargs)
        return cls(tokenizer=tokenizer, **init)
