"""Tokenizer layer: wraps the HF `tokenizers` Rust library.

Reference behavior: paddlenlp/transformers/tokenizer_utils_fast.py — in the
reference, "fast tokenizers" are also a thin wrapper over the same Rust
library (tokenizer_utils_fast.py:27-29), so this is a dependency both
frameworks share rather than ported code.  The Python-side API surface
mirrors PretrainedTokenizer (tokenizer_utils_base.py): encode/decode/
batch, text pairs with token_type_ids, truncation strategies
(longest_first / only_first / only_second), left/right padding with
pad_to_multiple_of, offsets mapping, special-token registration and
chat templates.
"""
from __future__ import annotations

import json
import os
from typing import Dict, List, Optional, Union

from ..utils.env import TOKENIZER_CONFIG_NAME

__all__ = ["PretrainedTokenizer"]

SPECIAL_TOKENS_MAP_NAME = "special_tokens_map.json"


class PretrainedTokenizer:
    """Tokenizer backed by a tokenizers.Tokenizer (tokenizer.json)."""

    padding_side: str = "left"   # decoder-only default; encoders pass "right"

    def __init__(
        self,
        tokenizer=None,
        bos_token: Optional[str] = None,
        eos_token: Optional[str] = None,
        pad_token: Optional[str] = None,
        unk_token: Optional[str] = None,
        cls_token: Optional[str] = None,
        sep_token: Optional[str] = None,
        mask_token: Optional[str] = None,
        additional_special_tokens: Optional[List[str]] = None,
        chat_template: Optional[str] = None,
        model_max_length: int = 1 << 30,
        padding_side: Optional[str] = None,
        **kwargs,
    ):
        self._tokenizer = tokenizer
        self.bos_token = bos_token
        self.eos_token = eos_token
        self.pad_token = pad_token or eos_token
        self.unk_token = unk_token
        self.cls_token = cls_token
        self.sep_token = sep_token
        self.mask_token = mask_token
        self.additional_special_tokens = list(additional_special_tokens or [])
        self.chat_template = chat_template
        self.model_max_length = model_max_length
        if padding_side is not None:
            self.padding_side = padding_side
        self.init_kwargs = kwargs

    # ---- special tokens ----
    def _token_id(self, token):
        if token is None or self._tokenizer is None:
            return None
        return self._tokenizer.token_to_id(token)

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
    def unk_token_id(self):
        return self._token_id(self.unk_token)

    @property
    def cls_token_id(self):
        return self._token_id(self.cls_token)

    @property
    def sep_token_id(self):
        return self._token_id(self.sep_token)

    @property
    def mask_token_id(self):
        return self._token_id(self.mask_token)

    @property
    def all_special_tokens(self) -> List[str]:
        toks = [self.bos_token, self.eos_token, self.pad_token, self.unk_token,
                self.cls_token, self.sep_token, self.mask_token]
        toks += self.additional_special_tokens
        seen, out = set(), []
        for t in toks:
            if t is not None and t not in seen:
                seen.add(t)
                out.append(t)
        return out

    @property
    def all_special_ids(self) -> List[int]:
        return [i for i in (self._token_id(t) for t in self.all_special_tokens)
                if i is not None]

    def add_tokens(self, new_tokens: Union[str, List[str]]) -> int:
        """Register ordinary new tokens (reference add_tokens)."""
        if isinstance(new_tokens, str):
            new_tokens = [new_tokens]
        return self._tokenizer.add_tokens(new_tokens)

    def add_special_tokens(self, special_tokens_dict: Dict) -> int:
        """Register special tokens; returns the number added (reference
        tokenizer_utils_base add_special_tokens)."""
        from tokenizers import AddedToken

        added = 0
        extra = special_tokens_dict.pop("additional_special_tokens", [])
        for key, value in special_tokens_dict.items():
            assert key.endswith("_token"), key
            content = value.get("content") if isinstance(value, dict) else value
            setattr(self, key, content)
            if self._tokenizer.token_to_id(content) is None:
                added += self._tokenizer.add_special_tokens(
                    [AddedToken(content, special=True)])
        for t in extra:
            content = t.get("content") if isinstance(t, dict) else t
            if content not in self.additional_special_tokens:
                self.additional_special_tokens.append(content)
            if self._tokenizer.token_to_id(content) is None:
                added += self._tokenizer.add_special_tokens(
                    [AddedToken(content, special=True)])
        return added

    @property
    def vocab_size(self):
        return self._tokenizer.get_vocab_size() if self._tokenizer else 0

    def get_vocab(self) -> Dict[str, int]:
        return self._tokenizer.get_vocab()

    def __len__(self):
        return self.vocab_size

    # ---- encode / decode ----
    def encode(self, text: str, text_pair: Optional[str] = None,
               add_special_tokens: bool = True) -> List[int]:
        enc = self._tokenizer.encode(text, pair=text_pair,
                                     add_special_tokens=add_special_tokens)
        return enc.ids

    @staticmethod
    def _truncate(ids_a, ids_b, max_length, strategy):
        """Pair-aware truncation (reference truncate_sequences):
        longest_first trims tokens one at a time from the longer side."""
        if max_length is None:
            return ids_a, ids_b
        if ids_b is None:
            return ids_a[:max_length], None
        if strategy == "only_first":
            keep = max(0, max_length - len(ids_b))
            return ids_a[:keep], ids_b
        if strategy == "only_second":
            keep = max(0, max_length - len(ids_a))
            return ids_a, ids_b[:keep]
        # longest_first
        a, b = list(ids_a), list(ids_b)
        while len(a) + len(b) > max_length:
            if len(a) >= len(b):
                a.pop()
            else:
                b.pop()
        return a, b

    def __call__(
        self,
        text: Union[str, List[str]],
        text_pair: Optional[Union[str, List[str]]] = None,
        add_special_tokens: bool = True,
        padding: Union[bool, str] = False,
        truncation: Union[bool, str] = False,
        max_length: Optional[int] = None,
        return_tensors: Optional[str] = None,
        return_token_type_ids: Optional[bool] = None,
        return_offsets_mapping: bool = False,
        return_attention_mask: bool = True,
        return_length: bool = False,
        pad_to_multiple_of: Optional[int] = None,
        padding_side: Optional[str] = None,
    ) -> Dict:
        single = isinstance(text, str)
        texts = [text] if single else list(text)
        pairs = None
        if text_pair is not None:
            pairs = [text_pair] if isinstance(text_pair, str) else list(text_pair)
            assert len(pairs) == len(texts)
        if return_token_type_ids is None:
            return_token_type_ids = pairs is not None
        if max_length is None and truncation:
            max_length = self.model_max_length

        trunc_strategy = truncation if isinstance(truncation, str) else "longest_first"
        ids, type_ids, offsets = [], [], []
        if pairs is not None and truncation:
            # encode sides separately so the pair-truncation strategy applies
            enc_a = self._tokenizer.encode_batch(texts, add_special_tokens=False)
            enc_b = self._tokenizer.encode_batch(pairs, add_special_tokens=False)
            for ea, eb in zip(enc_a, enc_b):
                a, b = self._truncate(ea.ids, eb.ids, max_length, trunc_strategy)
                ids.append(list(a) + list(b))
                type_ids.append([0] * len(a) + [1] * len(b))
                offs = list(ea.offsets[:len(a)]) + list(eb.offsets[:len(b)])
                offsets.append(offs)
        else:
            batch_in = list(zip(texts, pairs)) if pairs is not None else texts
            encs = self._tokenizer.encode_batch(batch_in,
                                                add_special_tokens=add_special_tokens)
            for e in encs:
                i = e.ids
                ti = e.type_ids
                off = list(e.offsets)
                if truncation and max_length:
                    i, ti, off = i[:max_length], ti[:max_length], off[:max_length]
                ids.append(i)
                type_ids.append(ti)
                offsets.append(off)

        side = padding_side or self.padding_side
        if padding:
            pad_id = self.pad_token_id if self.pad_token_id is not None else 0
            longest = (max_length if (padding == "max_length" and max_length)
                       else max(len(i) for i in ids))
            if pad_to_multiple_of:
                longest = ((longest + pad_to_multiple_of - 1)
                           // pad_to_multiple_of * pad_to_multiple_of)
            attn = []
            for j, i in enumerate(ids):
                npad = longest - len(i)
                if side == "left":
                    attn.append([0] * npad + [1] * len(i))
                    ids[j] = [pad_id] * npad + i
                    type_ids[j] = [0] * npad + type_ids[j]
                    offsets[j] = [(0, 0)] * npad + offsets[j]
                else:
                    attn.append([1] * len(i) + [0] * npad)
                    ids[j] = i + [pad_id] * npad
                    type_ids[j] = type_ids[j] + [0] * npad
                    offsets[j] = offsets[j] + [(0, 0)] * npad
        else:
            attn = [[1] * len(i) for i in ids]

        out = {"input_ids": ids}
        if return_attention_mask:
            out["attention_mask"] = attn
        if return_token_type_ids:
            out["token_type_ids"] = type_ids
        if return_offsets_mapping:
            out["offset_mapping"] = offsets
        if return_length:
            out["length"] = [len(i) for i in ids]
        if single and not return_tensors:
            out = {k: v[0] for k, v in out.items()}
        if return_tensors == "pt":
            import torch

            out = {k: torch.tensor(v, dtype=torch.int64)
                   for k, v in out.items() if k != "offset_mapping"}
            if return_offsets_mapping:
                import torch as _t

                out["offset_mapping"] = _t.tensor(offsets, dtype=_t.int64)
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

    def convert_tokens_to_string(self, tokens: List[str]) -> str:
        ids = [self._tokenizer.token_to_id(t) for t in tokens]
        return self._tokenizer.decode([i for i in ids if i is not None],
                                      skip_special_tokens=False)

    # ---- chat template (jinja-backed; minimal fallback) ----
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
            "cls_token": self.cls_token,
            "sep_token": self.sep_token,
            "mask_token": self.mask_token,
            "additional_special_tokens": self.additional_special_tokens,
            "chat_template": self.chat_template,
            "model_max_length": self.model_max_length,
            "padding_side": self.padding_side,
        }
        with open(os.path.join(save_directory, TOKENIZER_CONFIG_NAME), "w") as f:
            json.dump(cfg, f, indent=2)
        smap = {k: getattr(self, k) for k in
                ("bos_token", "eos_token", "pad_token", "unk_token",
                 "cls_token", "sep_token", "mask_token")
                if getattr(self, k) is not None}
        if self.additional_special_tokens:
            smap["additional_special_tokens"] = self.additional_special_tokens
        with open(os.path.join(save_directory, SPECIAL_TOKENS_MAP_NAME), "w") as f:
            json.dump(smap, f, indent=2)

    @classmethod
    def from_pretrained(cls, path: str, **kwargs):
        from tokenizers import Tokenizer

        tok_file = os.path.join(path, "tokenizer.json")
        if not os.path.isfile(tok_file):
            raise FileNotFoundError(f"tokenizer.json not found in {path}")
        tokenizer = Tokenizer.from_file(tok_file)
        cfg = {}
        cfg_file = os.path.join(path, TOKENIZER_CONFIG_NAME)
        if os.path.isfile(cfg_file):
            with open(cfg_file) as f:
                cfg = json.load(f)
        smap_file = os.path.join(path, SPECIAL_TOKENS_MAP_NAME)
        if os.path.isfile(smap_file):
            with open(smap_file) as f:
                smap = json.load(f)
            for k, v in smap.items():
                cfg.setdefault(k, v)
        cfg.pop("tokenizer_class", None)
        # HF-style token dicts -> plain strings
        for key in ("bos_token", "eos_token", "pad_token", "unk_token",
                    "cls_token", "sep_token", "mask_token"):
            v = cfg.get(key)
            if isinstance(v, dict):
                cfg[key] = v.get("content")
        ast = cfg.get("additional_special_tokens")
        if isinstance(ast, list):
            cfg["additional_special_tokens"] = [
                t.get("content") if isinstance(t, dict) else t for t in ast]
        known = {"bos_token", "eos_token", "pad_token", "unk_token",
                 "cls_token", "sep_token", "mask_token",
                 "additional_special_tokens", "chat_template",
                 "model_max_length", "padding_side"}
        init = {k: v for k, v in cfg.items() if k in known}
        init.update({k: v for k, v in kwargs.items() if k in known})
        return cls(tokenizer=tokenizer, **init)
