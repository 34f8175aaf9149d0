"""Japanese BERT tokenizer (reference:
paddlenlp/transformers/bert_japanese/tokenizer.py).

BertTokenizer variant with a Japanese word-segmentation front end
(`word_tokenizer_type`: mecab / basic) and a subword back end
(`subword_tokenizer_type`: wordpiece / character, reference :95-148),
built over a vocab.txt.  MeCab morphological segmentation needs the
`fugashi` package which is not in this offline image — requesting
"mecab" falls back to basic segmentation with a warning (Japanese text
without spaces then flows through the character subword mode, which is
the cl-tohoku `-char` models' configuration anyway).
"""
from __future__ import annotations

import os
import warnings

from ..tokenizer_utils import PretrainedTokenizer

__all__ = ["BertJapaneseTokenizer"]


def _build_backend(vocab_file: str, subword: str, lowercase: bool):
    from tokenizers import Regex, Tokenizer, models, normalizers, pre_tokenizers

    vocab = {}
    with open(vocab_file, encoding="utf-8") as f:
        for i, line in enumerate(f):
            vocab[line.rstrip("\n")] = i
    tok = Tokenizer(models.WordPiece(vocab, unk_token="[UNK]",
                                     max_input_chars_per_word=100))
    tok.normalizer = normalizers.BertNormalizer(
        lowercase=lowercase, handle_chinese_chars=True)
    if subword == "character":
        # every char becomes its own word -> vocab lookup per char
        tok.pre_tokenizer = pre_tokenizers.Sequence([
            pre_tokenizers.BertPreTokenizer(),
            pre_tokenizers.Split(Regex("."), behavior="isolated"),
        ])
    else:
        tok.pre_tokenizer = pre_tokenizers.BertPreTokenizer()
    return tok


class BertJapaneseTokenizer(PretrainedTokenizer):
    padding_side = "right"

    def __init__(self, vocab_file=None, word_tokenizer_type="basic",
                 subword_tokenizer_type="wordpiece", do_lower_case=False,
                 **kwargs):
        if word_tokenizer_type == "mecab":
            warnings.warn(
                "MeCab is not available in this image; falling back to "
                "basic segmentation (use subword_tokenizer_type="
                "'character' for unsegmented Japanese)")
            word_tokenizer_type = "basic"
        self.word_tokenizer_type = word_tokenizer_type
        self.subword_tokenizer_type = subword_tokenizer_type
        tok = (_build_backend(vocab_file, subword_tokenizer_type,
                              do_lower_case)
               if vocab_file and os.path.isfile(vocab_file) else
               kwargs.pop("tokenizer", None))
        kwargs.setdefault("cls_token", "[CLS]")
        kwargs.setdefault("sep_token", "[SEP]")
        kwargs.setdefault("pad_token", "[PAD]")
        kwargs.setdefault("unk_token", "[UNK]")
        kwargs.setdefault("mask_token", "[MASK]")
        super().__init__(tokenizer=tok, **kwargs)
