"""Corpus -> .bin/.idx preprocessing (reference: llm/tools/preprocess).

Reads jsonl ({"text": ...} per line), tokenizes, writes the Megatron-format
MMapIndexedDataset the pretrain entry consumes.

Usage:
  python create_pretraining_data.py --input corpus.jsonl \
      --tokenizer ./my-tokenizer-dir --output-prefix ./data/corpus
"""
from __future__ import annotations

import argparse
import json
import os
import sys

import numpy as np

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.dirname(
    os.path.dirname(os.path.abspath(__file__))))))

from paddlenlp_amd.data.indexed_dataset import MMapIndexedDatasetBuilder
from paddlenlp_amd.transformers import AutoTokenizer


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--input", required=True)
    p.add_argument("--tokenizer", required=True)
    p.add_argument("--output-prefix", required=True)
    p.add_argument("--append-eos", action="store_true", default=True)
    args = p.parse_args()

    tok = AutoTokenizer.from_pretrained(args.tokenizer)
    dtype = np.uint16 if tok.vocab_size < 65536 else np.int32
    os.makedirs(os.path.dirname(args.output_prefix) or ".", exist_ok=True)
    builder = MMapIndexedDatasetBuilder(args.output_prefix, dtype=dtype)
    n_docs = n_tokens = 0
    with open(args.input, "r", encoding="utf-8") as f:
        for line in f:
            line = line.strip()
            if not line:
                continue
            text = json.loads(line).get("text", "") if line.startswith("{") else line
            ids = tok.encode(text)
            if args.append_eos and tok.eos_token_id is not None:
                ids.append(tok.eos_token_id)
            builder.add_item(np.asarray(ids, dtype=dtype))
            builder.end_document()
            n_docs += 1
            n_tokens += len(ids)
    builder.finalize()
    print(f"wrote {n_docs} docs / {n_tokens} tokens -> {args.output_prefix}.bin/.idx")


if __name__ == "__main__":
    main()
