"""Static export of a CausalLM for deployment.

Reference behavior: llm/predict/export_model.py (dy2static `model.to_static`
+ jit.save + tokenizer/config copy).  The MI355X equivalent uses
torch.export: the forward graph is traced once with dynamic batch and
sequence dimensions and serialized as an ExportedProgram (.pt2) next to the
config and tokenizer, so a serving process can load and run it without the
Python modeling code.

Usage:
    python llm/predict/export_model.py --model_name_or_path ckpt/ \
        --output_path exported/ [--dtype bfloat16]
"""
from __future__ import annotations

import argparse
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.dirname(os.path.abspath(__file__)))))

import torch

from paddlenlp_amd.transformers import AutoModelForCausalLM
from paddlenlp_amd.transformers.tokenizer_utils import PretrainedTokenizer
from paddlenlp_amd.utils.log import logger

EXPORT_NAME = "model.pt2"


class _ForwardLogits(torch.nn.Module):
    """Export wrapper: (input_ids) -> logits, dropping aux outputs."""

    def __init__(self, model):
        super().__init__()
        self.model = model

    def forward(self, input_ids):
        out = self.model(input_ids=input_ids)
        if isinstance(out, tuple):
            out = out[0]
        elif isinstance(out, dict):
            out = out.get("logits", next(iter(out.values())))
        return out


def export_model(model, output_path: str, example_batch: int = 2,
                 example_seq: int = 8) -> str:
    """Trace + serialize the forward; returns the .pt2 path."""
    os.makedirs(output_path, exist_ok=True)
    model.eval()
    wrapper = _ForwardLogits(model)
    ids = torch.randint(0, model.config.vocab_size,
                        (example_batch, example_seq),
                        device=next(model.parameters()).device)
    # example batch >= 2: a size-1 example dim gets specialized by export
    b = torch.export.Dim("batch", min=2, max=4096)
    s = torch.export.Dim("seq", min=2, max=getattr(
        model.config, "max_position_embeddings", 8192))
    ep = torch.export.export(wrapper, (ids,),
                             dynamic_shapes={"input_ids": {0: b, 1: s}})
    path = os.path.join(output_path, EXPORT_NAME)
    torch.export.save(ep, path)
    if hasattr(model, "config"):
        model.config.save_pretrained(output_path)
    logger.info(f"Exported static program to {path}")
    return path


def load_exported(output_path: str):
    """Load the serialized program; returns a callable (input_ids -> logits)."""
    ep = torch.export.load(os.path.join(output_path, EXPORT_NAME))
    return ep.module()


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--model_name_or_path", required=True)
    p.add_argument("--output_path", required=True)
    p.add_argument("--dtype", default="bfloat16")
    args = p.parse_args()

    model = AutoModelForCausalLM.from_pretrained(args.model_name_or_path,
                                                 dtype=args.dtype)
    if torch.cuda.is_available():
        model = model.to("cuda")
    export_model(model, args.output_path)
    try:
        tok = PretrainedTokenizer.from_pretrained(args.model_name_or_path)
        tok.save_pretrained(args.output_path)
    except FileNotFoundError:
        logger.warning("no tokenizer found next to the checkpoint; skipped")


if __name__ == "__main__":
    main()
