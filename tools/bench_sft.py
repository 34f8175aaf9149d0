"""SFT-recipe throughput: Llama-3-8B with ZeroPadding packing + FlashMask.

The reference's SFT fast path packs samples to max_length and replaces
the attention mask with `attn_mask_startend_row_indices` (FlashMask).
This measures that exact training step — packed batches, FlashMask
attention (the v1 masked kernel), fused loss — in tokens/s, the same
convention as bench.py.

Run (GPU box): python tools/bench_sft.py --steps 6 --warmup 2
"""
from __future__ import annotations

import argparse
import json
import os
import sys
import time

import numpy as np
import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from paddlenlp_amd.datasets.zero_padding_dataset import (
    generate_startend_row_indices,
)
from paddlenlp_amd.trainer.optimizer import FusedAdamW
from paddlenlp_amd.transformers import LlamaConfig, LlamaForCausalLM

MODELS = {
    "llama3-8b": dict(
        vocab_size=128256, hidden_size=4096, intermediate_size=14336,
        num_hidden_layers=32, num_attention_heads=32, num_key_value_heads=8,
        max_position_embeddings=8192, rope_theta=500000.0,
    ),
}


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--model", default="llama3-8b")
    p.add_argument("--micro-batch", type=int, default=8)
    p.add_argument("--seq-len", type=int, default=4096)
    p.add_argument("--samples-per-pack", type=int, default=8)
    p.add_argument("--steps", type=int, default=6)
    p.add_argument("--warmup", type=int, default=2)
    args = p.parse_args()

    assert torch.cuda.is_available()
    device = torch.device("cuda:0")
    cfg = LlamaConfig(
        **MODELS[args.model], dtype="bfloat16",
        fuse_attention_qkv=True, fuse_attention_ffn=True,
        use_flash_attention=True, use_fused_rms_norm=True,
        use_fused_rope=True, use_fused_swiglu=True,
    )
    model = LlamaForCausalLM.from_config(cfg, dtype=torch.bfloat16,
                                         device=device)
    model.train()
    opt = FusedAdamW(model.parameters(), lr=1e-5, master_weights=True)

    # synthetic packed batch: samples_per_pack segments per row, the
    # FlashMask indices block cross-sample attention
    g = torch.Generator().manual_seed(0)
    S = args.seq_len
    seg = S // args.samples_per_pack
    boundaries = [seg * (i + 1) for i in range(args.samples_per_pack)]
    se_row = generate_startend_row_indices(boundaries, S)
    se = torch.from_numpy(np.asarray(se_row)).to(device)
    if se.dim() == 3:
        se = se.unsqueeze(0)
    se = se.expand(args.micro_batch, *se.shape[1:]).contiguous()
    ids = torch.randint(0, cfg.vocab_size, (args.micro_batch, S + 1),
                        generator=g).to(device)
    batch = {
        "input_ids": ids[:, :-1],
        "labels": ids[:, 1:].contiguous(),
        "attn_mask_startend_row_indices": se,
    }

    def step():
        opt.zero_grad(set_to_none=True)
        loss, _ = model(**batch)
        loss.backward()
        opt.step()
        return loss

    for _ in range(args.warmup):
        loss = step()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(args.steps):
        loss = step()
    torch.cuda.synchronize()
    dt = time.perf_counter() - t0

    tokens = args.micro_batch * S * args.steps
    print(json.dumps({
        "metric": "sft_flashmask_tokens_per_sec",
        "value": round(tokens / dt, 1),
        "ms_per_step": round(dt / args.steps * 1000, 1),
        "model": args.model, "seq_len": S,
        "samples_per_pack": args.samples_per_pack,
        "micro_batch": args.micro_batch,
        "loss": round(float(loss), 4),
    }))


if __name__ == "__main__":
    main()
