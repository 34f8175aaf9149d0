"""DPO step throughput: packed (FlashMask one-row) vs row-concat forward.

Llama-3-8B pairs at src 512 / response 1536 per side; one optimizer-free
forward+backward per step (the DPO cost profile is dominated by the two
policy/reference forwards + policy backward).

Run (GPU box): python tools/bench_dpo.py --steps 4 --warmup 1
"""
from __future__ import annotations

import argparse
import copy
import json
import os
import sys
import tempfile
import time

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from paddlenlp_amd.trainer import TrainingArguments
from paddlenlp_amd.transformers import LlamaConfig, LlamaForCausalLM
from paddlenlp_amd.trl.dpo_trainer import DPOTrainer

MODEL = dict(
    vocab_size=128256, hidden_size=4096, intermediate_size=14336,
    num_hidden_layers=32, num_attention_heads=32, num_key_value_heads=8,
    max_position_embeddings=8192, rope_theta=500000.0,
)


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--pairs", type=int, default=4)
    p.add_argument("--side-len", type=int, default=2048)
    p.add_argument("--steps", type=int, default=4)
    p.add_argument("--warmup", type=int, default=1)
    args = p.parse_args()

    assert torch.cuda.is_available()
    cfg = LlamaConfig(**MODEL, dtype="bfloat16", fuse_attention_qkv=True,
                      fuse_attention_ffn=True, use_flash_attention=True,
                      use_fused_rms_norm=True, use_fused_rope=True,
                      use_fused_swiglu=True)
    model = LlamaForCausalLM.from_config(cfg, dtype=torch.bfloat16,
                                         device="cuda:0")
    ref = copy.deepcopy(model)

    g = torch.Generator().manual_seed(0)
    L = args.side_len
    inputs = {
        "chosen_input_ids": torch.randint(2, cfg.vocab_size,
                                          (args.pairs, L), generator=g).cuda(),
        "rejected_input_ids": torch.randint(2, cfg.vocab_size,
                                            (args.pairs, L), generator=g).cuda(),
    }
    inputs["chosen_labels"] = inputs["chosen_input_ids"].clone()
    inputs["rejected_labels"] = inputs["rejected_input_ids"].clone()

    with tempfile.TemporaryDirectory() as d:
        targs = TrainingArguments(output_dir=d, max_steps=1,
                                  per_device_train_batch_size=args.pairs,
                                  bf16=True)
        tr = DPOTrainer(model=model, args=targs, beta=0.1,
                        loss_type="sigmoid", reference_model=ref)

        results = {}
        for packing in (True, False):
            tr.dpo_packing = packing
            for _ in range(args.warmup):
                loss = tr.compute_loss(model, inputs)
                loss.backward()
                model.zero_grad(set_to_none=True)
            torch.cuda.synchronize()
            t0 = time.perf_counter()
            for _ in range(args.steps):
                loss = tr.compute_loss(model, inputs)
                loss.backward()
                model.zero_grad(set_to_none=True)
            torch.cuda.synchronize()
            dt = (time.perf_counter() - t0) / args.steps
            tokens = args.pairs * 2 * L
            results["packed" if packing else "rowwise"] = {
                "ms_per_step": round(dt * 1000, 1),
                "tokens_per_sec": round(tokens / dt, 1),
            }
        print(json.dumps({"metric": "dpo_step", "pairs": args.pairs,
                          "side_len": L, **results}))


if __name__ == "__main__":
    main()
