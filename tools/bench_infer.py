"""Decode-throughput benchmark for the paged-KV inference engine.

Measures steady-state decode tokens/s for Llama-3-8B (random weights) at
several batch sizes, bf16 vs fp8 weight-only, with and without hipGraph
capture of the decode step.
"""
from __future__ import annotations

import argparse
import json
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

from paddlenlp_amd.experimental import BlockManager, FusedMultiTransformer
from paddlenlp_amd.transformers import LlamaConfig, LlamaForCausalLM

MODELS = {
    "llama3-8b": dict(
        vocab_size=128256, hidden_size=4096, intermediate_size=14336,
        num_hidden_layers=32, num_attention_heads=32, num_key_value_heads=8,
        max_position_embeddings=8192, rope_theta=500000.0,
    ),
    "llama-tiny": dict(
        vocab_size=32000, hidden_size=1024, intermediate_size=2816,
        num_hidden_layers=4, num_attention_heads=8, num_key_value_heads=8,
        max_position_embeddings=8192,
    ),
}


def bench_decode(eng, B, prompt_len, steps, warmup, device, use_graph=False):
    c = eng.config
    mgr = BlockManager(eng.k_caches[0].shape[0], c.block_size,
                       (prompt_len + steps + warmup + c.block_size) // c.block_size + 2, B)
    ids = torch.randint(3, c.vocab_size, (B, prompt_len), device=device)
    lens = torch.full((B,), prompt_len, dtype=torch.int32, device=device)
    slots = [mgr.allocate_slot(prompt_len) for _ in range(B)]
    bt = torch.stack([mgr.block_table[s] for s in slots]).to(device, torch.int32)
    logits = eng.prefill(ids, bt, lens)
    torch.cuda.synchronize()

    tokens = logits.argmax(-1, keepdim=True)
    graph = None
    static = {}

    def one_decode(tok, bt_dev, lens_before):
        return eng.decode_step(tok, bt_dev, lens_before)

    t0 = None
    done = 0
    for i in range(warmup + steps):
        lens_before = torch.tensor([int(mgr.seq_lens[s]) for s in slots],
                                   dtype=torch.int32, device=device)
        for s in slots:
            assert mgr.extend(s, 1)
        bt = torch.stack([mgr.block_table[s] for s in slots]).to(device, torch.int32)

        if use_graph:
            if graph is None:
                # capture with static buffers after one eager step
                static = {
                    "tok": tokens.clone(), "bt": bt.clone(), "lens": lens_before.clone(),
                }
                eng.decode_step(static["tok"], static["bt"], static["lens"])  # warm alloc
                torch.cuda.synchronize()
                graph = torch.cuda.CUDAGraph()
                with torch.cuda.graph(graph):
                    static["logits"] = eng.decode_step(static["tok"], static["bt"], static["lens"])
            static["tok"].copy_(tokens)
            static["bt"].copy_(bt)
            static["lens"].copy_(lens_before)
            graph.replay()
            logits = static["logits"]
        else:
            logits = eng.decode_step(tokens, bt, lens_before)
        tokens = logits.argmax(-1, keepdim=True)
        if i + 1 == warmup:
            torch.cuda.synchronize()
            t0 = time.perf_counter()
    torch.cuda.synchronize()
    elapsed = time.perf_counter() - t0
    return B * steps / elapsed, elapsed / steps * 1000


def bench_device_loop(eng, B, prompt_len, steps, warmup, device):
    """Zero-host-sync decode: fused sampling + in-kernel block scheduler."""
    from paddlenlp_amd.experimental.device_scheduler import DeviceDecodeLoop

    c = eng.config
    nblocks = eng.k_caches[0].shape[0]
    mbs = (prompt_len + steps + warmup) // c.block_size + 4
    loop = DeviceDecodeLoop(eng, max_batch=B, num_blocks=nblocks,
                            max_blocks_per_seq=mbs, device=device,
                            eos_ids=[], max_gen_len=steps + warmup + 8)
    ids = torch.randint(3, c.vocab_size, (B, prompt_len), device=device)
    lens = torch.full((B,), prompt_len, dtype=torch.int32, device=device)
    blocks = [loop.allocate_for_prefill(prompt_len) for _ in range(B)]
    bt = torch.stack([torch.cat([b, torch.full((mbs - b.numel(),), -1,
                                               dtype=torch.int32, device=device)])
                      for b in blocks])
    logits = eng.prefill(ids, bt, lens)
    first = logits.argmax(-1)
    for i in range(B):
        loop.add_request(i, prompt_len, int(first[i]), top_p=0.0, blocks=blocks[i])
    loop.decode_steps(warmup)
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    loop.decode_steps(steps)
    torch.cuda.synchronize()
    elapsed = time.perf_counter() - t0
    return B * steps / elapsed, elapsed / steps * 1000


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--model", default="llama3-8b")
    p.add_argument("--batches", default="1,8,32,64")
    p.add_argument("--prompt-len", type=int, default=512)
    p.add_argument("--steps", type=int, default=64)
    p.add_argument("--warmup", type=int, default=8)
    p.add_argument("--quant", default="")          # "" | fp8 | fp8_ffn | weight_only_int8
    p.add_argument("--graph", action="store_true")
    p.add_argument("--device-sched", action="store_true")
    p.add_argument("--cachekv", default="bf16", choices=["bf16", "int8", "int4"])
    args = p.parse_args()

    device = "cuda:0"
    cfg = LlamaConfig(**MODELS[args.model], dtype="bfloat16")
    print(f"[bench_infer] building {args.model}...", file=sys.stderr)
    model = LlamaForCausalLM.from_config(cfg, dtype=torch.bfloat16, device=device)
    model.eval()
    eng = FusedMultiTransformer.from_llama(model, block_size=64, max_seq_len=8192).to(device)
    del model
    torch.cuda.empty_cache()
    if args.quant == "fp8_ffn":
        # bandwidth-bound subset only: big FFN projections + lm head
        eng.quantize("fp8", names=("gate_up_weights", "down_weights", "lm_head"))
        torch.cuda.empty_cache()
    elif args.quant:
        eng.quantize(args.quant)
        torch.cuda.empty_cache()

    max_b = max(int(b) for b in args.batches.split(","))
    blocks_per_seq = (args.prompt_len + args.steps + args.warmup) // 64 + 2
    eng.allocate_caches(max_b * blocks_per_seq + 8, device, cachekv_dtype=args.cachekv)

    for b in args.batches.split(","):
        B = int(b)
        if args.device_sched:
            tps, ms = bench_device_loop(eng, B, args.prompt_len, args.steps,
                                        args.warmup, device)
        else:
            tps, ms = bench_decode(eng, B, args.prompt_len, args.steps, args.warmup,
                                   device, use_graph=args.graph)
        print(json.dumps({
            "metric": "decode_tokens_per_sec", "model": args.model, "batch": B,
            "prompt_len": args.prompt_len, "value": round(tps, 1),
            "ms_per_step": round(ms, 2),
            "quant": args.quant or "bf16", "hipgraph": bool(args.graph),
            "device_sched": bool(args.device_sched),
            "cachekv": args.cachekv,
        }))


if __name__ == "__main__":
    main()
