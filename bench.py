"""Flagship training benchmark: Llama-3-8B bf16 pretrain on MI355X.

Driver contract (BASELINE.json): measures tokens/sec (whole node) on the
Llama-3-8B config with synthetic data and random-init weights.
  python bench.py --gpus N --steps K --warmup W
N>1 is launched by the driver via torch.distributed.run (one rank per GPU,
RCCL); parallelism is data parallel + ZeRO stage-2 sharding (the reference's
headline "sd8_stage2"-style recipe).

Prints ONE JSON line from rank 0 with the whole-job aggregate.
"""
from __future__ import annotations

import argparse
import json
import os
import sys
import time

import torch
import torch.distributed as dist

sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))

from paddlenlp_amd.parallel.topology import init_parallel_env
from paddlenlp_amd.parallel.data_parallel import broadcast_parameters
from paddlenlp_amd.parallel.zero import ZeroShardedEngine
from paddlenlp_amd.trainer.optimizer import FusedAdamW
from paddlenlp_amd.trainer.trainer_utils import (
    MI355X_BF16_PEAK_FLOPS,
    caculate_llm_flops,
    set_seed,
)
from paddlenlp_amd.transformers import LlamaConfig, LlamaForCausalLM

# Closest published reference number: meta-llama/Llama-2-7b
# pretrain on 8xA100-80G, tp2sd4_stage2, seq 4096 = 3754.73 tokens/card/sec.
BASELINE_TOKENS_PER_CARD = 3754.73

MODELS = {
    "llama3-8b": dict(
        vocab_size=128256, hidden_size=4096, intermediate_size=14336,
        num_hidden_layers=32, num_attention_heads=32, num_key_value_heads=8,
        max_position_embeddings=8192, rope_theta=500000.0,
    ),
    # small config for quick plumbing runs
    "llama-tiny": dict(
        vocab_size=32000, hidden_size=1024, intermediate_size=2816,
        num_hidden_layers=4, num_attention_heads=8, num_key_value_heads=8,
        max_position_embeddings=8192,
    ),
}


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--gpus", type=int, default=1)
    p.add_argument("--steps", type=int, default=10)
    p.add_argument("--warmup", type=int, default=3)
    p.add_argument("--model", type=str, default="llama3-8b")
    p.add_argument("--micro-batch", type=int, default=8)
    p.add_argument("--seq-len", type=int, default=4096)
    p.add_argument("--accum", type=int, default=1)
    p.add_argument("--recompute", action="store_true")
    p.add_argument("--lr", type=float, default=3e-4)
    args = p.parse_args()

    world_size = int(os.environ.get("WORLD_SIZE", "1"))
    rank = int(os.environ.get("RANK", "0"))
    local_rank = int(os.environ.get("LOCAL_RANK", "0"))
    n_gpus = world_size if world_size > 1 else args.gpus

    assert torch.cuda.is_available(), "bench.py requires a GPU"
    torch.cuda.set_device(local_rank)
    device = torch.device("cuda", local_rank)

    # Pre-tuned hipBLASLt/rocBLAS GEMM algo selections for the bench shapes
    # (tools/tune_gemms.py).  Loaded read-only: no runtime tuning cost.
    tuned_csv = os.environ.get(
        "PNLP_TUNABLEOP_CSV",
        os.path.join(os.path.dirname(os.path.abspath(__file__)),
                     "paddlenlp_amd", "ops", "tunableop_gfx950.csv"))
    if os.path.exists(tuned_csv) and os.environ.get("PNLP_TUNABLEOP", "1") == "1":
        import torch.cuda.tunable as tunable
        tunable.enable(True)
        tunable.tuning_enable(False)
        tunable.read_file(tuned_csv)
        if local_rank == 0:
            print(f"[bench] loaded {len(tunable.get_results())} tuned GEMM "
                  f"algos from {tuned_csv}", file=sys.stderr)

    # data-parallel + ZeRO stage2 over the sharding axis (reference recipe)
    use_sharding = world_size > 1
    topo = init_parallel_env(
        dp_degree=1,
        sharding_degree=world_size if use_sharding else 1,
    )
    set_seed(42 + rank)

    cfg = LlamaConfig(
        **MODELS[args.model],
        dtype="bfloat16",
        fuse_attention_qkv=True,
        fuse_attention_ffn=True,
        recompute=args.recompute,
        use_flash_attention=True,
        use_fused_rms_norm=True,
        use_fused_rope=True,
        use_fused_swiglu=True,
        # chunked head+CE (no [tokens, vocab] logits tensor); A/B via env
        use_fused_linear_cross_entropy=(
            os.environ.get("PNLP_FUSED_CE", "0") == "1"),
    )
    if rank == 0:
        n_params = (
            cfg.vocab_size * cfg.hidden_size * 2
            + cfg.num_hidden_layers * (
                cfg.hidden_size * (cfg.hidden_size + 2 * cfg.num_key_value_heads * cfg.head_dim)
                + cfg.hidden_size * cfg.hidden_size
                + 3 * cfg.hidden_size * cfg.intermediate_size
                + 2 * cfg.hidden_size
            )
            + cfg.hidden_size
        )
        print(f"[bench] building {args.model} (~{n_params/1e9:.2f}B params) on {device}",
              file=sys.stderr)

    model = LlamaForCausalLM.from_config(cfg, dtype=torch.bfloat16, device=device)
    model.train()

    if use_sharding:
        broadcast_parameters(model, topo.sharding_parallel_group)

    decay, no_decay = [], []
    for name, prm in model.named_parameters():
        prm.param_name = name
        (no_decay if (name.endswith("bias") or "norm" in name) else decay).append(prm)
    optimizer = FusedAdamW(
        [{"params": decay, "weight_decay": 0.1},
         {"params": no_decay, "weight_decay": 0.0}],
        lr=args.lr, betas=(0.9, 0.95), eps=1e-8, master_weights=True,
    )
    zero = None
    if use_sharding:
        zero = ZeroShardedEngine(model, optimizer, stage=2,
                                 group=topo.sharding_parallel_group, bucket_mb=256)
        # backward-overlapped grad reduce (PNLP_NO_OVERLAP=1 disables)
        if os.environ.get("PNLP_NO_OVERLAP", "0") != "1":
            zero.enable_overlap_comm()

    # synthetic data of the benchmark shape (no network: random tokens)
    g = torch.Generator(device="cpu").manual_seed(1234 + rank)
    batches = []
    for _ in range(4):
        ids = torch.randint(0, cfg.vocab_size, (args.micro_batch, args.seq_len + 1), generator=g)
        batches.append({
            "input_ids": ids[:, :-1].to(device),
            "labels": ids[:, 1:].contiguous().to(device),
        })

    def one_step(step_idx: int):
        if zero is not None:
            # flat-bucket engine: zero the persistent grad buffers and
            # re-attach views so backward accumulates in place
            zero.zero_grad()
        else:
            optimizer.zero_grad(set_to_none=True)
        for a in range(args.accum):
            batch = batches[(step_idx * args.accum + a) % len(batches)]
            if zero is not None and getattr(zero, "_overlap", False) \
                    and a == args.accum - 1:
                zero.overlap_active = True  # reduce buckets during backward
            loss, _ = model(**batch)
            (loss / args.accum).backward()
        if zero is not None:
            zero.reduce_gradients_and_step_pre()
        optimizer.step()
        if zero is not None:
            zero.step_post()
        return loss

    def barrier_sync():
        if dist.is_initialized():
            dist.barrier()
        torch.cuda.synchronize()

    for i in range(args.warmup):
        loss = one_step(i)
    barrier_sync()

    t0 = time.perf_counter()
    for i in range(args.steps):
        loss = one_step(args.warmup + i)
    barrier_sync()
    elapsed = time.perf_counter() - t0

    # MAX over ranks
    if dist.is_initialized():
        t = torch.tensor([elapsed], device=device)
        dist.all_reduce(t, op=dist.ReduceOp.MAX)
        elapsed = t.item()

    tokens_per_step = args.micro_batch * args.seq_len * args.accum * n_gpus
    total_tokens = tokens_per_step * args.steps
    tokens_per_sec = total_tokens / elapsed
    ms_per_step = elapsed / args.steps * 1000

    flops_per_step_per_dev = caculate_llm_flops(
        cfg.hidden_size, cfg.intermediate_size, cfg.num_hidden_layers,
        cfg.vocab_size, args.seq_len,
        batch_size=args.micro_batch * args.accum, recompute=args.recompute,
    )
    mfu = flops_per_step_per_dev * args.steps / elapsed / MI355X_BF16_PEAK_FLOPS * 100

    if rank == 0:
        result = {
            "metric": "tokens_per_sec_whole_node_llama3_8b_pretrain",
            "value": round(tokens_per_sec, 1),
            "unit": "tokens/s",
            "n_gpus": n_gpus,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": round(ms_per_step, 1),
            "higher_is_better": True,
            "scaling": "weak",
            "vs_baseline": round(tokens_per_sec / (BASELINE_TOKENS_PER_CARD * n_gpus), 3),
            "dtype": "bf16",
            "data": "synthetic",
            "mfu_percent": round(mfu, 2),
            "loss": round(loss.item(), 4),
            "config": {
                "model": args.model,
                "global_batch": args.micro_batch * args.accum * n_gpus,
                "seq_len": args.seq_len,
                "parallelism": f"dp{n_gpus}_zero2" if use_sharding else "single",
                "recompute": args.recompute,
            },
        }
        print(json.dumps(result))

    if dist.is_initialized():
        dist.destroy_process_group()


if __name__ == "__main__":
    main()
