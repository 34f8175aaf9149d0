"""Offline TunableOp GEMM tuning for the training-step shapes.

Tunes every hipBLASLt/rocBLAS GEMM that appears in one Llama-3-8B training
step at the bench shape (micro-batch 8 x seq 4096 = 32768 tokens): forward,
dgrad and wgrad for qkv / o / gate_up / down / lm_head.  Results are written
to the CSV named by PYTORCH_TUNABLEOP_FILENAME (flushed at exit); commit the
merged file as paddlenlp_amd/ops/tunableop_gfx950.csv and bench.py loads it
read-only via torch.cuda.tunable.read_file().

Run (GPU box):
  PYTORCH_TUNABLEOP_ENABLED=1 PYTORCH_TUNABLEOP_TUNING=1 \
  PYTORCH_TUNABLEOP_FILENAME=gpurun_out/tunableop.csv \
  python tools/tune_gemms.py [--seed existing.csv] [--iters 10]
"""
from __future__ import annotations

import argparse
import os
import sys
import time

import torch
import torch.nn.functional as F
import torch.cuda.tunable as tunable

# (out_features, in_features) of every linear in the step; tokens = B*S
LLAMA3_8B_LINEARS = [
    (6144, 4096),     # fused qkv (32 q + 8 k + 8 v heads, d=128)
    (4096, 4096),     # o_proj
    (28672, 4096),    # fused gate_up
    (4096, 14336),    # down_proj
    (128256, 4096),   # lm head
]


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--tokens", type=int, default=32768)
    ap.add_argument("--seed", type=str, default=None,
                    help="existing results CSV to pre-load (those shapes skip re-tuning)")
    ap.add_argument("--iters", type=int, default=10)
    ap.add_argument("--duration-ms", type=int, default=50)
    args = ap.parse_args()

    assert tunable.is_enabled(), "set PYTORCH_TUNABLEOP_ENABLED=1"
    tunable.set_max_tuning_iterations(args.iters)
    tunable.set_max_tuning_duration(args.duration_ms)
    if args.seed and os.path.exists(args.seed):
        tunable.read_file(args.seed)
        print(f"[tune] seeded {len(tunable.get_results())} results from {args.seed}")

    dev = torch.device("cuda:0")
    t0 = time.time()
    for out_f, in_f in LLAMA3_8B_LINEARS:
        x = torch.randn(args.tokens, in_f, device=dev, dtype=torch.bfloat16,
                        requires_grad=True)
        w = torch.randn(out_f, in_f, device=dev, dtype=torch.bfloat16,
                        requires_grad=True)
        y = F.linear(x, w)           # tn fwd
        y.backward(torch.randn_like(y))  # nn dgrad + nt wgrad
        torch.cuda.synchronize()
        print(f"[tune] ({out_f},{in_f}) done at {time.time()-t0:.0f}s, "
              f"{len(tunable.get_results())} results", flush=True)
        del x, w, y
        torch.cuda.empty_cache()
    print(f"[tune] total {time.time()-t0:.0f}s; results flush to "
          f"{os.environ.get('PYTORCH_TUNABLEOP_FILENAME')} at exit")


if __name__ == "__main__":
    sys.exit(main())
