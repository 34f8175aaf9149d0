"""Within-probe A/B microbenchmark for the flash-attention kernels.

Guide rule #13: cross-run/cross-box deltas under 10% are noise — variants
must be timed interleaved in ONE process on ONE box.
"""
import argparse
import sys
import os
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

from paddlenlp_amd.ops.functional import _load_extension

VARIANTS = {6: "v2-32x32", 7: "v1-linear", 5: "MF1+pipe"}


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--B", type=int, default=8)
    p.add_argument("--S", type=int, default=4096)
    p.add_argument("--Hq", type=int, default=32)
    p.add_argument("--Hk", type=int, default=8)
    p.add_argument("--D", type=int, default=128)
    p.add_argument("--reps", type=int, default=10)
    args = p.parse_args()

    C = _load_extension()
    torch.manual_seed(0)
    dev = "cuda:0"
    q = torch.randn(args.B, args.S, args.Hq, args.D, device=dev, dtype=torch.bfloat16)
    k = torch.randn(args.B, args.S, args.Hk, args.D, device=dev, dtype=torch.bfloat16)
    v = torch.randn(args.B, args.S, args.Hk, args.D, device=dev, dtype=torch.bfloat16)

    # causal attention matmul FLOPs (0.5 visibility): 2 matmuls x 2 flops
    flops = 0.5 * 2 * 2 * args.B * args.Hq * args.S * args.S * args.D

    # correctness cross-check between variants first
    ref_o, ref_lse = C.flash_attn_fwd_ex(q, k, v, True, 7)
    for var in VARIANTS:
        o, lse = C.flash_attn_fwd_ex(q, k, v, True, var)
        err = (o.float() - ref_o.float()).abs().max().item()
        print(f"  variant {var} vs v1 max err: {err:.4f}")
        assert err < 3e-2, (var, err)

    results = {v: [] for v in VARIANTS}
    for rep in range(args.reps):
        for var in VARIANTS:  # interleaved
            torch.cuda.synchronize()
            t0 = time.perf_counter()
            C.flash_attn_fwd_ex(q, k, v, True, var)
            torch.cuda.synchronize()
            results[var].append(time.perf_counter() - t0)

    print(f"shape B{args.B} S{args.S} Hq{args.Hq} Hk{args.Hk} D{args.D}, {args.reps} reps:")
    for var, name in VARIANTS.items():
        ts = sorted(results[var])[1:-1] or results[var]  # trim outliers
        mean = sum(ts) / len(ts)
        print(f"  fwd variant {var} ({name:<11}): {mean*1e3:8.2f} ms  {flops/mean/1e12:7.1f} TF/s")

    # backward variants A/B (2=v2, 3=v1 linear)
    o, lse = C.flash_attn_fwd_ex(q, k, v, True, 0)
    do = torch.randn_like(o)
    ref_dq, ref_dk, ref_dv = C.flash_attn_bwd_ex(do, q, k, v, o, lse, True, 3)
    dq2, dk2, dv2 = C.flash_attn_bwd_ex(do, q, k, v, o, lse, True, 2)
    for nm, a, b in (("dq", ref_dq, dq2), ("dk", ref_dk, dk2), ("dv", ref_dv, dv2)):
        err = (a.float() - b.float()).abs().max().item()
        print(f"  bwd v2 vs v1 {nm} max err: {err:.4f}")
    bres = {2: [], 3: []}
    for rep in range(args.reps):
        for var in (2, 3):
            torch.cuda.synchronize()
            t0 = time.perf_counter()
            C.flash_attn_bwd_ex(do, q, k, v, o, lse, True, var)
            torch.cuda.synchronize()
            bres[var].append(time.perf_counter() - t0)
    bwd_flops = flops * 3.5  # 7 matmuls vs fwd's 2
    for var, name in ((2, "v2-32x32"), (3, "v1-linear")):
        ts = sorted(bres[var])[1:-1] or bres[var]
        mean = sum(ts) / len(ts)
        print(f"  bwd variant {var} ({name:<9}): {mean*1e3:8.2f} ms  {bwd_flops/mean/1e12:7.1f} TF/s")


if __name__ == "__main__":
    main()
