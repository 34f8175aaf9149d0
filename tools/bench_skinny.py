"""skinny_gemm vs hipBLASLt at the decode shapes (M=64)."""
import sys, os, time
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from paddlenlp_amd.ops.functional import _load_extension
C = _load_extension()

def bench(fn, reps=50):
    for _ in range(5): fn()
    torch.cuda.synchronize(); t0 = time.perf_counter()
    for _ in range(reps): fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / reps * 1e6

M = 64
for (N, K) in [(6144, 4096), (4096, 4096), (28672, 4096), (4096, 14336), (128256, 4096)]:
    x = torch.randn(M, K, device="cuda", dtype=torch.bfloat16)
    w = torch.randn(N, K, device="cuda", dtype=torch.bfloat16)
    ref = (x.float() @ w.float().t())
    best = (1e9, None)
    for ks in (2, 4, 8, 16):
        y = C.skinny_gemm(x, w, ks)
        err = (y.float() - ref).abs().max().item() / ref.abs().max().item()
        t = bench(lambda: C.skinny_gemm(x, w, ks))
        if t < best[0]: best = (t, ks)
        print(f"N={N:6d} K={K:6d} ks={ks:2d}: {t:7.1f}us relerr {err:.4f}")
    t_blas = bench(lambda: x @ w.t())
    print(f"N={N:6d} K={K:6d}: hipBLASLt {t_blas:7.1f}us | best skinny {best[0]:.1f}us (ks={best[1]})")
