import torch, time

def bench(fn, reps=50):
    for _ in range(5): fn()
    torch.cuda.synchronize(); t0 = time.perf_counter()
    for _ in range(reps): fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / reps * 1e6

M = 64
for (N, K) in [(4096, 4096), (28672, 4096), (4096, 14336)]:
    x = torch.randn(M, K, device="cuda", dtype=torch.bfloat16)
    w = torch.randn(N, K, device="cuda", dtype=torch.bfloat16)
    wf = w.to(torch.float8_e4m3fn)
    sa = torch.ones((), device="cuda", dtype=torch.float32)
    sb = torch.ones((), device="cuda", dtype=torch.float32)
    # fixed dynamic per-tensor (fp32 scale)
    def dynq():
        s = (x.float().abs().amax().clamp(min=1e-8) / 448.0)
        x8 = (x.float() / s).clamp(-448, 448).to(torch.float8_e4m3fn)
        return torch._scaled_mm(x8, wf.t(), scale_a=s, scale_b=sb, out_dtype=torch.bfloat16)
    try:
        t_dyn = bench(dynq)
    except Exception as e:
        t_dyn = float("nan"); print("dyn fail:", repr(e)[:120])
    # rowwise scales
    sa_r = torch.ones(M, 1, device="cuda", dtype=torch.float32)
    sb_c = torch.ones(1, N, device="cuda", dtype=torch.float32)
    x8 = x.to(torch.float8_e4m3fn)
    try:
        t_row = bench(lambda: torch._scaled_mm(x8, wf.t(), scale_a=sa_r, scale_b=sb_c, out_dtype=torch.bfloat16))
    except Exception as e:
        t_row = float("nan"); print("rowwise fail:", repr(e)[:120])
    # quant-op overhead alone
    def quant_only():
        s = (x.float().abs().amax(dim=1, keepdim=True).clamp(min=1e-8) / 448.0)
        return (x.float() / s).clamp(-448, 448).to(torch.float8_e4m3fn), s
    t_q = bench(quant_only)
    print(f"N={N:6d} K={K:6d}: dyn(per-tensor) {t_dyn:7.1f}us  rowwise-mm {t_row:7.1f}us  act-quant-alone {t_q:7.1f}us")
