// Fused rotary position embedding (Llama rotate-half convention) — gfx950.
//
// Replaces the reference's fused_rotary_position_embedding.
// q: [B, S, Hq, D], k: [B, S, Hk, D] bf16; cos/sin: [S, D] fp32 host-built
// tables (guide Appendix B: precompute trig on host, never sinf on device).
//
//   out[..., i]       = x[i] * cos[i] - x[i + D/2] * sin[i]        (i < D/2)
//   out[..., i+D/2]   = x[i+D/2] * cos[i+D/2] + x[i] * sin[i+D/2]
// backward = rotation by -theta (sign flip on sin).
#include "common.h"

// one thread handles 4 (i, i+D/2) pairs => 8 bf16 loads/stores, coalesced
// within the half-rows.  Layout: token-row = (b*S + s)*H*D + h*D.
__global__ void rope_kernel(
    const ushort_t* __restrict__ x, ushort_t* __restrict__ out,
    const float* __restrict__ cos_t, const float* __restrict__ sin_t,
    long long n_rows,  // B*S*H (q and k launched separately)
    int H, int D, int S, float sign) {
    int half = D / 2;
    long long total = n_rows * half;  // one work-item per rotation pair
    for (long long idx = (long long)blockIdx.x * blockDim.x * 4 + threadIdx.x * 4;
         idx < total; idx += (long long)gridDim.x * blockDim.x * 4) {
        long long row = idx / half;
        int i = (int)(idx % half);
        if (i + 4 > half) continue;  // tails handled by alignment (D%8==0)
        long long s = (row / H) % S;
        const ushort_t* xr = x + row * D;
        ushort_t* outr = out + row * D;
        const float* cr = cos_t + s * D;
        const float* sr = sin_t + s * D;

        short4v x1 = *reinterpret_cast<const short4v*>(xr + i);
        short4v x2 = *reinterpret_cast<const short4v*>(xr + i + half);
        f32x4 c1 = *reinterpret_cast<const f32x4*>(cr + i);
        f32x4 s1 = *reinterpret_cast<const f32x4*>(sr + i);
        f32x4 c2 = *reinterpret_cast<const f32x4*>(cr + i + half);
        f32x4 s2 = *reinterpret_cast<const f32x4*>(sr + i + half);
        short4v o1, o2;
#pragma unroll
        for (int j = 0; j < 4; j++) {
            float a = bf16_to_f32((ushort_t)x1[j]);
            float b = bf16_to_f32((ushort_t)x2[j]);
            o1[j] = (short)f32_to_bf16(a * c1[j] - sign * b * s1[j]);
            o2[j] = (short)f32_to_bf16(b * c2[j] + sign * a * s2[j]);
        }
        *reinterpret_cast<short4v*>(outr + i) = o1;
        *reinterpret_cast<short4v*>(outr + i + half) = o2;
    }
}

void launch_rope(const void* x, void* out, const float* cos_t, const float* sin_t,
                 long long n_rows, int H, int D, int S, bool backward,
                 hipStream_t stream) {
    long long total = n_rows * (D / 2) / 4;
    int grid = memgrid(total, 256);
    hipLaunchKernelGGL(rope_kernel, dim3(grid), dim3(256), 0, stream,
                       (const ushort_t*)x, (ushort_t*)out, cos_t, sin_t,
                       n_rows, H, D, S, backward ? -1.0f : 1.0f);
}
