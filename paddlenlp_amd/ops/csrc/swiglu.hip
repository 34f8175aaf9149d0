// SwiGLU activation — gfx950, bf16, fused over the [gate | up] layout the
// fused gate_up projection produces (reference: paddle `swiglu` fused op,
// SURVEY §2.9; llama/fusion_ops.py swiglu call).
//
// fwd:  y = silu(g) * u,      x = [N, 2I] with g = x[:, :I], u = x[:, I:]
// bwd:  dg = dy * u * silu'(g);  du = dy * silu(g)
//       silu(g) = g * sigmoid(g); silu'(g) = sig(g) * (1 + g * (1 - sig(g)))
#include "common.h"

__global__ void swiglu_fwd_kernel(
    const ushort_t* __restrict__ x, ushort_t* __restrict__ y,
    long long N, int I) {
    long long total = N * (long long)(I / 8);
    for (long long idx = (long long)blockIdx.x * blockDim.x + threadIdx.x;
         idx < total; idx += (long long)gridDim.x * blockDim.x) {
        long long row = idx / (I / 8);
        int col = (int)(idx % (I / 8)) * 8;
        const ushort_t* g = x + row * 2 * I + col;
        const ushort_t* u = g + I;
        short8v gv = *reinterpret_cast<const short8v*>(g);
        short8v uv = *reinterpret_cast<const short8v*>(u);
        short8v out;
#pragma unroll
        for (int j = 0; j < 8; j++) {
            float gf = bf16_to_f32((ushort_t)gv[j]);
            float uf = bf16_to_f32((ushort_t)uv[j]);
            float sig = 1.0f / (1.0f + __expf(-gf));
            out[j] = (short)f32_to_bf16(gf * sig * uf);
        }
        *reinterpret_cast<short8v*>(y + row * I + col) = out;
    }
}

__global__ void swiglu_bwd_kernel(
    const ushort_t* __restrict__ dy, const ushort_t* __restrict__ x,
    ushort_t* __restrict__ dx, long long N, int I) {
    long long total = N * (long long)(I / 8);
    for (long long idx = (long long)blockIdx.x * blockDim.x + threadIdx.x;
         idx < total; idx += (long long)gridDim.x * blockDim.x) {
        long long row = idx / (I / 8);
        int col = (int)(idx % (I / 8)) * 8;
        const ushort_t* g = x + row * 2 * I + col;
        const ushort_t* u = g + I;
        short8v gv = *reinterpret_cast<const short8v*>(g);
        short8v uv = *reinterpret_cast<const short8v*>(u);
        short8v dv = *reinterpret_cast<const short8v*>(dy + row * I + col);
        short8v dg, du;
#pragma unroll
        for (int j = 0; j < 8; j++) {
            float gf = bf16_to_f32((ushort_t)gv[j]);
            float uf = bf16_to_f32((ushort_t)uv[j]);
            float dyf = bf16_to_f32((ushort_t)dv[j]);
            float sig = 1.0f / (1.0f + __expf(-gf));
            float silu = gf * sig;
            dg[j] = (short)f32_to_bf16(dyf * uf * sig * (1.0f + gf * (1.0f - sig)));
            du[j] = (short)f32_to_bf16(dyf * silu);
        }
        ushort_t* dgp = dx + row * 2 * I + col;
        *reinterpret_cast<short8v*>(dgp) = dg;
        *reinterpret_cast<short8v*>(dgp + I) = du;
    }
}

void launch_swiglu_fwd(const void* x, void* y, long long N, int I, hipStream_t stream) {
    int grid = memgrid(N * (long long)(I / 8), 256);
    hipLaunchKernelGGL(swiglu_fwd_kernel, dim3(grid), dim3(256), 0, stream,
                       (const ushort_t*)x, (ushort_t*)y, N, I);
}

void launch_swiglu_bwd(const void* dy, const void* x, void* dx, long long N, int I,
                       hipStream_t stream) {
    int grid = memgrid(N * (long long)(I / 8), 256);
    hipLaunchKernelGGL(swiglu_bwd_kernel, dim3(grid), dim3(256), 0, stream,
                       (const ushort_t*)dy, (const ushort_t*)x, (ushort_t*)dx, N, I);
}
