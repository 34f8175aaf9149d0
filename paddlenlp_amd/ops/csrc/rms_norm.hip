// RMSNorm forward/backward — gfx950, bf16 in / bf16 out, fp32 math.
//
// Replaces the reference's fused_rms_norm (paddlenlp fused op, SURVEY §2.9
// "training-side fused ops").  Memory-bound: vectorized short8 loads
// (guide G13), one workgroup per row bucket, fp32 accumulation.
//
// fwd:  y = x * rsqrt(mean(x^2) + eps) * w          (saves invrms per row)
// bwd:  dx = invrms * (dy*w - x * invrms^2 * mean(dy*w*x))
//       dw = sum_rows(dy * x * invrms)   (two-pass partial reduction)
#include "common.h"

#define RMS_BLOCK 256

// one block per row; H must be a multiple of 8 (hidden sizes are)
template <typename T_VEC>
__global__ void rms_norm_fwd_kernel(
    const ushort_t* __restrict__ x, const ushort_t* __restrict__ w,
    ushort_t* __restrict__ y, float* __restrict__ invrms,
    int H, float eps, long long rows) {
    __shared__ float scratch[16];
    for (long long row = blockIdx.x; row < rows; row += gridDim.x) {
        const ushort_t* xr = x + row * H;
        ushort_t* yr = y + row * H;
        float ss = 0.f;
        for (int i = threadIdx.x * 8; i < H; i += blockDim.x * 8) {
            short8v v = *reinterpret_cast<const short8v*>(xr + i);
#pragma unroll
            for (int j = 0; j < 8; j++) {
                float f = bf16_to_f32((ushort_t)v[j]);
                ss += f * f;
            }
        }
        ss = block_reduce_sum(ss, scratch);
        float inv = rsqrtf(ss / H + eps);
        if (threadIdx.x == 0) invrms[row] = inv;
        for (int i = threadIdx.x * 8; i < H; i += blockDim.x * 8) {
            short8v v = *reinterpret_cast<const short8v*>(xr + i);
            short8v wv = *reinterpret_cast<const short8v*>(w + i);
            short8v out;
#pragma unroll
            for (int j = 0; j < 8; j++) {
                float f = bf16_to_f32((ushort_t)v[j]) * inv * bf16_to_f32((ushort_t)wv[j]);
                out[j] = (short)f32_to_bf16(f);
            }
            *reinterpret_cast<short8v*>(yr + i) = out;
        }
        __syncthreads();
    }
}

// dx per row + per-block partial dw accumulation in fp32 workspace
__global__ void rms_norm_bwd_dx_kernel(
    const ushort_t* __restrict__ dy, const ushort_t* __restrict__ x,
    const ushort_t* __restrict__ w, const float* __restrict__ invrms,
    ushort_t* __restrict__ dx, float* __restrict__ dw_partial,
    int H, long long rows) {
    __shared__ float scratch[16];
    float* dwp = dw_partial + (long long)blockIdx.x * H;
    for (long long row = blockIdx.x; row < rows; row += gridDim.x) {
        const ushort_t* dyr = dy + row * H;
        const ushort_t* xr = x + row * H;
        ushort_t* dxr = dx + row * H;
        float inv = invrms[row];
        // pass 1: c = sum(dy * w * x) (fp32)
        float c = 0.f;
        for (int i = threadIdx.x * 8; i < H; i += blockDim.x * 8) {
            short8v dv = *reinterpret_cast<const short8v*>(dyr + i);
            short8v xv = *reinterpret_cast<const short8v*>(xr + i);
            short8v wv = *reinterpret_cast<const short8v*>(w + i);
#pragma unroll
            for (int j = 0; j < 8; j++) {
                c += bf16_to_f32((ushort_t)dv[j]) * bf16_to_f32((ushort_t)wv[j]) * bf16_to_f32((ushort_t)xv[j]);
            }
        }
        c = block_reduce_sum(c, scratch);
        float k = c * inv * inv / H;
        // pass 2: dx + dw partials
        for (int i = threadIdx.x * 8; i < H; i += blockDim.x * 8) {
            short8v dv = *reinterpret_cast<const short8v*>(dyr + i);
            short8v xv = *reinterpret_cast<const short8v*>(xr + i);
            short8v wv = *reinterpret_cast<const short8v*>(w + i);
            short8v out;
#pragma unroll
            for (int j = 0; j < 8; j++) {
                float dyf = bf16_to_f32((ushort_t)dv[j]);
                float xf = bf16_to_f32((ushort_t)xv[j]);
                float wf = bf16_to_f32((ushort_t)wv[j]);
                float dxf = inv * (dyf * wf - xf * k);
                out[j] = (short)f32_to_bf16(dxf);
                dwp[i + j] += dyf * xf * inv;
            }
            *reinterpret_cast<short8v*>(dxr + i) = out;
        }
        __syncthreads();
    }
}

// reduce dw partials [P, H] -> dw32 [H] fp32.  Parallelized over BOTH axes:
// grid.x covers columns, grid.y slices the P partials; one atomicAdd per
// (column, slice) — ~grid.y adders per address, negligible contention (G12).
__global__ void rms_norm_bwd_dw_reduce_kernel(
    const float* __restrict__ dw_partial, float* __restrict__ dw32,
    int H, int P) {
    int i = blockIdx.x * blockDim.x + threadIdx.x;
    if (i >= H) return;
    int p0 = blockIdx.y;
    float acc = 0.f;
    for (int p = p0; p < P; p += gridDim.y) acc += dw_partial[(long long)p * H + i];
    atomicAdd(&dw32[i], acc);
}

__global__ void rms_norm_bwd_dw_cast_kernel(
    const float* __restrict__ dw32, ushort_t* __restrict__ dw, int H) {
    int i = blockIdx.x * blockDim.x + threadIdx.x;
    if (i < H) dw[i] = f32_to_bf16(dw32[i]);
}

// ---- launchers (called from bindings.cpp) ----
void launch_rms_norm_fwd(const void* x, const void* w, void* y, float* invrms,
                         long long rows, int H, float eps, hipStream_t stream) {
    int grid = (int)(rows < 8192 ? rows : 8192);
    hipLaunchKernelGGL(rms_norm_fwd_kernel<short8v>, dim3(grid), dim3(RMS_BLOCK), 0, stream,
                       (const ushort_t*)x, (const ushort_t*)w, (ushort_t*)y, invrms, H, eps, rows);
}

void launch_rms_norm_bwd(const void* dy, const void* x, const void* w,
                         const float* invrms, void* dx, float* dw_partial,
                         float* dw32, void* dw, bool dw_is_bf16,
                         long long rows, int H, int P, hipStream_t stream) {
    hipLaunchKernelGGL(rms_norm_bwd_dx_kernel, dim3(P), dim3(RMS_BLOCK), 0, stream,
                       (const ushort_t*)dy, (const ushort_t*)x, (const ushort_t*)w,
                       invrms, (ushort_t*)dx, dw_partial, H, rows);
    int rgrid = (H + 255) / 256;
    int pslices = P > 64 ? 64 : P;
    hipLaunchKernelGGL(rms_norm_bwd_dw_reduce_kernel, dim3(rgrid, pslices), dim3(256), 0, stream,
                       dw_partial, dw32, H, P);
    if (dw_is_bf16) {
        hipLaunchKernelGGL(rms_norm_bwd_dw_cast_kernel, dim3(rgrid), dim3(256), 0, stream,
                           dw32, (ushort_t*)dw, H);
    }
}
