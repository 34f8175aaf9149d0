// Weight-only int8 fused dequant GEMV/GEMM for decode — gfx950.
//
// Replaces the reference's weight_only_linear int8 runtime (SURVEY §2.9
// "weight_only_linear (int4/int8 gemm)") with a CDNA4-native kernel for the
// skinny decode shapes: y[M, N] = x[M, K] @ dequant(Wq[N, K])^T * scale[N].
// M is the decode batch (1..16 here; larger M goes to hipBLASLt after
// dequant), so the kernel is memory-bound on the int8 weight stream: each
// workgroup owns 32 output rows, waves stream 16-byte int8 chunks of Wq,
// dequantize in-register and dot against an x chunk staged in LDS as fp32
// (same-address broadcast reads are conflict-free).
#include "common.h"
#include <hip/hip_fp8.h>

#define WG_BLOCK 256
#define ROWS_PER_WG 32   // output rows (N) per workgroup, 8 lanes per row

typedef __attribute__((ext_vector_type(4))) int int4v;

template <int MAX_M>
__global__ __launch_bounds__(WG_BLOCK) void wint8_gemv_kernel(
    const ushort_t* __restrict__ x, const signed char* __restrict__ wq,
    const float* __restrict__ scale, ushort_t* __restrict__ y,
    int M, int N, int K, int KC) {
    extern __shared__ float x_lds[];  // [M, KC] fp32 chunk

    const int tid = threadIdx.x;
    const int n0 = blockIdx.x * ROWS_PER_WG;
    const int lane_in_row = tid & 7;       // 8 lanes per output row
    const int row_local = tid >> 3;        // 0..31
    const int n = n0 + row_local;
    const bool live = n < N;
    const signed char* wrow = wq + (long long)n * K;

    float acc[MAX_M];
#pragma unroll
    for (int m = 0; m < MAX_M; m++) acc[m] = 0.f;

    for (int kc = 0; kc < K; kc += KC) {
        const int kend = min(K, kc + KC);
        const int klen = kend - kc;
        // stage this K-chunk of x (fp32) — all threads participate
        for (int i = tid; i < M * klen; i += WG_BLOCK) {
            int m = i / klen, kk = i % klen;
            x_lds[m * KC + kk] = bf16_to_f32(x[(long long)m * K + kc + kk]);
        }
        __syncthreads();

        if (live) {
            for (int k = lane_in_row * 16; k < klen; k += 8 * 16) {
                int4v w4 = *reinterpret_cast<const int4v*>(wrow + kc + k);
#pragma unroll
                for (int c = 0; c < 4; c++) {
                    unsigned int word = (unsigned int)w4[c];
#pragma unroll
                    for (int j = 0; j < 4; j++) {
                        float wf = (float)(signed char)((word >> (8 * j)) & 0xFF);
                        int kk = k + c * 4 + j;
#pragma unroll
                        for (int m = 0; m < MAX_M; m++) {
                            if (m >= M) break;
                            acc[m] += wf * x_lds[m * KC + kk];
                        }
                    }
                }
            }
        }
        __syncthreads();
    }

    if (!live) return;
#pragma unroll
    for (int m = 0; m < MAX_M; m++) {
        if (m >= M) break;
        float v = acc[m];
#pragma unroll
        for (int off = 1; off < 8; off <<= 1) v += __shfl_xor(v, off, 64);
        if (lane_in_row == 0)
            y[(long long)m * N + n] = f32_to_bf16(v * scale[n]);
    }
}

void launch_wint8_gemv(const void* x, const void* wq, const float* scale, void* y,
                       int M, int N, int K, hipStream_t stream) {
    dim3 grid((N + ROWS_PER_WG - 1) / ROWS_PER_WG);
    // LDS budget ~96 KB for the fp32 x chunk; KC multiple of 128
    int KC = (96 * 1024 / 4) / (M > 0 ? M : 1);
    KC = (KC / 128) * 128;
    if (KC > K) KC = K;
    if (KC < 128) KC = 128;
    size_t lds = (size_t)M * KC * sizeof(float);
    if (M <= 4)
        hipLaunchKernelGGL(wint8_gemv_kernel<4>, grid, dim3(WG_BLOCK), lds, stream,
                           (const ushort_t*)x, (const signed char*)wq, scale,
                           (ushort_t*)y, M, N, K, KC);
    else if (M <= 8)
        hipLaunchKernelGGL(wint8_gemv_kernel<8>, grid, dim3(WG_BLOCK), lds, stream,
                           (const ushort_t*)x, (const signed char*)wq, scale,
                           (ushort_t*)y, M, N, K, KC);
    else
        hipLaunchKernelGGL(wint8_gemv_kernel<16>, grid, dim3(WG_BLOCK), lds, stream,
                           (const ushort_t*)x, (const signed char*)wq, scale,
                           (ushort_t*)y, M, N, K, KC);
}

// ---------------------------------------------------------------------------
// fused rowwise fp8 activation quant: one wave per row computes the row
// absmax and casts to OCP e4m3fn in a single launch (the torch-op chain
// costs ~6 kernel launches ≈ 40 us at decode shapes; this is one).
// Feeds torch._scaled_mm rowwise-scaled fp8 GEMMs (reference fp8 A8W8
// path, fused_transformer_layers.py FP8 variant).
// ---------------------------------------------------------------------------
#define FP8_E4M3_MAX 448.0f

__global__ __launch_bounds__(256) void fp8_rowwise_quant_kernel(
    const ushort_t* __restrict__ x,   // [R, K] bf16
    unsigned char* __restrict__ y,    // [R, K] fp8 e4m3fn bit patterns
    float* __restrict__ scales,       // [R]
    long long R, int K) {
    const int wave_in_block = threadIdx.x >> 6;
    const int lane = threadIdx.x & 63;
    const long long row = (long long)blockIdx.x * 4 + wave_in_block;
    if (row >= R) return;
    const ushort_t* xr = x + row * K;
    float amax = 0.f;
    for (int k = lane * 8; k < K; k += 64 * 8) {
        short8v v = *reinterpret_cast<const short8v*>(xr + k);
#pragma unroll
        for (int j = 0; j < 8; j++)
            amax = fmaxf(amax, fabsf(bf16_to_f32((ushort_t)v[j])));
    }
    amax = wave_reduce_max(amax);
    const float sc = fmaxf(amax, 1e-8f) / FP8_E4M3_MAX;
    const float inv = 1.0f / sc;
    unsigned char* yr = y + row * K;
    for (int k = lane * 8; k < K; k += 64 * 8) {
        short8v v = *reinterpret_cast<const short8v*>(xr + k);
        unsigned char out8[8];
#pragma unroll
        for (int j = 0; j < 8; j++) {
            float f = bf16_to_f32((ushort_t)v[j]) * inv;
            f = fminf(fmaxf(f, -FP8_E4M3_MAX), FP8_E4M3_MAX);
            __hip_fp8_e4m3 h(f);
            out8[j] = h.__x;
        }
        *reinterpret_cast<unsigned long long*>(yr + k) =
            *reinterpret_cast<unsigned long long*>(out8);
    }
    if (lane == 0) scales[row] = sc;
}

void launch_fp8_rowwise_quant(const void* x, void* y, float* scales,
                              long long R, int K, hipStream_t stream) {
    long long blocks = (R + 3) / 4;
    hipLaunchKernelGGL(fp8_rowwise_quant_kernel, dim3((unsigned)blocks), dim3(256),
                       0, stream, (const ushort_t*)x, (unsigned char*)y, scales, R, K);
}
