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
