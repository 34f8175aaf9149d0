// Skinny-M GEMM for decode projections — gfx950.
//
// y[M, N] = x[M, K] @ W[N, K]^T for M <= 64 (decode batch).  hipBLASLt's
// kernels floor at ~19 us for these shapes regardless of byte count
// (measured: tools/probe_qmm.py; TunableOp retuning at M=64 is neutral),
// ~3.5x above the weight-streaming roofline.  This kernel is built around
// that roofline: weights stream ONCE as direct per-lane MFMA A-fragments
// (16-byte loads, L1-absorbed), the x panel stages per K-chunk in
// XOR-swizzled LDS, and the K dimension splits across blocks into fp32
// partials reduced by a second tiny kernel (deterministic; no atomics).
//
// Reference behavior: the reference's decode GEMMs run through
// cublasLt/cutlass with offline tuning (csrc tune_cublaslt_gemm).
//
// MEASURED RESULT (tools/bench_skinny.py, MI355X): even with the whole
// chunk's weight fragments issued up front, this structure loses to
// hipBLASLt at every decode shape (best 20.8 vs 19.1 us at N=K=4096;
// 71.8 vs 36.7 at N=28672) — the per-lane 16-byte fragment stream reaches
// ~3.3 TB/s vs the library's ~6 TB/s, and the x panel re-stage per
// N-block adds up at large N.  The kernel is kept as a correct,
// deterministic split-K GEMM (useful reference / TP-shard shapes) but the
// engine keeps hipBLASLt for decode projections.
#include "common.h"

#define SG_WAVES 4
#define SG_BLOCK (SG_WAVES * 64)
#define SG_NB 128            // N rows per block (32 per wave)
#define SG_CK 256            // K chunk staged in LDS

typedef short8v sg_frag8;
typedef __attribute__((ext_vector_type(4))) float sg_f32x4;

static __device__ __forceinline__ sg_f32x4 sg_mfma16(sg_frag8 a, sg_frag8 b, sg_f32x4 c) {
    return __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, c, 0, 0, 0);
}

template <int LD>
static __device__ __forceinline__ char* sg_swz(ushort_t* base, int row, int col_elem) {
    return reinterpret_cast<char*>(base) +
           (((row * LD + col_elem) * 2) ^ ((row & 7) << 4));
}

// MT = number of 16-row M tiles (M <= MT*16)
template <int MT>
__global__ __launch_bounds__(SG_BLOCK) void skinny_gemm_kernel(
    const ushort_t* __restrict__ x,   // [M, K]
    const ushort_t* __restrict__ w,   // [N, K]
    float* __restrict__ partial,      // [ksplit, M, N]
    int M, int N, int K, int kper) {
    __shared__ ushort_t x_lds[64 * SG_CK];   // 32 KB, swizzled rows

    const int tid = threadIdx.x;
    const int wave = tid >> 6;
    const int lane = tid & 63;
    const int l16 = lane & 15;
    const int lg = lane >> 4;

    const int n0 = blockIdx.x * SG_NB + wave * 32;
    const int z = blockIdx.y;
    const int k0 = z * kper;
    const int k1 = min(K, k0 + kper);
    if (k0 >= K) return;

    sg_f32x4 acc[2][MT];
#pragma unroll
    for (int rt = 0; rt < 2; rt++)
#pragma unroll
        for (int mt = 0; mt < MT; mt++)
#pragma unroll
            for (int r = 0; r < 4; r++) acc[rt][mt][r] = 0.f;

    // staging map: 256 threads cover 64 rows x 256 cols; thread owns one
    // row and a 64-col quarter (8 x short8)
    const int s_row = tid & 63;
    const int s_c = (tid >> 6) * 64;

    for (int kc = k0; kc < k1; kc += SG_CK) {
        // ---- stage x chunk (zero-padded beyond M) ----
#pragma unroll
        for (int cc = 0; cc < 8; cc++) {
            int col = kc + s_c + cc * 8;
            short8v v8 = short8v{0,0,0,0,0,0,0,0};
            if (s_row < M && col < K)
                v8 = *reinterpret_cast<const short8v*>(x + (long long)s_row * K + col);
            *reinterpret_cast<short8v*>(sg_swz<SG_CK>(x_lds, s_row, s_c + cc * 8)) = v8;
        }
        __syncthreads();

        const int kend = min(SG_CK, k1 - kc);
        // issue the ENTIRE chunk's W fragment loads first (16 x 16B per
        // wave in flight) — per-kk loads leave only 32 B outstanding and
        // the weight stream goes latency-bound
        sg_frag8 aw[SG_CK / 32][2];
#pragma unroll
        for (int kk = 0; kk < SG_CK; kk += 32) {
            if (kk < kend) {
#pragma unroll
                for (int rt = 0; rt < 2; rt++) {
                    int n = n0 + rt * 16 + l16;
                    aw[kk / 32][rt] = short8v{0,0,0,0,0,0,0,0};
                    if (n < N)
                        aw[kk / 32][rt] = *reinterpret_cast<const sg_frag8*>(
                            w + (long long)n * K + kc + kk + lg * 8);
                }
            }
        }
#pragma unroll
        for (int kk = 0; kk < SG_CK; kk += 32) {
            if (kk >= kend) break;
            sg_frag8 bx[MT];
#pragma unroll
            for (int mt = 0; mt < MT; mt++)
                bx[mt] = *reinterpret_cast<const sg_frag8*>(
                    sg_swz<SG_CK>(x_lds, mt * 16 + l16, kk + lg * 8));
            __builtin_amdgcn_s_setprio(1);
#pragma unroll
            for (int rt = 0; rt < 2; rt++)
#pragma unroll
                for (int mt = 0; mt < MT; mt++)
                    acc[rt][mt] = sg_mfma16(aw[kk / 32][rt], bx[mt], acc[rt][mt]);
            __builtin_amdgcn_s_setprio(0);
        }
        __syncthreads();
    }

    // partial[z][m][n]: C row = n-in-tile = lg*4 + r, col = m = l16
    float* pz = partial + (long long)z * M * N;
#pragma unroll
    for (int rt = 0; rt < 2; rt++) {
        int n = n0 + rt * 16 + lg * 4;
#pragma unroll
        for (int mt = 0; mt < MT; mt++) {
            int m = mt * 16 + l16;
            if (m >= M) continue;
#pragma unroll
            for (int r = 0; r < 4; r++) {
                if (n + r < N)
                    pz[(long long)m * N + n + r] = acc[rt][mt][r];
            }
        }
    }
}

__global__ void skinny_gemm_reduce_kernel(
    const float* __restrict__ partial, ushort_t* __restrict__ y,
    long long MN, int ksplit) {
    long long i = (long long)blockIdx.x * blockDim.x + threadIdx.x;
    if (i >= MN) return;
    float acc = 0.f;
    for (int z = 0; z < ksplit; z++)
        acc += partial[(long long)z * MN + i];
    y[i] = f32_to_bf16(acc);
}

int skinny_gemm_ksplit(int N, int K, int ksplit_req) {
    int nchunks = (K + SG_CK - 1) / SG_CK;
    int per = (nchunks + ksplit_req - 1) / ksplit_req;
    return (nchunks + per - 1) / per;   // effective splits, all non-empty
}

void launch_skinny_gemm(const void* x, const void* w, float* partial, void* y,
                        int M, int N, int K, int ksplit, hipStream_t stream) {
    int nchunks = (K + SG_CK - 1) / SG_CK;
    int per = (nchunks + ksplit - 1) / ksplit;
    int ks = (nchunks + per - 1) / per;
    int kper = per * SG_CK;
    dim3 grid((N + SG_NB - 1) / SG_NB, ks);
    const int MT = (M + 15) / 16;
#define SG_LAUNCH(MM)                                                        \
    hipLaunchKernelGGL((skinny_gemm_kernel<MM>), grid, dim3(SG_BLOCK), 0,    \
                       stream, (const ushort_t*)x, (const ushort_t*)w,       \
                       partial, M, N, K, kper)
    if (MT <= 1) SG_LAUNCH(1);
    else if (MT == 2) SG_LAUNCH(2);
    else if (MT == 3) SG_LAUNCH(3);
    else SG_LAUNCH(4);
#undef SG_LAUNCH
    long long MN = (long long)M * N;
    int thr = 256;
    hipLaunchKernelGGL(skinny_gemm_reduce_kernel,
                       dim3((unsigned)((MN + thr - 1) / thr)), dim3(thr), 0,
                       stream, partial, (ushort_t*)y, MN, ks);
}
