// FlashAttention-2 forward + backward — hand-written gfx950 (CDNA4) MFMA.
//
// MI355X-native replacement for the reference's training attention
// (paddle `flash_attention` fused op; the csrc append_attn
// family is the inference-side sibling).  Not a CUDA port: tiles are sized
// for 64-wide waves and mfma_f32_16x16x32_bf16, K/V staged through LDS with
// +8-element padded rows (≤2-way bank aliasing, free on CDNA4 per guide G4),
// fp32 online-softmax state in registers.
//
// v1 structure: block = 4 waves (256 thr); Q tile 64 rows (16/wave);
// KV tile 64.  Within-probe A/B results at the bench shape
// (B8 S4096 Hq32 Hk8 D128, tools/bench_fa.py):
//   fwd: MF1+padded-linear ~167 TF (default).  Tried and rejected:
//     MF2 (298 VGPR -> 1 wave/SIMD, 80 TF), XOR swizzle (codegen-unstable
//     149-174 TF; conflicts are hidden behind the staging barrier in this
//     2-phase structure, guide T2 gate), global_load_lds K staging
//     (128 TF: unpadded tile re-exposes 16-way read conflicts).
//   bwd: linear ~200 TF (default); swizzle -15%.
// The planned next structure is the guide's 256^2-style multi-phase
// pipeline with counted vmcnt (T3+T4), which is what unlocks T2/T5.
//
// MFMA fragment layouts for mfma_f32_16x16x32_bf16 (HW-validated by the
// mfma_layout_probe kernel + tests/test_ops_gpu.py asymmetric check):
//   A (16x32): lane l holds A[l%16][(l/16)*8 + j], j=0..7   (one short8)
//   B (32x16): lane l holds B[(l/16)*8 + j][l%16]
//   C (16x16): lane l, reg r holds C[(l/16)*4 + r][l%16]
//
// Operand-layout cheat sheet (X row-major [row][col] in LDS):
//   "B-frag contraction over col of X" (e.g. S=QK^T contracts d):
//       read x_lds[n*16 + l16][kk*32 + lk8 .. +8]        (contiguous ✓)
//   "B-frag contraction over row of X" (e.g. O=PV contracts kv):
//       needs X transposed in LDS: xt_lds[col][row]
#include "common.h"

#define FA_WAVES 4
#define FA_BLOCK (FA_WAVES * 64)
#define BLK_M 64
#define BLK_N 64
#define LDS_PAD 8

typedef short8v frag_ab;
typedef f32x4 frag_c;

__device__ __forceinline__ frag_c mfma16(frag_ab a, frag_ab b, frag_c c) {
    return __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, c, 0, 0, 0);
}

// ---------------------------------------------------------------------------
// LDS XOR swizzle (guide §6 G4 / T2): a row-major bf16 tile read as B-frags
// has 16 lanes x 4 row-groups hitting (4r + 4g) mod 32 banks = 8-way
// conflicts (measured 1.8e8 conflicts/dispatch on the unswizzled kernels).
// XOR-ing the byte offset with (row & 7) << 4 spreads each column across 8
// distinct 16-byte slots.  Writes and reads must apply the SAME swizzle;
// the XOR is a multiple of 16 so 4/8/16-byte access alignment is preserved.
// ---------------------------------------------------------------------------
template <int LD>
__device__ __forceinline__ char* swz(ushort_t (*tile)[LD], int row, int col_elem) {
    return reinterpret_cast<char*>(&tile[row][0]) + ((col_elem * 2) ^ ((row & 7) << 4));
}

// compile-time on/off variant for within-probe A/B (guide rule #13)
template <bool SWZ, int LD>
__device__ __forceinline__ char* swzb(ushort_t (*tile)[LD], int row, int col_elem) {
    if constexpr (SWZ)
        return reinterpret_cast<char*>(&tile[row][0]) + ((col_elem * 2) ^ ((row & 7) << 4));
    else
        return reinterpret_cast<char*>(&tile[row][0]) + col_elem * 2;
}

// ---------------------------------------------------------------------------
// layout probe: one wave computes C = A @ B for 16x32 @ 32x16.  Used by the
// GPU layout-validation test with asymmetric random inputs (transpose-
// detecting, guide G9).
// ---------------------------------------------------------------------------
__global__ void mfma_layout_probe(const ushort_t* A, const ushort_t* B, float* C) {
    int l = threadIdx.x;
    frag_ab a, b;
#pragma unroll
    for (int j = 0; j < 8; j++) {
        a[j] = (short)A[(l % 16) * 32 + (l / 16) * 8 + j];
        b[j] = (short)B[((l / 16) * 8 + j) * 16 + (l % 16)];
    }
    frag_c c = {0.f, 0.f, 0.f, 0.f};
    c = mfma16(a, b, c);
#pragma unroll
    for (int r = 0; r < 4; r++) C[((l / 16) * 4 + r) * 16 + (l % 16)] = c[r];
}

// ---------------------------------------------------------------------------
// forward.  Each wave owns MF x 16 q rows (MF M-frags), so one K/V staging
// round feeds 2x the MFMA work (BLK_M = FA_WAVES * MF * 16 = 128).
// ---------------------------------------------------------------------------
template <int D, int MF, bool SWZ, bool GLL = false>
__global__ __launch_bounds__(FA_BLOCK) void flash_fwd_kernel(
    const ushort_t* __restrict__ q,   // [B, Sq, Hq, D]
    const ushort_t* __restrict__ k,   // [B, Skv, Hk, D]
    const ushort_t* __restrict__ v,   // [B, Skv, Hk, D]
    ushort_t* __restrict__ o,         // [B, Sq, Hq, D]
    float* __restrict__ lse,          // [B, Hq, Sq]
    const int* __restrict__ startend, // FlashMask [B, Skv] or nullptr:
                                      // key j visible to queries j<=i<start[j]
    int B, int Sq, int Skv, int Hq, int Hk, float scale, int causal) {
    constexpr int FWD_BLKM = FA_WAVES * MF * 16;
    constexpr int KD = D / 32;   // MFMA K-steps over the head dim
    constexpr int ND = D / 16;   // d-frags of the O accumulator
    constexpr int NN = BLK_N / 16;
    // global_load_lds requires a linear (unpadded, unswizzled) K tile
    constexpr int LDK = GLL ? D : (D + LDS_PAD);
    constexpr int LDT = BLK_N + LDS_PAD;

    __shared__ ushort_t k_lds[BLK_N][LDK];       // K row-major [kv][d]
    __shared__ ushort_t vt_lds[D][LDT];          // V transposed [d][kv]
    __shared__ ushort_t p_lds[FWD_BLKM][LDT];   // P row-major [q][kv]

    const int tid = threadIdx.x;
    const int wave = tid >> 6;
    const int lane = tid & 63;
    const int l16 = lane & 15;
    const int lk8 = (lane >> 4) * 8;

    int qt, bh;
    if ((gridDim.y & 7) == 0) {
        // XCD-aware remap (see flash_attn_v2.hip): one (b,h) per XCD
        const int flat = blockIdx.x + gridDim.x * blockIdx.y;
        const int idx = flat >> 3;
        qt = idx % gridDim.x;
        bh = (flat & 7) * (gridDim.y >> 3) + idx / gridDim.x;
    } else {
        qt = blockIdx.x;
        bh = blockIdx.y;
    }
    const int b = bh / Hq, hq = bh % Hq;
    const int hk = hq / (Hq / Hk);
    const int q_base = qt * FWD_BLKM;
    const int causal_off = Skv - Sq;  // kv visible iff kv <= q + off

    const long long q_row_stride = (long long)Hq * D;
    const long long kv_row_stride = (long long)Hk * D;
    const ushort_t* q_ptr = q + ((long long)b * Sq * Hq + hq) * D;
    const ushort_t* k_ptr = k + ((long long)b * Skv * Hk + hk) * D;
    const ushort_t* v_ptr = v + ((long long)b * Skv * Hk + hk) * D;

    // Q A-frags in registers (MF x 16 rows x D per wave)
    frag_ab aq[MF][KD];
#pragma unroll
    for (int mf = 0; mf < MF; mf++) {
        int qrow = q_base + (wave * MF + mf) * 16 + l16;
        if (qrow < Sq) {
#pragma unroll
            for (int kk = 0; kk < KD; kk++)
                aq[mf][kk] = *reinterpret_cast<const frag_ab*>(
                    q_ptr + (long long)qrow * q_row_stride + kk * 32 + lk8);
        } else {
#pragma unroll
            for (int kk = 0; kk < KD; kk++) aq[mf][kk] = frag_ab{0};
        }
    }

    float m_run[MF][4], l_run[MF][4];
#pragma unroll
    for (int mf = 0; mf < MF; mf++)
#pragma unroll
        for (int r = 0; r < 4; r++) { m_run[mf][r] = -INFINITY; l_run[mf][r] = 0.f; }
    frag_c acc_o[MF][ND];
#pragma unroll
    for (int mf = 0; mf < MF; mf++)
#pragma unroll
        for (int n = 0; n < ND; n++) acc_o[mf][n] = frag_c{0.f, 0.f, 0.f, 0.f};

    int n_kv_tiles = (Skv + BLK_N - 1) / BLK_N;
    if (causal) {
        int max_kv = q_base + FWD_BLKM - 1 + causal_off;
        int lim = (max_kv + BLK_N) / BLK_N;
        n_kv_tiles = min(n_kv_tiles, max(lim, 0));
    }

    for (int kvt = 0; kvt < n_kv_tiles; kvt++) {
        const int kv_base = kvt * BLK_N;
        if constexpr (GLL) {
            // direct HBM -> LDS for the K tile (guide §5: the compiler never
            // auto-emits global_load_lds; width 16 removes the VGPR staging
            // round-trip).  Dest is wave-uniform base + lane*16.
            for (int chunk = wave; chunk < BLK_N * D * 2 / 1024; chunk += FA_WAVES) {
                int byte_off = chunk * 1024 + lane * 16;
                int row = byte_off / (D * 2);
                int col = (byte_off % (D * 2)) / 2;
                int kvg = kv_base + row;
                if (kvg >= Skv) kvg = Skv - 1;  // clamp: masked later, avoids NaN garbage
                const ushort_t* src = k_ptr + (long long)kvg * kv_row_stride + col;
                __builtin_amdgcn_global_load_lds(
                    reinterpret_cast<const unsigned int*>(src),
                    reinterpret_cast<unsigned int*>(&k_lds[0][0]) + byte_off / 4,
                    16, 0, 0);
            }
            for (int idx = tid * 8; idx < (BLK_N / 2) * D; idx += FA_BLOCK * 8) {
                int rr = idx / D, col = idx % D;
                int row0 = rr * 2;
                int kvg0 = kv_base + row0;
                short8v v8a = {0,0,0,0,0,0,0,0}, v8b = {0,0,0,0,0,0,0,0};
                if (kvg0 < Skv)
                    v8a = *reinterpret_cast<const short8v*>(v_ptr + (long long)kvg0 * kv_row_stride + col);
                if (kvg0 + 1 < Skv)
                    v8b = *reinterpret_cast<const short8v*>(v_ptr + (long long)(kvg0 + 1) * kv_row_stride + col);
#pragma unroll
                for (int j = 0; j < 8; j++) {
                    unsigned int packed = ((unsigned int)(unsigned short)v8a[j]) |
                                          (((unsigned int)(unsigned short)v8b[j]) << 16);
                    *reinterpret_cast<unsigned int*>(swzb<SWZ>(vt_lds, col + j, row0)) = packed;
                }
            }
        } else {
        // stage K row-major (vector) + V transposed (b32-packed: two kv rows
        // per thread so the LDS transpose writes are 4-byte, halving ds ops)
        for (int idx = tid * 8; idx < (BLK_N / 2) * D; idx += FA_BLOCK * 8) {
            int rr = idx / D, col = idx % D;
            int row0 = rr * 2;
            int kvg0 = kv_base + row0;
            short8v k8a = {0,0,0,0,0,0,0,0}, k8b = {0,0,0,0,0,0,0,0};
            short8v v8a = {0,0,0,0,0,0,0,0}, v8b = {0,0,0,0,0,0,0,0};
            if (kvg0 < Skv) {
                k8a = *reinterpret_cast<const short8v*>(k_ptr + (long long)kvg0 * kv_row_stride + col);
                v8a = *reinterpret_cast<const short8v*>(v_ptr + (long long)kvg0 * kv_row_stride + col);
            }
            if (kvg0 + 1 < Skv) {
                k8b = *reinterpret_cast<const short8v*>(k_ptr + (long long)(kvg0 + 1) * kv_row_stride + col);
                v8b = *reinterpret_cast<const short8v*>(v_ptr + (long long)(kvg0 + 1) * kv_row_stride + col);
            }
            *reinterpret_cast<short8v*>(swzb<SWZ>(k_lds, row0, col)) = k8a;
            *reinterpret_cast<short8v*>(swzb<SWZ>(k_lds, row0 + 1, col)) = k8b;
#pragma unroll
            for (int j = 0; j < 8; j++) {
                unsigned int packed = ((unsigned int)(unsigned short)v8a[j]) |
                                      (((unsigned int)(unsigned short)v8b[j]) << 16);
                *reinterpret_cast<unsigned int*>(swzb<SWZ>(vt_lds, col + j, row0)) = packed;
            }
        }
        }
        __syncthreads();

        // S = Q K^T (contract d; B-frag from K row-major), then softmax with
        // P written in place into the S accumulator, then P -> LDS
#pragma unroll
        for (int mf = 0; mf < MF; mf++) {
            frag_c acc_s[NN];
#pragma unroll
            for (int n = 0; n < NN; n++) {
                acc_s[n] = frag_c{0.f, 0.f, 0.f, 0.f};
#pragma unroll
                for (int kk = 0; kk < KD; kk++) {
                    frag_ab bk = *reinterpret_cast<const frag_ab*>(swzb<SWZ>(k_lds, n * 16 + l16, kk * 32 + lk8));
                    acc_s[n] = mfma16(aq[mf][kk], bk, acc_s[n]);
                }
            }
            const int qrow0 = q_base + (wave * MF + mf) * 16 + (lane >> 4) * 4;
#pragma unroll
            for (int n = 0; n < NN; n++) {
                int kvg = kv_base + n * 16 + l16;
                int kv_end = (startend && kvg < Skv)
                                 ? startend[(long long)b * Skv + kvg] : Sq + causal_off;
#pragma unroll
                for (int r = 0; r < 4; r++) {
                    int qg = qrow0 + r;
                    bool vis = (kvg < Skv) && (qg < Sq);
                    if (causal) vis = vis && (kvg <= qg + causal_off);
                    if (startend) vis = vis && (qg < kv_end);
                    acc_s[n][r] = vis ? acc_s[n][r] * scale : -INFINITY;
                }
            }
#pragma unroll
            for (int r = 0; r < 4; r++) {
                float row_max = -INFINITY;
#pragma unroll
                for (int n = 0; n < NN; n++) row_max = fmaxf(row_max, acc_s[n][r]);
#pragma unroll
                for (int off = 8; off > 0; off >>= 1)
                    row_max = fmaxf(row_max, __shfl_xor(row_max, off, 64));
                float m_new = fmaxf(m_run[mf][r], row_max);
                float alpha;
                if (m_new == -INFINITY) {
                    alpha = 1.f;
                } else if (m_run[mf][r] == -INFINITY) {
                    alpha = 0.f;
                } else {
                    alpha = __expf(m_run[mf][r] - m_new);
                }
                float row_sum = 0.f;
#pragma unroll
                for (int n = 0; n < NN; n++) {
                    float p = (m_new == -INFINITY || acc_s[n][r] == -INFINITY)
                                  ? 0.f : __expf(acc_s[n][r] - m_new);
                    acc_s[n][r] = p;  // reuse the S accumulator for P
                    row_sum += p;
                }
#pragma unroll
                for (int off = 8; off > 0; off >>= 1) row_sum += __shfl_xor(row_sum, off, 64);
                l_run[mf][r] = l_run[mf][r] * alpha + row_sum;
                m_run[mf][r] = m_new;
#pragma unroll
                for (int n = 0; n < ND; n++) acc_o[mf][n][r] *= alpha;
            }
#pragma unroll
            for (int n = 0; n < NN; n++)
#pragma unroll
                for (int r = 0; r < 4; r++)
                    *reinterpret_cast<ushort_t*>(swzb<SWZ>(p_lds, (wave * MF + mf) * 16 + (lane >> 4) * 4 + r, n * 16 + l16)) =
                        f32_to_bf16(acc_s[n][r]);
        }
        // each wave reads back only its own p_lds rows: a wave-local LDS
        // drain is enough (no cross-wave barrier needed here)
        asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");

        // O += P V  (contract kv; A from p_lds, B from vt_lds)
#pragma unroll
        for (int kk = 0; kk < BLK_N / 32; kk++) {
#pragma unroll
            for (int mf = 0; mf < MF; mf++) {
                frag_ab ap = *reinterpret_cast<const frag_ab*>(swzb<SWZ>(p_lds, (wave * MF + mf) * 16 + l16, kk * 32 + lk8));
#pragma unroll
                for (int n = 0; n < ND; n++) {
                    frag_ab bv = *reinterpret_cast<const frag_ab*>(swzb<SWZ>(vt_lds, n * 16 + l16, kk * 32 + lk8));
                    acc_o[mf][n] = mfma16(ap, bv, acc_o[mf][n]);
                }
            }
        }
        __syncthreads();
    }

    // epilogue: O/l and LSE
#pragma unroll
    for (int mf = 0; mf < MF; mf++) {
        const int qrow0 = q_base + (wave * MF + mf) * 16 + (lane >> 4) * 4;
#pragma unroll
        for (int r = 0; r < 4; r++) {
            int qg = qrow0 + r;
            if (qg >= Sq) continue;
            float inv_l = (l_run[mf][r] > 0.f) ? 1.0f / l_run[mf][r] : 0.f;
            ushort_t* orow = o + ((long long)b * Sq + qg) * q_row_stride + (long long)hq * D;
#pragma unroll
            for (int n = 0; n < ND; n++)
                orow[n * 16 + l16] = f32_to_bf16(acc_o[mf][n][r] * inv_l);
            if (l16 == 0) {
                float lv = (l_run[mf][r] > 0.f) ? (m_run[mf][r] + __logf(l_run[mf][r])) : -INFINITY;
                lse[((long long)b * Hq + hq) * Sq + qg] = lv;
            }
        }
    }
}


// ---------------------------------------------------------------------------
// forward, software-pipelined (guide T14 async-stage split): the next KV
// tile's global loads are ISSUED before this tile's PV MFMAs so the ~500-
// cycle HBM latency hides under compute; the register-staged data is written
// to LDS after the barrier.  Same barrier count per tile as the base kernel.
// ---------------------------------------------------------------------------
template <int D>
__global__ __launch_bounds__(FA_BLOCK) void flash_fwd_pipe_kernel(
    const ushort_t* __restrict__ q, const ushort_t* __restrict__ k,
    const ushort_t* __restrict__ v, ushort_t* __restrict__ o,
    float* __restrict__ lse,
    int B, int Sq, int Skv, int Hq, int Hk, float scale, int causal) {
    constexpr int BLKM = FA_WAVES * 16;
    constexpr int KD = D / 32;
    constexpr int ND = D / 16;
    constexpr int NN = BLK_N / 16;
    constexpr int LDK = D + LDS_PAD;
    constexpr int LDT = BLK_N + LDS_PAD;
    constexpr int ITERS = (BLK_N / 2) * D / (FA_BLOCK * 8);  // reg-staged chunks/thread

    __shared__ ushort_t k_lds[BLK_N][LDK];
    __shared__ ushort_t vt_lds[D][LDT];
    __shared__ ushort_t p_lds[BLKM][LDT];

    const int tid = threadIdx.x;
    const int wave = tid >> 6;
    const int lane = tid & 63;
    const int l16 = lane & 15;
    const int lk8 = (lane >> 4) * 8;

    int qt, bh;
    if ((gridDim.y & 7) == 0) {
        // XCD-aware remap (see flash_attn_v2.hip): one (b,h) per XCD
        const int flat = blockIdx.x + gridDim.x * blockIdx.y;
        const int idx = flat >> 3;
        qt = idx % gridDim.x;
        bh = (flat & 7) * (gridDim.y >> 3) + idx / gridDim.x;
    } else {
        qt = blockIdx.x;
        bh = blockIdx.y;
    }
    const int b = bh / Hq, hq = bh % Hq;
    const int hk = hq / (Hq / Hk);
    const int q_base = qt * BLKM;
    const int causal_off = Skv - Sq;

    const long long q_row_stride = (long long)Hq * D;
    const long long kv_row_stride = (long long)Hk * D;
    const ushort_t* q_ptr = q + ((long long)b * Sq * Hq + hq) * D;
    const ushort_t* k_ptr = k + ((long long)b * Skv * Hk + hk) * D;
    const ushort_t* v_ptr = v + ((long long)b * Skv * Hk + hk) * D;

    frag_ab aq[KD];
    {
        int qrow = q_base + wave * 16 + l16;
        if (qrow < Sq) {
#pragma unroll
            for (int kk = 0; kk < KD; kk++)
                aq[kk] = *reinterpret_cast<const frag_ab*>(
                    q_ptr + (long long)qrow * q_row_stride + kk * 32 + lk8);
        } else {
#pragma unroll
            for (int kk = 0; kk < KD; kk++) aq[kk] = frag_ab{0};
        }
    }

    float m_run[4], l_run[4];
#pragma unroll
    for (int r = 0; r < 4; r++) { m_run[r] = -INFINITY; l_run[r] = 0.f; }
    frag_c acc_o[ND];
#pragma unroll
    for (int n = 0; n < ND; n++) acc_o[n] = frag_c{0.f, 0.f, 0.f, 0.f};

    int n_kv_tiles = (Skv + BLK_N - 1) / BLK_N;
    if (causal) {
        int max_kv = q_base + BLKM - 1 + causal_off;
        int lim = (max_kv + BLK_N) / BLK_N;
        n_kv_tiles = min(n_kv_tiles, max(lim, 0));
    }

    // register staging buffers: this thread's chunks of the NEXT tile
    short8v sk[ITERS][2], sv[ITERS][2];

    auto load_tile = [&](int kv_base) {
#pragma unroll
        for (int it = 0; it < ITERS; it++) {
            int idx = (tid + it * FA_BLOCK) * 8;
            int rr = idx / D, col = idx % D;
            int row0 = rr * 2;
            int kvg0 = kv_base + row0;
            sk[it][0] = short8v{0,0,0,0,0,0,0,0}; sk[it][1] = short8v{0,0,0,0,0,0,0,0};
            sv[it][0] = short8v{0,0,0,0,0,0,0,0}; sv[it][1] = short8v{0,0,0,0,0,0,0,0};
            if (kvg0 < Skv) {
                sk[it][0] = *reinterpret_cast<const short8v*>(k_ptr + (long long)kvg0 * kv_row_stride + col);
                sv[it][0] = *reinterpret_cast<const short8v*>(v_ptr + (long long)kvg0 * kv_row_stride + col);
            }
            if (kvg0 + 1 < Skv) {
                sk[it][1] = *reinterpret_cast<const short8v*>(k_ptr + (long long)(kvg0 + 1) * kv_row_stride + col);
                sv[it][1] = *reinterpret_cast<const short8v*>(v_ptr + (long long)(kvg0 + 1) * kv_row_stride + col);
            }
        }
    };

    auto write_tile = [&]() {
#pragma unroll
        for (int it = 0; it < ITERS; it++) {
            int idx = (tid + it * FA_BLOCK) * 8;
            int rr = idx / D, col = idx % D;
            int row0 = rr * 2;
            *reinterpret_cast<short8v*>(&k_lds[row0][col]) = sk[it][0];
            *reinterpret_cast<short8v*>(&k_lds[row0 + 1][col]) = sk[it][1];
#pragma unroll
            for (int j = 0; j < 8; j++) {
                unsigned int packed = ((unsigned int)(unsigned short)sv[it][0][j]) |
                                      (((unsigned int)(unsigned short)sv[it][1][j]) << 16);
                *reinterpret_cast<unsigned int*>(&vt_lds[col + j][row0]) = packed;
            }
        }
    };

    // prologue: tile 0 into LDS
    load_tile(0);
    write_tile();
    __syncthreads();

    for (int kvt = 0; kvt < n_kv_tiles; kvt++) {
        const int kv_base = kvt * BLK_N;

        frag_c acc_s[NN];
#pragma unroll
        for (int n = 0; n < NN; n++) {
            acc_s[n] = frag_c{0.f, 0.f, 0.f, 0.f};
#pragma unroll
            for (int kk = 0; kk < KD; kk++) {
                frag_ab bk = *reinterpret_cast<const frag_ab*>(&k_lds[n * 16 + l16][kk * 32 + lk8]);
                acc_s[n] = mfma16(aq[kk], bk, acc_s[n]);
            }
        }

        // issue the NEXT tile's global loads now: their latency hides under
        // the softmax + PV work below (guide T14)
        if (kvt + 1 < n_kv_tiles) load_tile(kv_base + BLK_N);

        const int qrow0 = q_base + wave * 16 + (lane >> 4) * 4;
#pragma unroll
        for (int n = 0; n < NN; n++) {
            int kvg = kv_base + n * 16 + l16;
#pragma unroll
            for (int r = 0; r < 4; r++) {
                int qg = qrow0 + r;
                bool vis = (kvg < Skv) && (qg < Sq);
                if (causal) vis = vis && (kvg <= qg + causal_off);
                acc_s[n][r] = vis ? acc_s[n][r] * scale : -INFINITY;
            }
        }
#pragma unroll
        for (int r = 0; r < 4; r++) {
            float row_max = -INFINITY;
#pragma unroll
            for (int n = 0; n < NN; n++) row_max = fmaxf(row_max, acc_s[n][r]);
#pragma unroll
            for (int off = 8; off > 0; off >>= 1)
                row_max = fmaxf(row_max, __shfl_xor(row_max, off, 64));
            float m_new = fmaxf(m_run[r], row_max);
            float alpha;
            if (m_new == -INFINITY) alpha = 1.f;
            else if (m_run[r] == -INFINITY) alpha = 0.f;
            else alpha = __expf(m_run[r] - m_new);
            float row_sum = 0.f;
#pragma unroll
            for (int n = 0; n < NN; n++) {
                float pp = (m_new == -INFINITY || acc_s[n][r] == -INFINITY)
                               ? 0.f : __expf(acc_s[n][r] - m_new);
                acc_s[n][r] = pp;
                row_sum += pp;
            }
#pragma unroll
            for (int off = 8; off > 0; off >>= 1) row_sum += __shfl_xor(row_sum, off, 64);
            l_run[r] = l_run[r] * alpha + row_sum;
            m_run[r] = m_new;
#pragma unroll
            for (int n = 0; n < ND; n++) acc_o[n][r] *= alpha;
        }
#pragma unroll
        for (int n = 0; n < NN; n++)
#pragma unroll
            for (int r = 0; r < 4; r++)
                p_lds[wave * 16 + (lane >> 4) * 4 + r][n * 16 + l16] =
                    f32_to_bf16(acc_s[n][r]);
        asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");

#pragma unroll
        for (int kk = 0; kk < BLK_N / 32; kk++) {
            frag_ab ap = *reinterpret_cast<const frag_ab*>(&p_lds[wave * 16 + l16][kk * 32 + lk8]);
#pragma unroll
            for (int n = 0; n < ND; n++) {
                frag_ab bv = *reinterpret_cast<const frag_ab*>(&vt_lds[n * 16 + l16][kk * 32 + lk8]);
                acc_o[n] = mfma16(ap, bv, acc_o[n]);
            }
        }
        __syncthreads();      // everyone done reading k_lds / vt_lds
        if (kvt + 1 < n_kv_tiles) {
            write_tile();      // staged regs -> LDS for the next iteration
            __syncthreads();
        }
    }

    const int qrow0 = q_base + wave * 16 + (lane >> 4) * 4;
#pragma unroll
    for (int r = 0; r < 4; r++) {
        int qg = qrow0 + r;
        if (qg >= Sq) continue;
        float inv_l = (l_run[r] > 0.f) ? 1.0f / l_run[r] : 0.f;
        ushort_t* orow = o + ((long long)b * Sq + qg) * q_row_stride + (long long)hq * D;
#pragma unroll
        for (int n = 0; n < ND; n++)
            orow[n * 16 + l16] = f32_to_bf16(acc_o[n][r] * inv_l);
        if (l16 == 0) {
            float lv = (l_run[r] > 0.f) ? (m_run[r] + __logf(l_run[r])) : -INFINITY;
            lse[((long long)b * Hq + hq) * Sq + qg] = lv;
        }
    }
}

// ---------------------------------------------------------------------------
// backward preprocess: delta[b,h,q] = rowsum(dO * O)
// ---------------------------------------------------------------------------
template <int D>
__global__ void flash_bwd_delta_kernel(
    const ushort_t* __restrict__ dout, const ushort_t* __restrict__ o,
    float* __restrict__ delta, int B, int Sq, int Hq) {
    long long row = (long long)blockIdx.x * (blockDim.x >> 6) + (threadIdx.x >> 6);
    long long total = (long long)B * Sq * Hq;
    if (row >= total) return;
    int lane = threadIdx.x & 63;
    const ushort_t* dor = dout + row * D;
    const ushort_t* orow = o + row * D;
    float acc = 0.f;
    for (int i = lane * 2; i < D; i += 128) {
        acc += bf16_to_f32(dor[i]) * bf16_to_f32(orow[i]);
        acc += bf16_to_f32(dor[i + 1]) * bf16_to_f32(orow[i + 1]);
    }
    acc = wave_reduce_sum(acc);
    if (lane == 0) {
        long long h = row % Hq;
        long long bq = row / Hq;
        long long bb = bq / Sq, qi = bq % Sq;
        delta[(bb * Hq + h) * Sq + qi] = acc;
    }
}

// ---------------------------------------------------------------------------
// backward dQ: fixed q tile, loop kv tiles.
//   S = scale*QK^T; P = exp(S - lse); dP = dO V^T;
//   dS = P*(dP - delta)*scale; dQ += dS K
// LDS: K row-major (S B-frag), V row-major (dP B-frag),
//      K^T (dQ B-frag), dS staging.
// ---------------------------------------------------------------------------
template <int D, bool SWZ>
__global__ __launch_bounds__(FA_BLOCK) void flash_bwd_dq_kernel(
    const ushort_t* __restrict__ q, const ushort_t* __restrict__ k,
    const ushort_t* __restrict__ v, const ushort_t* __restrict__ dout,
    const float* __restrict__ lse, const float* __restrict__ delta,
    ushort_t* __restrict__ dq, const int* __restrict__ startend,
    int B, int Sq, int Skv, int Hq, int Hk, float scale, int causal) {
    constexpr int KD = D / 32;
    constexpr int ND = D / 16;
    constexpr int NN = BLK_N / 16;
    constexpr int LDK = D + LDS_PAD;
    constexpr int LDT = BLK_N + LDS_PAD;

    __shared__ ushort_t k_lds[BLK_N][LDK];
    __shared__ ushort_t v_lds[BLK_N][LDK];
    __shared__ ushort_t kt_lds[D][LDT];
    __shared__ ushort_t ds_lds[BLK_M][LDT];

    const int tid = threadIdx.x;
    const int wave = tid >> 6;
    const int lane = tid & 63;
    const int l16 = lane & 15;
    const int lk8 = (lane >> 4) * 8;

    int qt, bh;
    if ((gridDim.y & 7) == 0) {
        // XCD-aware remap (see flash_attn_v2.hip): one (b,h) per XCD
        const int flat = blockIdx.x + gridDim.x * blockIdx.y;
        const int idx = flat >> 3;
        qt = idx % gridDim.x;
        bh = (flat & 7) * (gridDim.y >> 3) + idx / gridDim.x;
    } else {
        qt = blockIdx.x;
        bh = blockIdx.y;
    }
    const int b = bh / Hq, hq = bh % Hq;
    const int hk = hq / (Hq / Hk);
    const int q_base = qt * BLK_M;
    const int causal_off = Skv - Sq;

    const long long q_row_stride = (long long)Hq * D;
    const long long kv_row_stride = (long long)Hk * D;
    const ushort_t* q_ptr = q + ((long long)b * Sq * Hq + hq) * D;
    const ushort_t* k_ptr = k + ((long long)b * Skv * Hk + hk) * D;
    const ushort_t* v_ptr = v + ((long long)b * Skv * Hk + hk) * D;
    const ushort_t* do_ptr = dout + ((long long)b * Sq * Hq + hq) * D;
    const float* lse_row = lse + ((long long)b * Hq + hq) * Sq;
    const float* dl_row = delta + ((long long)b * Hq + hq) * Sq;

    frag_ab aq[KD], ado[KD];
    {
        int qrow = q_base + wave * 16 + l16;
        if (qrow < Sq) {
#pragma unroll
            for (int kk = 0; kk < KD; kk++) {
                aq[kk] = *reinterpret_cast<const frag_ab*>(
                    q_ptr + (long long)qrow * q_row_stride + kk * 32 + lk8);
                ado[kk] = *reinterpret_cast<const frag_ab*>(
                    do_ptr + (long long)qrow * q_row_stride + kk * 32 + lk8);
            }
        } else {
#pragma unroll
            for (int kk = 0; kk < KD; kk++) { aq[kk] = frag_ab{0}; ado[kk] = frag_ab{0}; }
        }
    }
    const int qrow0 = q_base + wave * 16 + (lane >> 4) * 4;
    float lse_r[4], dl_r[4];
#pragma unroll
    for (int r = 0; r < 4; r++) {
        int qg = qrow0 + r;
        lse_r[r] = (qg < Sq) ? lse_row[qg] : INFINITY;
        dl_r[r] = (qg < Sq) ? dl_row[qg] : 0.f;
    }

    frag_c acc_dq[ND];
#pragma unroll
    for (int n = 0; n < ND; n++) acc_dq[n] = frag_c{0.f, 0.f, 0.f, 0.f};

    int n_kv_tiles = (Skv + BLK_N - 1) / BLK_N;
    if (causal) {
        int max_kv = q_base + BLK_M - 1 + causal_off;
        int lim = (max_kv + BLK_N) / BLK_N;
        n_kv_tiles = min(n_kv_tiles, max(lim, 0));
    }

    for (int kvt = 0; kvt < n_kv_tiles; kvt++) {
        const int kv_base = kvt * BLK_N;
        for (int idx = tid * 8; idx < (BLK_N / 2) * D; idx += FA_BLOCK * 8) {
            int rr = idx / D, col = idx % D;
            int row0 = rr * 2;
            int kvg0 = kv_base + row0;
            short8v k8a = {0,0,0,0,0,0,0,0}, k8b = {0,0,0,0,0,0,0,0};
            short8v v8a = {0,0,0,0,0,0,0,0}, v8b = {0,0,0,0,0,0,0,0};
            if (kvg0 < Skv) {
                k8a = *reinterpret_cast<const short8v*>(k_ptr + (long long)kvg0 * kv_row_stride + col);
                v8a = *reinterpret_cast<const short8v*>(v_ptr + (long long)kvg0 * kv_row_stride + col);
            }
            if (kvg0 + 1 < Skv) {
                k8b = *reinterpret_cast<const short8v*>(k_ptr + (long long)(kvg0 + 1) * kv_row_stride + col);
                v8b = *reinterpret_cast<const short8v*>(v_ptr + (long long)(kvg0 + 1) * kv_row_stride + col);
            }
            *reinterpret_cast<short8v*>(swzb<SWZ>(k_lds, row0, col)) = k8a;
            *reinterpret_cast<short8v*>(swzb<SWZ>(k_lds, row0 + 1, col)) = k8b;
            *reinterpret_cast<short8v*>(swzb<SWZ>(v_lds, row0, col)) = v8a;
            *reinterpret_cast<short8v*>(swzb<SWZ>(v_lds, row0 + 1, col)) = v8b;
#pragma unroll
            for (int j = 0; j < 8; j++) {
                unsigned int packed = ((unsigned int)(unsigned short)k8a[j]) |
                                      (((unsigned int)(unsigned short)k8b[j]) << 16);
                *reinterpret_cast<unsigned int*>(swzb<SWZ>(kt_lds, col + j, row0)) = packed;
            }
        }
        __syncthreads();

        // S and dP
        frag_c acc_s[NN], acc_dp[NN];
#pragma unroll
        for (int n = 0; n < NN; n++) {
            acc_s[n] = frag_c{0.f, 0.f, 0.f, 0.f};
            acc_dp[n] = frag_c{0.f, 0.f, 0.f, 0.f};
#pragma unroll
            for (int kk = 0; kk < KD; kk++) {
                frag_ab bk = *reinterpret_cast<const frag_ab*>(swzb<SWZ>(k_lds, n * 16 + l16, kk * 32 + lk8));
                frag_ab bv = *reinterpret_cast<const frag_ab*>(swzb<SWZ>(v_lds, n * 16 + l16, kk * 32 + lk8));
                acc_s[n] = mfma16(aq[kk], bk, acc_s[n]);
                acc_dp[n] = mfma16(ado[kk], bv, acc_dp[n]);
            }
        }

        // dS = P * (dP - delta) * scale   (C layout)
#pragma unroll
        for (int n = 0; n < NN; n++) {
            int kvg = kv_base + n * 16 + l16;
            int kv_end = (startend && kvg < Skv)
                             ? startend[(long long)b * Skv + kvg] : Sq + causal_off;
#pragma unroll
            for (int r = 0; r < 4; r++) {
                int qg = qrow0 + r;
                bool vis = (kvg < Skv) && (qg < Sq);
                if (causal) vis = vis && (kvg <= qg + causal_off);
                if (startend) vis = vis && (qg < kv_end);
                float p = vis ? __expf(acc_s[n][r] * scale - lse_r[r]) : 0.f;
                float ds = p * (acc_dp[n][r] - dl_r[r]) * scale;
                *reinterpret_cast<ushort_t*>(swzb<SWZ>(ds_lds, wave * 16 + (lane >> 4) * 4 + r, n * 16 + l16)) = f32_to_bf16(ds);
            }
        }
        // ds_lds rows are wave-private: wave-local LDS drain suffices
        asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");

        // dQ += dS K  (contract kv; B from kt_lds)
#pragma unroll
        for (int kk = 0; kk < BLK_N / 32; kk++) {
            frag_ab ads = *reinterpret_cast<const frag_ab*>(swzb<SWZ>(ds_lds, wave * 16 + l16, kk * 32 + lk8));
#pragma unroll
            for (int n = 0; n < ND; n++) {
                frag_ab bkt = *reinterpret_cast<const frag_ab*>(swzb<SWZ>(kt_lds, n * 16 + l16, kk * 32 + lk8));
                acc_dq[n] = mfma16(ads, bkt, acc_dq[n]);
            }
        }
        __syncthreads();
    }

    // store dQ
#pragma unroll
    for (int r = 0; r < 4; r++) {
        int qg = qrow0 + r;
        if (qg >= Sq) continue;
        ushort_t* dqr = dq + ((long long)b * Sq + qg) * q_row_stride + (long long)hq * D;
#pragma unroll
        for (int n = 0; n < ND; n++)
            dqr[n * 16 + l16] = f32_to_bf16(acc_dq[n][r]);
    }
}

// ---------------------------------------------------------------------------
// backward dK/dV: fixed kv tile, loop q tiles.  Wave owns 16 kv rows.
//   S^T = scale*K Q^T;  P^T = exp(S^T - lse[q]);
//   dV += P^T dO;  dP^T = V dO^T;  dS^T = P^T*(dP^T - delta[q])*scale;
//   dK += dS^T Q
// GQA: outputs are PER Q-HEAD ([B, Skv, Hq, D]); python sums head groups.
// LDS: Q row-major (S^T B), Q^T (dK B), dO row-major (dP^T B),
//      dO^T (dV B), P^T/dS^T staging (one reused buffer).
// ---------------------------------------------------------------------------
template <int D, bool SWZ>
__global__ __launch_bounds__(FA_BLOCK) void flash_bwd_dkv_kernel(
    const ushort_t* __restrict__ q, const ushort_t* __restrict__ k,
    const ushort_t* __restrict__ v, const ushort_t* __restrict__ dout,
    const float* __restrict__ lse, const float* __restrict__ delta,
    ushort_t* __restrict__ dk,   // [B, Skv, Hk, D] (GQA group summed in-kernel)
    ushort_t* __restrict__ dv,   // [B, Skv, Hk, D]
    const int* __restrict__ startend,
    int B, int Sq, int Skv, int Hq, int Hk, float scale, int causal) {
    constexpr int KD = D / 32;
    constexpr int ND = D / 16;
    constexpr int NN = BLK_M / 16;  // q-frags per tile
    constexpr int LDK = D + LDS_PAD;
    constexpr int LDT = BLK_M + LDS_PAD;

    __shared__ ushort_t q_lds[BLK_M][LDK];
    __shared__ ushort_t do_lds[BLK_M][LDK];
    __shared__ ushort_t qt_lds[D][LDT];
    __shared__ ushort_t dot_lds[D][LDT];
    __shared__ ushort_t pt_lds[BLK_N][LDT];  // P^T then dS^T (reused)

    const int tid = threadIdx.x;
    const int wave = tid >> 6;
    const int lane = tid & 63;
    const int l16 = lane & 15;
    const int lk8 = (lane >> 4) * 8;

    int kvt, bh;
    if ((gridDim.y & 7) == 0) {
        // XCD-aware remap (see flash_attn_v2.hip): one (b,h) per XCD
        const int flat = blockIdx.x + gridDim.x * blockIdx.y;
        const int idx = flat >> 3;
        kvt = idx % gridDim.x;
        bh = (flat & 7) * (gridDim.y >> 3) + idx / gridDim.x;
    } else {
        kvt = blockIdx.x;
        bh = blockIdx.y;
    }
    const int b = bh / Hk, hk = bh % Hk;
    const int G = Hq / Hk;               // q heads accumulated in-kernel
    const int kv_base = kvt * BLK_N;
    const int causal_off = Skv - Sq;

    const long long q_row_stride = (long long)Hq * D;
    const long long kv_row_stride = (long long)Hk * D;
    const ushort_t* k_ptr = k + ((long long)b * Skv * Hk + hk) * D;
    const ushort_t* v_ptr = v + ((long long)b * Skv * Hk + hk) * D;

    // K and V A-frags in registers (wave's 16 kv rows, fixed all kernel)
    frag_ab ak[KD], av[KD];
    {
        int kvrow = kv_base + wave * 16 + l16;
        if (kvrow < Skv) {
#pragma unroll
            for (int kk = 0; kk < KD; kk++) {
                ak[kk] = *reinterpret_cast<const frag_ab*>(
                    k_ptr + (long long)kvrow * kv_row_stride + kk * 32 + lk8);
                av[kk] = *reinterpret_cast<const frag_ab*>(
                    v_ptr + (long long)kvrow * kv_row_stride + kk * 32 + lk8);
            }
        } else {
#pragma unroll
            for (int kk = 0; kk < KD; kk++) { ak[kk] = frag_ab{0}; av[kk] = frag_ab{0}; }
        }
    }

    frag_c acc_dk[ND], acc_dv[ND];
#pragma unroll
    for (int n = 0; n < ND; n++) {
        acc_dk[n] = frag_c{0.f, 0.f, 0.f, 0.f};
        acc_dv[n] = frag_c{0.f, 0.f, 0.f, 0.f};
    }

    int qt_start = 0;
    if (causal) {
        // first q that can see kv_base: q >= kv_base - off
        int first_q = kv_base - causal_off;
        if (first_q > 0) qt_start = first_q / BLK_M;
    }
    int n_q_tiles = (Sq + BLK_M - 1) / BLK_M;

    // accumulate the whole GQA group into acc_dk/acc_dv: K/V stream from HBM
    // once per kv head, and the outputs are written [B, Skv, Hk, D] directly
    // (no per-q-head buffers + reduction pass)
    for (int g = 0; g < G; g++) {
    const int hq = hk * G + g;
    const ushort_t* q_ptr = q + ((long long)b * Sq * Hq + hq) * D;
    const ushort_t* do_ptr = dout + ((long long)b * Sq * Hq + hq) * D;
    const float* lse_row = lse + ((long long)b * Hq + hq) * Sq;
    const float* dl_row = delta + ((long long)b * Hq + hq) * Sq;
    for (int qt = qt_start; qt < n_q_tiles; qt++) {
        const int q_base = qt * BLK_M;
        for (int idx = tid * 8; idx < (BLK_M / 2) * D; idx += FA_BLOCK * 8) {
            int rr = idx / D, col = idx % D;
            int row0 = rr * 2;
            int qg0 = q_base + row0;
            short8v q8a = {0,0,0,0,0,0,0,0}, q8b = {0,0,0,0,0,0,0,0};
            short8v d8a = {0,0,0,0,0,0,0,0}, d8b = {0,0,0,0,0,0,0,0};
            if (qg0 < Sq) {
                q8a = *reinterpret_cast<const short8v*>(q_ptr + (long long)qg0 * q_row_stride + col);
                d8a = *reinterpret_cast<const short8v*>(do_ptr + (long long)qg0 * q_row_stride + col);
            }
            if (qg0 + 1 < Sq) {
                q8b = *reinterpret_cast<const short8v*>(q_ptr + (long long)(qg0 + 1) * q_row_stride + col);
                d8b = *reinterpret_cast<const short8v*>(do_ptr + (long long)(qg0 + 1) * q_row_stride + col);
            }
            *reinterpret_cast<short8v*>(swzb<SWZ>(q_lds, row0, col)) = q8a;
            *reinterpret_cast<short8v*>(swzb<SWZ>(q_lds, row0 + 1, col)) = q8b;
            *reinterpret_cast<short8v*>(swzb<SWZ>(do_lds, row0, col)) = d8a;
            *reinterpret_cast<short8v*>(swzb<SWZ>(do_lds, row0 + 1, col)) = d8b;
#pragma unroll
            for (int j = 0; j < 8; j++) {
                unsigned int pq = ((unsigned int)(unsigned short)q8a[j]) |
                                  (((unsigned int)(unsigned short)q8b[j]) << 16);
                unsigned int pd = ((unsigned int)(unsigned short)d8a[j]) |
                                  (((unsigned int)(unsigned short)d8b[j]) << 16);
                *reinterpret_cast<unsigned int*>(swzb<SWZ>(qt_lds, col + j, row0)) = pq;
                *reinterpret_cast<unsigned int*>(swzb<SWZ>(dot_lds, col + j, row0)) = pd;
            }
        }
        __syncthreads();

        // S^T = K Q^T and dP^T = V dO^T   (M = kv, N = q, contract d)
        frag_c acc_s[NN], acc_dp[NN];
#pragma unroll
        for (int n = 0; n < NN; n++) {
            acc_s[n] = frag_c{0.f, 0.f, 0.f, 0.f};
            acc_dp[n] = frag_c{0.f, 0.f, 0.f, 0.f};
#pragma unroll
            for (int kk = 0; kk < KD; kk++) {
                frag_ab bq = *reinterpret_cast<const frag_ab*>(swzb<SWZ>(q_lds, n * 16 + l16, kk * 32 + lk8));
                frag_ab bdo = *reinterpret_cast<const frag_ab*>(swzb<SWZ>(do_lds, n * 16 + l16, kk * 32 + lk8));
                acc_s[n] = mfma16(ak[kk], bq, acc_s[n]);
                acc_dp[n] = mfma16(av[kk], bdo, acc_dp[n]);
            }
        }

        // P^T (C layout: row = kv, col = q); lse/delta gathered per col
        const int kvrow0 = kv_base + wave * 16 + (lane >> 4) * 4;
        float pt_vals[NN][4], dst_vals[NN][4];
#pragma unroll
        for (int n = 0; n < NN; n++) {
            int qg = q_base + n * 16 + l16;
            float lse_q = (qg < Sq) ? lse_row[qg] : INFINITY;
            float dl_q = (qg < Sq) ? dl_row[qg] : 0.f;
#pragma unroll
            for (int r = 0; r < 4; r++) {
                int kvg = kvrow0 + r;
                bool vis = (kvg < Skv) && (qg < Sq);
                if (causal) vis = vis && (kvg <= qg + causal_off);
                if (startend) vis = vis && (kvg < Skv) &&
                    (qg < startend[(long long)b * Skv + kvg]);
                float p = vis ? __expf(acc_s[n][r] * scale - lse_q) : 0.f;
                pt_vals[n][r] = p;
                dst_vals[n][r] = p * (acc_dp[n][r] - dl_q) * scale;
            }
        }

        // stage P^T; dV += P^T dO (contract q; B from dot_lds)
#pragma unroll
        for (int n = 0; n < NN; n++)
#pragma unroll
            for (int r = 0; r < 4; r++)
                *reinterpret_cast<ushort_t*>(swzb<SWZ>(pt_lds, wave * 16 + (lane >> 4) * 4 + r, n * 16 + l16)) =
                    f32_to_bf16(pt_vals[n][r]);
        // pt_lds rows are wave-private: wave-local drain, no block barrier
        asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
#pragma unroll
        for (int kk = 0; kk < BLK_M / 32; kk++) {
            frag_ab apt = *reinterpret_cast<const frag_ab*>(swzb<SWZ>(pt_lds, wave * 16 + l16, kk * 32 + lk8));
#pragma unroll
            for (int n = 0; n < ND; n++) {
                frag_ab bdot = *reinterpret_cast<const frag_ab*>(swzb<SWZ>(dot_lds, n * 16 + l16, kk * 32 + lk8));
                acc_dv[n] = mfma16(apt, bdot, acc_dv[n]);
            }
        }
        // the pt_lds reads above are this wave's own rows; overwriting with
        // dS^T next is also wave-local (no cross-wave barrier)

        // stage dS^T; dK += dS^T Q (contract q; B from qt_lds)
#pragma unroll
        for (int n = 0; n < NN; n++)
#pragma unroll
            for (int r = 0; r < 4; r++)
                *reinterpret_cast<ushort_t*>(swzb<SWZ>(pt_lds, wave * 16 + (lane >> 4) * 4 + r, n * 16 + l16)) =
                    f32_to_bf16(dst_vals[n][r]);
        asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
#pragma unroll
        for (int kk = 0; kk < BLK_M / 32; kk++) {
            frag_ab adst = *reinterpret_cast<const frag_ab*>(swzb<SWZ>(pt_lds, wave * 16 + l16, kk * 32 + lk8));
#pragma unroll
            for (int n = 0; n < ND; n++) {
                frag_ab bqt = *reinterpret_cast<const frag_ab*>(swzb<SWZ>(qt_lds, n * 16 + l16, kk * 32 + lk8));
                acc_dk[n] = mfma16(adst, bqt, acc_dk[n]);
            }
        }
        __syncthreads();
    }
    }  // g loop over the GQA group

    // store dK/dV directly in [B, Skv, Hk, D]
    const int kvrow0 = kv_base + wave * 16 + (lane >> 4) * 4;
    const long long out_row_stride = (long long)Hk * D;
#pragma unroll
    for (int r = 0; r < 4; r++) {
        int kvg = kvrow0 + r;
        if (kvg >= Skv) continue;
        long long base = ((long long)b * Skv + kvg) * out_row_stride + (long long)hk * D;
#pragma unroll
        for (int n = 0; n < ND; n++) {
            dk[base + n * 16 + l16] = f32_to_bf16(acc_dk[n][r]);
            dv[base + n * 16 + l16] = f32_to_bf16(acc_dv[n][r]);
        }
    }
}

// ---------------------------------------------------------------------------
// launchers
// ---------------------------------------------------------------------------
void launch_mfma_probe(const void* A, const void* B, float* C, hipStream_t stream) {
    hipLaunchKernelGGL(mfma_layout_probe, dim3(1), dim3(64), 0, stream,
                       (const ushort_t*)A, (const ushort_t*)B, C);
}

template <int D, int MF, bool SWZ>
static void flash_fwd_t(const void* q, const void* k, const void* v, void* o,
                        float* lse, int B, int Sq, int Skv, int Hq, int Hk,
                        float scale, bool causal, hipStream_t stream,
                        const int* startend = nullptr) {
    constexpr int BLKM = FA_WAVES * MF * 16;
    dim3 grid((Sq + BLKM - 1) / BLKM, B * Hq);
    hipLaunchKernelGGL((flash_fwd_kernel<D, MF, SWZ>), grid, dim3(FA_BLOCK), 0, stream,
                       (const ushort_t*)q, (const ushort_t*)k, (const ushort_t*)v,
                       (ushort_t*)o, lse, startend, B, Sq, Skv, Hq, Hk, scale,
                       causal ? 1 : 0);
}

bool launch_flash_fwd2_mask(const void* q, const void* k, const void* v,
                            void* o, float* lse, const int* startend,
                            int B, int Sq, int Skv, int Hq, int Hk,
                            int D, float scale, hipStream_t stream);

void launch_flash_fwd_mask(const void* q, const void* k, const void* v, void* o,
                           float* lse, const int* startend,
                           int B, int Sq, int Skv, int Hq, int Hk, int D,
                           float scale, hipStream_t stream) {
    // v2 masked kernel on the D=128 hot path (PNLP_FLASHMASK_KERNEL=1
    // forces the v1 fallback for A/B)
    static const bool force_v1 = [] {
        const char* e = getenv("PNLP_FLASHMASK_KERNEL");
        return e && e[0] == '1';
    }();
    if (!force_v1 &&
        launch_flash_fwd2_mask(q, k, v, o, lse, startend,
                               B, Sq, Skv, Hq, Hk, D, scale, stream))
        return;
    if (D == 128) flash_fwd_t<128, 1, false>(q, k, v, o, lse, B, Sq, Skv, Hq, Hk, scale, true, stream, startend);
    else if (D == 64) flash_fwd_t<64, 1, false>(q, k, v, o, lse, B, Sq, Skv, Hq, Hk, scale, true, stream, startend);
    else if (D == 32) flash_fwd_t<32, 1, false>(q, k, v, o, lse, B, Sq, Skv, Hq, Hk, scale, true, stream, startend);
}

// v2 kernel (flash_attn_v2.hip): 8-wave 32x32 swapped-QK^T structure
bool launch_flash_fwd2(const void* q, const void* k, const void* v, void* o,
                       float* lse, int B, int Sq, int Skv, int Hq, int Hk,
                       int D, float scale, bool causal, hipStream_t stream);

// variant dispatch for A/B benchmarking (variant: 0=default [v2 when D=128],
// 1=MF2+swz, 2=MF1 linear, 3=MF2 linear, 4=GLL, 5=16x16 pipe, 6=v2 forced,
// 7=v1 MF1 linear explicitly)
void launch_flash_fwd_variant(const void* q, const void* k, const void* v, void* o,
                              float* lse, int B, int Sq, int Skv, int Hq, int Hk,
                              int D, float scale, bool causal, int variant,
                              hipStream_t stream) {
    if (variant == 6) {
        if (launch_flash_fwd2(q, k, v, o, lse, B, Sq, Skv, Hq, Hk, D, scale, causal, stream))
            return;
        variant = 0;
    }
    if (variant == 0 &&
        launch_flash_fwd2(q, k, v, o, lse, B, Sq, Skv, Hq, Hk, D, scale, causal, stream))
        return;
    if (variant == 7) variant = 2;
    if (D != 128) variant = 0;
    switch (variant) {
        case 5: {
            constexpr int BLKM = FA_WAVES * 16;
            dim3 grid((Sq + BLKM - 1) / BLKM, B * Hq);
            hipLaunchKernelGGL((flash_fwd_pipe_kernel<128>), grid, dim3(FA_BLOCK), 0, stream,
                               (const ushort_t*)q, (const ushort_t*)k, (const ushort_t*)v,
                               (ushort_t*)o, lse, B, Sq, Skv, Hq, Hk, scale, causal ? 1 : 0);
            break;
        }
        case 4: {
            constexpr int BLKM = FA_WAVES * 1 * 16;
            dim3 grid((Sq + BLKM - 1) / BLKM, B * Hq);
            hipLaunchKernelGGL((flash_fwd_kernel<128, 1, false, true>), grid, dim3(FA_BLOCK), 0, stream,
                               (const ushort_t*)q, (const ushort_t*)k, (const ushort_t*)v,
                               (ushort_t*)o, lse, nullptr, B, Sq, Skv, Hq, Hk, scale, causal ? 1 : 0);
            break;
        }
        case 1: flash_fwd_t<128, 2, true>(q, k, v, o, lse, B, Sq, Skv, Hq, Hk, scale, causal, stream); break;
        case 2: flash_fwd_t<128, 1, false>(q, k, v, o, lse, B, Sq, Skv, Hq, Hk, scale, causal, stream); break;
        case 3: flash_fwd_t<128, 2, false>(q, k, v, o, lse, B, Sq, Skv, Hq, Hk, scale, causal, stream); break;
        default:
            // within-probe A/B across builds: MF1+linear is stable at
            // ~167-170 TF while MF1+swz swings 149-174 with unrelated code
            // changes (guide rule #19: co-compiled variants perturb regalloc).
            // Default to the stable variant.
            if (D == 128) flash_fwd_t<128, 1, false>(q, k, v, o, lse, B, Sq, Skv, Hq, Hk, scale, causal, stream);
            else if (D == 64) flash_fwd_t<64, 1, false>(q, k, v, o, lse, B, Sq, Skv, Hq, Hk, scale, causal, stream);
            else if (D == 32) flash_fwd_t<32, 1, false>(q, k, v, o, lse, B, Sq, Skv, Hq, Hk, scale, causal, stream);
    }
}

void launch_flash_fwd(const void* q, const void* k, const void* v, void* o,
                      float* lse, int B, int Sq, int Skv, int Hq, int Hk, int D,
                      float scale, bool causal, hipStream_t stream) {
    launch_flash_fwd_variant(q, k, v, o, lse, B, Sq, Skv, Hq, Hk, D, scale, causal, 0, stream);
}

template <int D, bool SWZ>
static void flash_bwd_t(const void* dout, const void* q, const void* k, const void* v,
                        const void* o, const float* lse, float* delta,
                        void* dq, void* dk, void* dv,
                        int B, int Sq, int Skv, int Hq, int Hk,
                        float scale, bool causal, hipStream_t stream,
                        const int* startend = nullptr) {
    long long rows = (long long)B * Sq * Hq;
    int waves_per_block = FA_BLOCK / 64;
    int dgrid = (int)((rows + waves_per_block - 1) / waves_per_block);
    hipLaunchKernelGGL(flash_bwd_delta_kernel<D>, dim3(dgrid), dim3(FA_BLOCK), 0, stream,
                       (const ushort_t*)dout, (const ushort_t*)o, delta, B, Sq, Hq);
    dim3 gq((Sq + BLK_M - 1) / BLK_M, B * Hq);
    hipLaunchKernelGGL((flash_bwd_dq_kernel<D, SWZ>), gq, dim3(FA_BLOCK), 0, stream,
                       (const ushort_t*)q, (const ushort_t*)k, (const ushort_t*)v,
                       (const ushort_t*)dout, lse, delta, (ushort_t*)dq, startend,
                       B, Sq, Skv, Hq, Hk, scale, causal ? 1 : 0);
    dim3 gkv((Skv + BLK_N - 1) / BLK_N, B * Hk);
    hipLaunchKernelGGL((flash_bwd_dkv_kernel<D, SWZ>), gkv, dim3(FA_BLOCK), 0, stream,
                       (const ushort_t*)q, (const ushort_t*)k, (const ushort_t*)v,
                       (const ushort_t*)dout, lse, delta, (ushort_t*)dk, (ushort_t*)dv,
                       startend, B, Sq, Skv, Hq, Hk, scale, causal ? 1 : 0);
}

bool launch_flash_bwd2_mask(const void* dout, const void* q, const void* k,
                            const void* v, const void* o, const float* lse,
                            float* delta, void* dq, void* dk, void* dv,
                            const int* startend,
                            int B, int Sq, int Skv, int Hq, int Hk, int D,
                            float scale, hipStream_t stream);

void launch_flash_bwd_mask(const void* dout, const void* q, const void* k, const void* v,
                           const void* o, const float* lse, float* delta,
                           void* dq, void* dk, void* dv, const int* startend,
                           int B, int Sq, int Skv, int Hq, int Hk, int D,
                           float scale, hipStream_t stream) {
    static const bool force_v1 = [] {
        const char* e = getenv("PNLP_FLASHMASK_KERNEL");
        return e && e[0] == '1';
    }();
    if (!force_v1 &&
        launch_flash_bwd2_mask(dout, q, k, v, o, lse, delta, dq, dk, dv,
                               startend, B, Sq, Skv, Hq, Hk, D, scale, stream))
        return;
    if (D == 128) flash_bwd_t<128, false>(dout, q, k, v, o, lse, delta, dq, dk, dv, B, Sq, Skv, Hq, Hk, scale, true, stream, startend);
    else if (D == 64) flash_bwd_t<64, false>(dout, q, k, v, o, lse, delta, dq, dk, dv, B, Sq, Skv, Hq, Hk, scale, true, stream, startend);
    else if (D == 32) flash_bwd_t<32, false>(dout, q, k, v, o, lse, delta, dq, dk, dv, B, Sq, Skv, Hq, Hk, scale, true, stream, startend);
}

// v2 (flash_attn_bwd_v2.hip): 8-wave 32x32 swapped-operand structure
bool launch_flash_bwd2(const void* dout, const void* q, const void* k, const void* v,
                       const void* o, const float* lse, float* delta,
                       void* dq, void* dk, void* dv,
                       int B, int Sq, int Skv, int Hq, int Hk, int D,
                       float scale, bool causal, hipStream_t stream);

void launch_flash_bwd_variant(const void* dout, const void* q, const void* k, const void* v,
                              const void* o, const float* lse, float* delta,
                              void* dq, void* dk, void* dv,
                              int B, int Sq, int Skv, int Hq, int Hk, int D,
                              float scale, bool causal, int variant, hipStream_t stream) {
    // variant 0 = default (v2 when D=128), 2 = v2 forced, 3 = v1 linear,
    // 1 = v1 swizzled
    if (variant == 2 || variant == 0) {
        if (launch_flash_bwd2(dout, q, k, v, o, lse, delta, dq, dk, dv,
                              B, Sq, Skv, Hq, Hk, D, scale, causal, stream))
            return;
        variant = 0;
    }
    if (variant == 3) variant = 0;
    // within-probe A/B: linear 202 TF > swizzled 172 for the bwd pair
    // (their extra LDS tiles make the swizzle's address VALU the bottleneck)
    if (variant == 0 && D == 128) {
        flash_bwd_t<128, false>(dout, q, k, v, o, lse, delta, dq, dk, dv, B, Sq, Skv, Hq, Hk, scale, causal, stream);
        return;
    }
    if (D == 128) flash_bwd_t<128, true>(dout, q, k, v, o, lse, delta, dq, dk, dv, B, Sq, Skv, Hq, Hk, scale, causal, stream);
    else if (D == 64) flash_bwd_t<64, false>(dout, q, k, v, o, lse, delta, dq, dk, dv, B, Sq, Skv, Hq, Hk, scale, causal, stream);
    else if (D == 32) flash_bwd_t<32, false>(dout, q, k, v, o, lse, delta, dq, dk, dv, B, Sq, Skv, Hq, Hk, scale, causal, stream);
}

void launch_flash_bwd(const void* dout, const void* q, const void* k, const void* v,
                      const void* o, const float* lse, float* delta,
                      void* dq, void* dk, void* dv,
                      int B, int Sq, int Skv, int Hq, int Hk, int D,
                      float scale, bool causal, hipStream_t stream) {
    launch_flash_bwd_variant(dout, q, k, v, o, lse, delta, dq, dk, dv,
                             B, Sq, Skv, Hq, Hk, D, scale, causal, 0, stream);
}
