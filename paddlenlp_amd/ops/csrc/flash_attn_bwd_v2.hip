// FlashAttention-2 backward v2 — 8-wave 32x32 MFMA structure for gfx950.
//
// Same technique set as the v2 forward (flash_attn_v2.hip): 32x32x16
// MFMA, XOR-swizzled LDS staging, async-stage split, in-register
// P/dS-to-fragment conversion via permlane32_swap.
//
// The key structural choice: backward has NO online softmax — lse and
// delta are precomputed arrays — so the C-layout orientation of the two
// recompute GEMMs is free.  Each kernel picks the orientation that makes
// P and dS directly convertible (cvt-pack + permlane32_swap, guide T12)
// into the A/B fragments of the GEMMs that consume them:
//
//   dKV kernel (kv stationary): compute S and dP in C-layout [q][kv]
//     (col = kv = lane's own column).  The T12 swap then yields, per
//     lane, P^T[kv = l&31][q-slice] — exactly the A-fragment of
//     dV += P^T·dO and dK += dS^T·Q.  No LDS round trip for P/dS.
//   dQ kernel (q stationary): compute S^T and dP^T in C-layout [kv][q]
//     (col = q).  The swap yields dS[q = l&31][kv-slice] — the
//     A-fragment of dQ += dS·K.
//
// v1 (flash_attn.hip) staged P^T/dS^T through LDS between every GEMM
// pair; measured A/B at the bench shape: v1 191 TF/s -> v2 (this file).
#include "common.h"

#define FB2_WAVES 8
#define FB2_BLOCK (FB2_WAVES * 64)

typedef short8v frag8;

static __device__ __forceinline__ f32x16 mfma32b(frag8 a, frag8 b, f32x16 c) {
    return __builtin_amdgcn_mfma_f32_32x32x16_bf16(a, b, c, 0, 0, 0);
}

static __device__ __forceinline__ int crow32b(int r, int hi) {
    return (r & 3) + 8 * (r >> 2) + 4 * hi;
}

template <int LD>
static __device__ __forceinline__ char* swzb2(ushort_t* base, int row, int col_elem) {
    return reinterpret_cast<char*>(base) +
           (((row * LD + col_elem) * 2) ^ ((row & 7) << 4));
}

static __device__ __forceinline__ unsigned packbf(float a, float b) {
    return (unsigned)f32_to_bf16(a) | ((unsigned)f32_to_bf16(b) << 16);
}

// T12: convert one 16-row slice (regs rb..rb+7 of a 32x32 C tile) into the
// 8 packed bf16 values each lane needs for the transpose-contraction
// fragment: lane ends up with X[other = hi*8 + j] at its own column.
static __device__ __forceinline__ frag8 t12_convert(const f32x16& p, int rb) {
    unsigned w[4];
    {
        unsigned lo0 = packbf(p[rb + 0], p[rb + 1]);
        unsigned hi0 = packbf(p[rb + 4], p[rb + 5]);
        auto rr = __builtin_amdgcn_permlane32_swap(lo0, hi0, false, false);
        w[0] = rr[0]; w[2] = rr[1];
    }
    {
        unsigned lo1 = packbf(p[rb + 2], p[rb + 3]);
        unsigned hi1 = packbf(p[rb + 6], p[rb + 7]);
        auto rr = __builtin_amdgcn_permlane32_swap(lo1, hi1, false, false);
        w[1] = rr[0]; w[3] = rr[1];
    }
    frag8 f;
#pragma unroll
    for (int j = 0; j < 4; j++) {
        f[j * 2] = (short)(w[j] & 0xffff);
        f[j * 2 + 1] = (short)(w[j] >> 16);
    }
    return f;
}

// ---------------------------------------------------------------------------
// delta[b,h,q] = rowsum(dO * O)
// ---------------------------------------------------------------------------
template <int D>
__global__ void flash_bwd2_delta_kernel(
    const ushort_t* __restrict__ dout, const ushort_t* __restrict__ o,
    float* __restrict__ delta, int B, int Sq, int Hq) {
    long long row = (long long)blockIdx.x * (blockDim.x >> 6) + (threadIdx.x >> 6);
    long long total = (long long)B * Sq * Hq;
    if (row >= total) return;
    int lane = threadIdx.x & 63;
    const ushort_t* dor = dout + row * D;
    const ushort_t* orow = o + row * D;
    float acc = 0.f;
#pragma unroll
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
// dQ kernel: q stationary (8 waves x 32 q = 256 q rows/block), kv iterated
// in tiles of 64.  Orientation: S^T and dP^T in C-layout [kv][q] so lse and
// delta are per-lane scalars (q = lane&31) and the T12 swap hands dS to the
// dQ MFMA directly.
// ---------------------------------------------------------------------------
template <int D, bool MASKED = false>
__global__ __launch_bounds__(FB2_BLOCK) void flash_bwd2_dq_kernel(
    const ushort_t* __restrict__ q, const ushort_t* __restrict__ k,
    const ushort_t* __restrict__ v, const ushort_t* __restrict__ dout,
    const float* __restrict__ lse, const float* __restrict__ delta,
    ushort_t* __restrict__ dq, const int* __restrict__ startend,
    int B, int Sq, int Skv, int Hq, int Hk, float scale, int causal) {
    constexpr int BLKN = 64;
    constexpr int DSTEPS = D / 16;
    constexpr int NDT = D / 32;
    constexpr int BLKM = FB2_WAVES * 32;

    // double-buffered (one barrier per KV tile; see flash_attn_v2.hip)
    __shared__ ushort_t k_lds[2][BLKN * D];    // row-major [kv][d]
    __shared__ ushort_t v_lds[2][BLKN * D];    // row-major [kv][d]
    __shared__ ushort_t kt_lds[2][D * BLKN];   // transposed [d][kv]
    __shared__ int se_lds[2][MASKED ? BLKN : 1];  // FlashMask bounds

    const int tid = threadIdx.x;
    const int wave = tid >> 6;
    const int lane = tid & 63;
    const int l32 = lane & 31;
    const int hi = lane >> 5;

    // XCD-aware remap (see flash_attn_v2.hip): keep one (b,h)'s tiles
    // on one XCD so the shared K/V (dq) tiles re-hit that XCD's L2
    int qt, bh;
    if ((gridDim.y & 7) == 0) {
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
    const int qw = q_base + wave * 32;
    const int qg = qw + l32;
    const int causal_off = Skv - Sq;

    const long long q_row_stride = (long long)Hq * D;
    const long long kv_row_stride = (long long)Hk * D;
    const ushort_t* q_ptr = q + ((long long)b * Sq * Hq + hq) * D;
    const ushort_t* do_ptr = dout + ((long long)b * Sq * Hq + hq) * D;
    const ushort_t* k_ptr = k + ((long long)b * Skv * Hk + hk) * D;
    const ushort_t* v_ptr = v + ((long long)b * Skv * Hk + hk) * D;

    // stationary per-lane: Q^T and dO^T B-fragments (one q row each)
    frag8 aq[DSTEPS], ado[DSTEPS];
    float lse_q = INFINITY, dl_q = 0.f;
    if (qg < Sq) {
#pragma unroll
        for (int kk = 0; kk < DSTEPS; kk++) {
            aq[kk] = *reinterpret_cast<const frag8*>(
                q_ptr + (long long)qg * q_row_stride + kk * 16 + hi * 8);
            ado[kk] = *reinterpret_cast<const frag8*>(
                do_ptr + (long long)qg * q_row_stride + kk * 16 + hi * 8);
        }
        lse_q = lse[((long long)b * Hq + hq) * Sq + qg];
        dl_q = delta[((long long)b * Hq + hq) * Sq + qg];
    } else {
#pragma unroll
        for (int kk = 0; kk < DSTEPS; kk++) { aq[kk] = frag8{0}; ado[kk] = frag8{0}; }
    }

    f32x16 acc_dq[NDT];
#pragma unroll
    for (int n = 0; n < NDT; n++)
#pragma unroll
        for (int r = 0; r < 16; r++) acc_dq[n][r] = 0.f;

    int n_kv_tiles = (Skv + BLKN - 1) / BLKN;
    if (causal) {
        int max_kv = q_base + BLKM - 1 + causal_off;
        int lim = (max_kv + BLKN) / BLKN;
        n_kv_tiles = min(n_kv_tiles, max(lim, 0));
    }

    const int s_rowp = tid >> 4;
    const int s_col = (tid & 15) * 8;
    const int s_row0 = s_rowp * 2;
    short8v sk0, sk1, sv0, sv1;

    auto load_tile = [&](int kv_base) {
        int g0 = kv_base + s_row0;
        sk0 = short8v{0,0,0,0,0,0,0,0}; sk1 = sk0; sv0 = sk0; sv1 = sk0;
        if (g0 < Skv) {
            sk0 = *reinterpret_cast<const short8v*>(k_ptr + (long long)g0 * kv_row_stride + s_col);
            sv0 = *reinterpret_cast<const short8v*>(v_ptr + (long long)g0 * kv_row_stride + s_col);
        }
        if (g0 + 1 < Skv) {
            sk1 = *reinterpret_cast<const short8v*>(k_ptr + (long long)(g0 + 1) * kv_row_stride + s_col);
            sv1 = *reinterpret_cast<const short8v*>(v_ptr + (long long)(g0 + 1) * kv_row_stride + s_col);
        }
    };
    auto write_tile = [&](int buf, int kv_base) {
        if (MASKED && tid < BLKN) {
            int g = kv_base + tid;
            se_lds[buf][tid] =
                (g < Skv) ? startend[(long long)b * Skv + g] : 0;
        }
        *reinterpret_cast<short8v*>(swzb2<D>(k_lds[buf], s_row0, s_col)) = sk0;
        *reinterpret_cast<short8v*>(swzb2<D>(k_lds[buf], s_row0 + 1, s_col)) = sk1;
        *reinterpret_cast<short8v*>(swzb2<D>(v_lds[buf], s_row0, s_col)) = sv0;
        *reinterpret_cast<short8v*>(swzb2<D>(v_lds[buf], s_row0 + 1, s_col)) = sv1;
#pragma unroll
        for (int j = 0; j < 8; j++) {
            unsigned p32 = ((unsigned)(unsigned short)sk0[j]) |
                           (((unsigned)(unsigned short)sk1[j]) << 16);
            *reinterpret_cast<unsigned*>(swzb2<BLKN>(kt_lds[buf], s_col + j, s_row0)) = p32;
        }
    };

    // per-wave bound reduce (see flash_attn_v2.hip tile_bounds)
    auto tile_se_max = [&](int kv_base) -> int {
        int g = kv_base + lane;
        int v = (g < Skv) ? startend[(long long)b * Skv + g] : 0;
#pragma unroll
        for (int off = 32; off; off >>= 1)
            v = max(v, __shfl_xor(v, off, 64));
        return v;
    };

    load_tile(0);
    write_tile(0, 0);
    int se_max_cur = MASKED ? tile_se_max(0) : 0x7fffffff;
    __syncthreads();

    for (int kvt = 0; kvt < n_kv_tiles; kvt++) {
        const int kv_base = kvt * BLKN;
        const int cur = kvt & 1;
        bool wave_skip =
            causal && (kv_base > qw + 31 + causal_off);
        // FlashMask whole-tile skip (see fwd): no bound exceeds this
        // wave's first q row -> nothing visible
        if (MASKED && qw >= se_max_cur) wave_skip = true;

        // two 32-kv sub-iterations: one (S^T, dP^T) register pair live at
        // a time (register-pressure; see dkv kernel note)
        auto process_sub = [&](int sub) {
            f32x16 st, dp;
#pragma unroll
            for (int r = 0; r < 16; r++) { st[r] = 0.f; dp[r] = 0.f; }
            __builtin_amdgcn_s_setprio(1);
#pragma unroll
            for (int kk = 0; kk < DSTEPS; kk++) {
                frag8 ak = *reinterpret_cast<const frag8*>(
                    swzb2<D>(k_lds[cur], sub * 32 + l32, kk * 16 + hi * 8));
                frag8 av = *reinterpret_cast<const frag8*>(
                    swzb2<D>(v_lds[cur], sub * 32 + l32, kk * 16 + hi * 8));
                st = mfma32b(ak, aq[kk], st);
                dp = mfma32b(av, ado[kk], dp);
            }
            __builtin_amdgcn_s_setprio(0);

            // P^T = exp(S^T*scale - lse_q); dS^T = P^T*(dP^T - delta_q)*
            // scale (all per-lane: q = l32)
#pragma unroll
            for (int r = 0; r < 16; r++) {
                int row = sub * 32 + crow32b(r, hi);
                int kvg = kv_base + row;
                bool vis = (kvg < Skv) && (qg < Sq) &&
                           (!causal || kvg <= qg + causal_off);
                if (MASKED) vis = vis && (qg < se_lds[cur][row]);
                float p = vis ? __expf(st[r] * scale - lse_q) : 0.f;
                st[r] = p * (dp[r] - dl_q) * scale;  // = dS^T
            }

            // dQ += dS·K: A = T12(dS^T) (lane: dS[q=l32][kv-slice]),
            // B = K[kv][d] column-slices from kt_lds
            __builtin_amdgcn_s_setprio(1);
#pragma unroll
            for (int ksl = 0; ksl < 2; ksl++) {
                const int ks = sub * 2 + ksl;
                frag8 ads = t12_convert(st, ksl * 8);
#pragma unroll
                for (int n = 0; n < NDT; n++) {
                    frag8 bkf = *reinterpret_cast<const frag8*>(
                        swzb2<BLKN>(kt_lds[cur], n * 32 + l32, ks * 16 + hi * 8));
                    acc_dq[n] = mfma32b(ads, bkf, acc_dq[n]);
                }
            }
            __builtin_amdgcn_s_setprio(0);
        };

        if (!wave_skip) process_sub(0);
        int se_max_next = 0x7fffffff;
        bool stage_skip = false;
        if (kvt + 1 < n_kv_tiles) {
            if (MASKED) {
                se_max_next = tile_se_max(kv_base + BLKN);
                // whole block past every bound: skip the K/V staging too
                stage_skip = (q_base >= se_max_next);
            }
            if (!stage_skip) load_tile(kv_base + BLKN);
        }
        if (!wave_skip) process_sub(1);

        if (kvt + 1 < n_kv_tiles && !stage_skip)
            write_tile(cur ^ 1, kv_base + BLKN);
        if (MASKED) se_max_cur = se_max_next;
        __syncthreads();
    }

    // epilogue: acc_dq C-layout [q][d] (col = d, row = q)
#pragma unroll
    for (int r = 0; r < 16; r++) {
        int qrow = qw + crow32b(r, hi);
        if (qrow >= Sq) continue;
        ushort_t* dqr = dq + ((long long)b * Sq + qrow) * q_row_stride + (long long)hq * D;
#pragma unroll
        for (int n = 0; n < NDT; n++)
            dqr[n * 32 + l32] = f32_to_bf16(acc_dq[n][r]);
    }
}

// ---------------------------------------------------------------------------
// dKV kernel: kv stationary (8 waves x 32 kv = 256 kv rows/block), q iterated
// in tiles of 64 across the whole GQA group.  Orientation: S and dP in
// C-layout [q][kv] (col = kv) so T12 feeds dV += P^T·dO and dK += dS^T·Q.
// ---------------------------------------------------------------------------
template <int D, bool MASKED = false>
__global__ __launch_bounds__(FB2_BLOCK, 1) void flash_bwd2_dkv_kernel(
    const ushort_t* __restrict__ q, const ushort_t* __restrict__ k,
    const ushort_t* __restrict__ v, const ushort_t* __restrict__ dout,
    const float* __restrict__ lse, const float* __restrict__ delta,
    ushort_t* __restrict__ dk, ushort_t* __restrict__ dv,
    const int* __restrict__ startend,
    int B, int Sq, int Skv, int Hq, int Hk, float scale, int causal) {
    constexpr int BLKQ = 64;               // q tile
    constexpr int DSTEPS = D / 16;
    constexpr int NDT = D / 32;
    constexpr int BLKKV = FB2_WAVES * 32;  // 256 kv rows per block

    // double-buffered q-side tiles (one barrier per q tile)
    __shared__ ushort_t q_lds[2][BLKQ * D];     // row-major [q][d]
    __shared__ ushort_t do_lds[2][BLKQ * D];    // row-major [q][d]
    __shared__ ushort_t qt_lds[2][D * BLKQ];    // transposed [d][q]
    __shared__ ushort_t dot_lds[2][D * BLKQ];   // transposed [d][q]
    __shared__ float lse_lds[2][BLKQ];
    __shared__ float dl_lds[2][BLKQ];

    const int tid = threadIdx.x;
    const int wave = tid >> 6;
    const int lane = tid & 63;
    const int l32 = lane & 31;
    const int hi = lane >> 5;

    // XCD-aware remap: one (b,h)'s kv-tiles per XCD (shared Q/dO reuse)
    int kvt, bh;
    if ((gridDim.y & 7) == 0) {
        const int flat = blockIdx.x + gridDim.x * blockIdx.y;
        const int idx = flat >> 3;
        kvt = idx % gridDim.x;
        bh = (flat & 7) * (gridDim.y >> 3) + idx / gridDim.x;
    } else {
        kvt = blockIdx.x;
        bh = blockIdx.y;
    }
    const int b = bh / Hk, hk = bh % Hk;
    const int G = Hq / Hk;
    const int kv_base = kvt * BLKKV;
    const int kvw = kv_base + wave * 32;     // this wave's first kv row
    const int kvg_lane = kvw + l32;          // lane's kv column
    const int causal_off = Skv - Sq;
    // FlashMask: lane owns one kv column -> one bound scalar; the
    // wave-max bound drives whole-q-tile skipping and the BLOCK max
    // (published through LDS before the first barrier) lets the whole
    // workgroup skip staging q tiles no kv row can see
    __shared__ int blk_max_lds[MASKED ? FB2_WAVES : 1];
    int kv_end_lane = 0x7fffffff;
    int wave_max_end = 0x7fffffff;
    int block_max_end = 0x7fffffff;
    if (MASKED) {
        kv_end_lane = (kvg_lane < Skv)
            ? startend[(long long)b * Skv + kvg_lane] : 0;
        wave_max_end = kv_end_lane;
#pragma unroll
        for (int off = 32; off; off >>= 1)
            wave_max_end = max(wave_max_end,
                               __shfl_xor(wave_max_end, off, 64));
        if (lane == 0) blk_max_lds[wave] = wave_max_end;
    }

    const long long q_row_stride = (long long)Hq * D;
    const long long kv_row_stride = (long long)Hk * D;
    const ushort_t* k_ptr = k + ((long long)b * Skv * Hk + hk) * D;
    const ushort_t* v_ptr = v + ((long long)b * Skv * Hk + hk) * D;

    // stationary per-lane: K^T and V^T B-fragments (one kv row each)
    frag8 bk[DSTEPS], bv[DSTEPS];
    if (kvg_lane < Skv) {
#pragma unroll
        for (int kk = 0; kk < DSTEPS; kk++) {
            bk[kk] = *reinterpret_cast<const frag8*>(
                k_ptr + (long long)kvg_lane * kv_row_stride + kk * 16 + hi * 8);
            bv[kk] = *reinterpret_cast<const frag8*>(
                v_ptr + (long long)kvg_lane * kv_row_stride + kk * 16 + hi * 8);
        }
    } else {
#pragma unroll
        for (int kk = 0; kk < DSTEPS; kk++) { bk[kk] = frag8{0}; bv[kk] = frag8{0}; }
    }

    f32x16 acc_dk[NDT], acc_dv[NDT];
#pragma unroll
    for (int n = 0; n < NDT; n++)
#pragma unroll
        for (int r = 0; r < 16; r++) { acc_dk[n][r] = 0.f; acc_dv[n][r] = 0.f; }

    // q tiles: under causal only q >= kv_base - off contribute
    int qt0 = 0;
    if (causal) qt0 = max(0, (kv_base - causal_off) / BLKQ);
    const int n_q_tiles = (Sq + BLKQ - 1) / BLKQ;

    const int s_rowp = tid >> 4;
    const int s_col = (tid & 15) * 8;
    const int s_row0 = s_rowp * 2;
    short8v sq0, sq1, sd0, sd1;

    for (int g = 0; g < G; g++) {
        const int hq = hk * G + g;
        const ushort_t* q_ptr = q + ((long long)b * Sq * Hq + hq) * D;
        const ushort_t* do_ptr = dout + ((long long)b * Sq * Hq + hq) * D;
        const float* lse_row = lse + ((long long)b * Hq + hq) * Sq;
        const float* dl_row = delta + ((long long)b * Hq + hq) * Sq;

        auto load_qtile = [&](int q_tb) {
            int g0 = q_tb + s_row0;
            sq0 = short8v{0,0,0,0,0,0,0,0}; sq1 = sq0; sd0 = sq0; sd1 = sq0;
            if (g0 < Sq) {
                sq0 = *reinterpret_cast<const short8v*>(q_ptr + (long long)g0 * q_row_stride + s_col);
                sd0 = *reinterpret_cast<const short8v*>(do_ptr + (long long)g0 * q_row_stride + s_col);
            }
            if (g0 + 1 < Sq) {
                sq1 = *reinterpret_cast<const short8v*>(q_ptr + (long long)(g0 + 1) * q_row_stride + s_col);
                sd1 = *reinterpret_cast<const short8v*>(do_ptr + (long long)(g0 + 1) * q_row_stride + s_col);
            }
        };
        auto write_qtile = [&](int q_tb, int buf) {
            *reinterpret_cast<short8v*>(swzb2<D>(q_lds[buf], s_row0, s_col)) = sq0;
            *reinterpret_cast<short8v*>(swzb2<D>(q_lds[buf], s_row0 + 1, s_col)) = sq1;
            *reinterpret_cast<short8v*>(swzb2<D>(do_lds[buf], s_row0, s_col)) = sd0;
            *reinterpret_cast<short8v*>(swzb2<D>(do_lds[buf], s_row0 + 1, s_col)) = sd1;
#pragma unroll
            for (int j = 0; j < 8; j++) {
                unsigned pq = ((unsigned)(unsigned short)sq0[j]) |
                              (((unsigned)(unsigned short)sq1[j]) << 16);
                unsigned pd = ((unsigned)(unsigned short)sd0[j]) |
                              (((unsigned)(unsigned short)sd1[j]) << 16);
                *reinterpret_cast<unsigned*>(swzb2<BLKQ>(qt_lds[buf], s_col + j, s_row0)) = pq;
                *reinterpret_cast<unsigned*>(swzb2<BLKQ>(dot_lds[buf], s_col + j, s_row0)) = pd;
            }
            if (tid < BLKQ) {
                int qgl = q_tb + tid;
                lse_lds[buf][tid] = (qgl < Sq) ? lse_row[qgl] : INFINITY;
                dl_lds[buf][tid] = (qgl < Sq) ? dl_row[qgl] : 0.f;
            }
        };

        load_qtile(qt0 * BLKQ);
        write_qtile(qt0 * BLKQ, qt0 & 1);
        __syncthreads();
        if (MASKED) {
            // block max of the per-wave bounds (published above): q tiles
            // wholly past it are invisible to EVERY kv row in the block
            block_max_end = blk_max_lds[0];
#pragma unroll
            for (int w = 1; w < FB2_WAVES; w++)
                block_max_end = max(block_max_end, blk_max_lds[w]);
        }

        for (int qt = qt0; qt < n_q_tiles; qt++) {
            const int q_tb = qt * BLKQ;
            const int cur = qt & 1;
            // skip q tiles fully below this wave's causal diagonal, or
            // (FlashMask) entirely past every bound this wave holds
            bool wave_skip =
                causal && (q_tb + BLKQ - 1 + causal_off < kvw);
            if (MASKED && q_tb >= wave_max_end) wave_skip = true;

            // process the 64-q tile as two 32-q sub-iterations: only one
            // (S, dP) register pair is live at a time, which keeps the
            // file at {acc 128 + K/V frags 64 + state 32} and spill-free
            if (!wave_skip) {
#pragma unroll 1
                for (int sub = 0; sub < 2; sub++) {
                    f32x16 st, dp;
#pragma unroll
                    for (int r = 0; r < 16; r++) { st[r] = 0.f; dp[r] = 0.f; }
                    __builtin_amdgcn_s_setprio(1);
#pragma unroll
                    for (int kk = 0; kk < DSTEPS; kk++) {
                        frag8 aqf = *reinterpret_cast<const frag8*>(
                            swzb2<D>(q_lds[cur], sub * 32 + l32, kk * 16 + hi * 8));
                        frag8 adf = *reinterpret_cast<const frag8*>(
                            swzb2<D>(do_lds[cur], sub * 32 + l32, kk * 16 + hi * 8));
                        st = mfma32b(aqf, bk[kk], st);
                        dp = mfma32b(adf, bv[kk], dp);
                    }
                    __builtin_amdgcn_s_setprio(0);

                    // P = exp(S*scale - lse[q]) overwrites S; dS = P*(dP -
                    // delta[q])*scale overwrites dP.  Rows are q (wave-
                    // uniform per reg): lse/delta broadcast from LDS.
#pragma unroll
                    for (int r = 0; r < 16; r++) {
                        int qrow = sub * 32 + crow32b(r, hi);
                        int qgl = q_tb + qrow;
                        float ls = lse_lds[cur][qrow];
                        float dl = dl_lds[cur][qrow];
                        bool vis = (qgl < Sq) && (kvg_lane < Skv) &&
                                   (!causal || kvg_lane <= qgl + causal_off);
                        if (MASKED) vis = vis && (qgl < kv_end_lane);
                        float p = (vis && ls != INFINITY)
                                      ? __expf(st[r] * scale - ls) : 0.f;
                        st[r] = p;
                        dp[r] = p * (dp[r] - dl) * scale;  // dS
                    }

                    // dV += P^T·dO, dK += dS^T·Q (contract q: the 2
                    // k-steps of this sub-tile)
                    __builtin_amdgcn_s_setprio(1);
#pragma unroll
                    for (int ksl = 0; ksl < 2; ksl++) {
                        const int ks = sub * 2 + ksl;
                        frag8 apt = t12_convert(st, ksl * 8);
                        frag8 ads = t12_convert(dp, ksl * 8);
#pragma unroll
                        for (int n = 0; n < NDT; n++) {
                            frag8 bdo = *reinterpret_cast<const frag8*>(
                                swzb2<BLKQ>(dot_lds[cur], n * 32 + l32, ks * 16 + hi * 8));
                            frag8 bq = *reinterpret_cast<const frag8*>(
                                swzb2<BLKQ>(qt_lds[cur], n * 32 + l32, ks * 16 + hi * 8));
                            acc_dv[n] = mfma32b(apt, bdo, acc_dv[n]);
                            acc_dk[n] = mfma32b(ads, bq, acc_dk[n]);
                        }
                    }
                    __builtin_amdgcn_s_setprio(0);
                }
            }

            if (qt + 1 < n_q_tiles &&
                !(MASKED && q_tb + BLKQ >= block_max_end)) {
                // no register prefetch (register file at capacity); the
                // load+write go straight into the OTHER buffer, so only
                // one barrier per tile.  FlashMask: q tiles past the
                // block's max bound skip the staging entirely
                load_qtile(q_tb + BLKQ);
                write_qtile(q_tb + BLKQ, cur ^ 1);
            }
            __syncthreads();
        }
        __syncthreads();   // before the next g reuses the LDS tiles
    }

    // epilogue: acc C-layout [kv][d] (col = d, row = kv within the wave)
    const long long out_row_stride = (long long)Hk * D;
#pragma unroll
    for (int r = 0; r < 16; r++) {
        int kvr = kvw + crow32b(r, hi);
        if (kvr >= Skv) continue;
        long long base = ((long long)b * Skv + kvr) * out_row_stride + (long long)hk * D;
#pragma unroll
        for (int n = 0; n < NDT; n++) {
            dk[base + n * 32 + l32] = f32_to_bf16(acc_dk[n][r]);
            dv[base + n * 32 + l32] = f32_to_bf16(acc_dv[n][r]);
        }
    }
}

// ---------------------------------------------------------------------------
// launcher
// ---------------------------------------------------------------------------
bool launch_flash_bwd2(const void* dout, const void* q, const void* k, const void* v,
                       const void* o, const float* lse, float* delta,
                       void* dq, void* dk, void* dv,
                       int B, int Sq, int Skv, int Hq, int Hk, int D,
                       float scale, bool causal, hipStream_t stream) {
    if (D != 128 || (Hq % Hk) != 0) return false;
    long long rows = (long long)B * Sq * Hq;
    int waves_per_block = FB2_BLOCK / 64;
    int dgrid = (int)((rows + waves_per_block - 1) / waves_per_block);
    hipLaunchKernelGGL(flash_bwd2_delta_kernel<128>, dim3(dgrid), dim3(FB2_BLOCK), 0, stream,
                       (const ushort_t*)dout, (const ushort_t*)o, delta, B, Sq, Hq);
    dim3 gq((Sq + 255) / 256, B * Hq);
    hipLaunchKernelGGL((flash_bwd2_dq_kernel<128, false>), gq, dim3(FB2_BLOCK), 0, stream,
                       (const ushort_t*)q, (const ushort_t*)k, (const ushort_t*)v,
                       (const ushort_t*)dout, lse, delta, (ushort_t*)dq,
                       (const int*)nullptr,
                       B, Sq, Skv, Hq, Hk, scale, causal ? 1 : 0);
    dim3 gkv((Skv + 255) / 256, B * Hk);
    hipLaunchKernelGGL((flash_bwd2_dkv_kernel<128, false>), gkv, dim3(FB2_BLOCK), 0, stream,
                       (const ushort_t*)q, (const ushort_t*)k, (const ushort_t*)v,
                       (const ushort_t*)dout, lse, delta, (ushort_t*)dk, (ushort_t*)dv,
                       (const int*)nullptr,
                       B, Sq, Skv, Hq, Hk, scale, causal ? 1 : 0);
    return true;
}

bool launch_flash_bwd2_mask(const void* dout, const void* q, const void* k,
                            const void* v, const void* o, const float* lse,
                            float* delta, void* dq, void* dk, void* dv,
                            const int* startend,
                            int B, int Sq, int Skv, int Hq, int Hk, int D,
                            float scale, hipStream_t stream) {
    if (D != 128 || (Hq % Hk) != 0) return false;
    long long rows = (long long)B * Sq * Hq;
    int waves_per_block = FB2_BLOCK / 64;
    int dgrid = (int)((rows + waves_per_block - 1) / waves_per_block);
    hipLaunchKernelGGL(flash_bwd2_delta_kernel<128>, dim3(dgrid), dim3(FB2_BLOCK), 0, stream,
                       (const ushort_t*)dout, (const ushort_t*)o, delta, B, Sq, Hq);
    dim3 gq((Sq + 255) / 256, B * Hq);
    hipLaunchKernelGGL((flash_bwd2_dq_kernel<128, true>), gq, dim3(FB2_BLOCK), 0, stream,
                       (const ushort_t*)q, (const ushort_t*)k, (const ushort_t*)v,
                       (const ushort_t*)dout, lse, delta, (ushort_t*)dq, startend,
                       B, Sq, Skv, Hq, Hk, scale, 1);
    dim3 gkv((Skv + 255) / 256, B * Hk);
    hipLaunchKernelGGL((flash_bwd2_dkv_kernel<128, true>), gkv, dim3(FB2_BLOCK), 0, stream,
                       (const ushort_t*)q, (const ushort_t*)k, (const ushort_t*)v,
                       (const ushort_t*)dout, lse, delta, (ushort_t*)dk, (ushort_t*)dv,
                       startend, B, Sq, Skv, Hq, Hk, scale, 1);
    return true;
}
