// FlashAttention-2 forward v2 — 8-wave 32x32 MFMA structure for gfx950.
//
// Replaces the v1 4-wave 16x16 kernel (flash_attn.hip) on the causal/dense
// D=128 hot path.  Structure follows the CDNA4 guide's verified attention
// ladder (§B "8-warp 32x32", ~900 TF at the GQA bench shape):
//
//   * mfma_f32_32x32x16_bf16 (8-cycle, 32 KFLOP per issue) instead of
//     16x16x32: 2x the FLOP per instruction slot at equal issue rate.
//   * SWAPPED QK^T: compute S^T = K·Q^T so the MFMA C-layout
//     (col = lane&31 = q, row = kv) puts a full softmax row (all kv for
//     one q) into ONE lane's registers — the row max/sum needs no
//     cross-lane shuffles beyond a single permlane32_swap combining the
//     two lane-halves.  (v1 spent 8 shfl_xor rounds per 4-row group.)
//   * In-register P -> PV A-fragment conversion via packed bf16 +
//     permlane32_swap (guide T12): no p_lds round trip at all.
//   * K and V^T staged in XOR-swizzled LDS (guide T2/G4: row-major
//     [*][128] bf16 read as column-slices is a 32-way bank conflict;
//     byte ^= (row&7)<<4 spreads it across 8 slots).
//   * Async-stage split (guide T14): next tile's global loads are issued
//     before this tile's softmax+PV so ~500-cycle HBM latency hides under
//     compute; ds_writes land after the barrier.
//   * Defer-max rescale (guide T13, THR=8): O-rescale only runs when the
//     running max actually grew, saving the per-tile O sweep.
//   * s_setprio(1) around the MFMA clusters (guide T5).
//
// Reference behavior: paddle `flash_attention` fused op;
// numerics oracle: tests/test_ops_gpu.py vs fp32 torch attention.
#include "common.h"

#define FA2_WAVES 8
#define FA2_BLOCK (FA2_WAVES * 64)
#define FA2_BLKN 64
#define FA2_QW 32            // q rows per wave
#define FA2_BLKM (FA2_WAVES * FA2_QW)
#define DEFER_THR 8.0f

typedef short8v frag8;

__device__ __forceinline__ f32x16 mfma32(frag8 a, frag8 b, f32x16 c) {
    return __builtin_amdgcn_mfma_f32_32x32x16_bf16(a, b, c, 0, 0, 0);
}

// C/D layout of mfma_f32_32x32x16_bf16 (guide §3, HW-verified m74/m101):
//   C[row][col]: col = lane&31, row = (reg&3) + 8*(reg>>2) + 4*(lane>>5)
__device__ __forceinline__ int crow32(int r, int hi) {
    return (r & 3) + 8 * (r >> 2) + 4 * hi;
}

// XOR swizzle for a row-major bf16 LDS tile (stride LD elements):
// byte offset ^ ((row&7)<<4) — bijective within an 8-row stripe, keeps
// 16B alignment so b128 reads/writes stay legal.
template <int LD>
__device__ __forceinline__ char* swz2(ushort_t* base, int row, int col_elem) {
    return reinterpret_cast<char*>(base) +
           (((row * LD + col_elem) * 2) ^ ((row & 7) << 4));
}

__device__ __forceinline__ unsigned pack_bf16(float a, float b) {
    return (unsigned)f32_to_bf16(a) | ((unsigned)f32_to_bf16(b) << 16);
}

// ---------------------------------------------------------------------------
// layout probe for mfma_f32_32x32x16_bf16: one wave computes C = A(32x16) @
// B(16x32); asymmetric-input GPU test validates the A/B/C lane mappings.
//   A: lane l holds A[l&31][(l>>5)*8 + j]
//   B: lane l holds B[(l>>5)*8 + j][l&31]
// ---------------------------------------------------------------------------
__global__ void mfma32_layout_probe(const ushort_t* A, const ushort_t* B, float* C) {
    int l = threadIdx.x;
    int l32 = l & 31, hi = l >> 5;
    frag8 a, b;
#pragma unroll
    for (int j = 0; j < 8; j++) {
        a[j] = (short)A[l32 * 16 + hi * 8 + j];
        b[j] = (short)B[(hi * 8 + j) * 32 + l32];
    }
    f32x16 c;
#pragma unroll
    for (int r = 0; r < 16; r++) c[r] = 0.f;
    c = mfma32(a, b, c);
#pragma unroll
    for (int r = 0; r < 16; r++) C[crow32(r, hi) * 32 + l32] = c[r];
}

// permlane32_swap semantics probe: out0/out1 = the two results of
// swap(x, y) where x = lane, y = 1000+lane.
__global__ void permlane_probe(int* out0, int* out1) {
    int l = threadIdx.x;
    auto r = __builtin_amdgcn_permlane32_swap((unsigned)l, (unsigned)(1000 + l),
                                              false, false);
    out0[l] = (int)r[0];
    out1[l] = (int)r[1];
}

// ---------------------------------------------------------------------------
// forward v2
// ---------------------------------------------------------------------------
template <int D, bool MASKED = false>
__global__ __launch_bounds__(FA2_BLOCK) void flash_fwd2_kernel(
    const ushort_t* __restrict__ q,   // [B, Sq, Hq, D]
    const ushort_t* __restrict__ k,   // [B, Skv, Hk, D]
    const ushort_t* __restrict__ v,   // [B, Skv, Hk, D]
    ushort_t* __restrict__ o,         // [B, Sq, Hq, D]
    float* __restrict__ lse,          // [B, Hq, Sq]
    const int* __restrict__ startend, // FlashMask [B, Skv] (MASKED only)
    int B, int Sq, int Skv, int Hq, int Hk, float scale, int causal) {
    constexpr int DSTEPS = D / 16;    // K-steps of the S^T MFMAs
    constexpr int NDT = D / 32;       // 32-col d-tiles of O
    constexpr int KVT = FA2_BLKN / 32;  // kv sub-tiles (2)

    // double-buffered: compute tile t from buf[t&1] while staging t+1 into
    // buf[(t+1)&1] -> ONE barrier per KV tile instead of two (the barrier
    // drain before s_barrier is the dominant structural stall, guide §5)
    __shared__ ushort_t k_lds[2][FA2_BLKN * D];
    __shared__ ushort_t vt_lds[2][D * FA2_BLKN];
    // per-tile FlashMask column bounds (staged with the K/V tile) and
    // the tile's max bound (wave 0 reduces it during staging) for
    // whole-tile skipping: a wave with qw >= max end sees nothing
    // per-tile FlashMask column bounds (staged with the K/V tile);
    // the tile min/max bounds travel in registers: every wave reduces
    // the 64-int column itself (256 B from L2, 6 shfl_xor) one tile
    // ahead, so whole-tile skips AND whole-block staging skips need no
    // extra barrier
    __shared__ int se_lds[2][MASKED ? FA2_BLKN : 1];

    const int tid = threadIdx.x;
    const int wave = tid >> 6;
    const int lane = tid & 63;
    const int l32 = lane & 31;
    const int hi = lane >> 5;

    // XCD-aware remap: the dispatcher round-robins flat workgroup ids
    // over the 8 XCDs, so with the natural (qt fastest) order every XCD
    // streams every head's KV and no L2 ever re-hits it.  Remap so one
    // (b,h)'s q-tiles run on ONE XCD back-to-back: its 16 workgroups
    // then consume each KV tile out of that XCD's L2 together.
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
    const int q_base = qt * FA2_BLKM;
    const int qw = q_base + wave * FA2_QW;   // this wave's first q row
    const int qg = qw + l32;                 // this lane's q row (softmax owner)
    const int causal_off = Skv - Sq;

    const long long q_row_stride = (long long)Hq * D;
    const long long kv_row_stride = (long long)Hk * D;
    const ushort_t* q_ptr = q + ((long long)b * Sq * Hq + hq) * D;
    const ushort_t* k_ptr = k + ((long long)b * Skv * Hk + hk) * D;
    const ushort_t* v_ptr = v + ((long long)b * Skv * Hk + hk) * D;

    // Q B-fragments in registers: lane l holds Q[qw + l32][kk*16 + hi*8 + j]
    frag8 aq[DSTEPS];
    if (qg < Sq) {
#pragma unroll
        for (int kk = 0; kk < DSTEPS; kk++)
            aq[kk] = *reinterpret_cast<const frag8*>(
                q_ptr + (long long)qg * q_row_stride + kk * 16 + hi * 8);
    } else {
#pragma unroll
        for (int kk = 0; kk < DSTEPS; kk++) aq[kk] = frag8{0};
    }

    float m_run = -INFINITY, l_run = 0.f;
    f32x16 acc_o[NDT];
#pragma unroll
    for (int n = 0; n < NDT; n++)
#pragma unroll
        for (int r = 0; r < 16; r++) acc_o[n][r] = 0.f;

    int n_kv_tiles = (Skv + FA2_BLKN - 1) / FA2_BLKN;
    if (causal) {
        int max_kv = q_base + FA2_BLKM - 1 + causal_off;
        int lim = (max_kv + FA2_BLKN) / FA2_BLKN;
        n_kv_tiles = min(n_kv_tiles, max(lim, 0));
    }

    // register staging: each thread owns 2 consecutive kv rows x 8 d cols
    const int s_rowp = tid >> 4;              // 0..31 row pair
    const int s_col = (tid & 15) * 8;         // d col (D=128: 16 chunks)
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
        if (MASKED && tid < FA2_BLKN) {
            int g = kv_base + tid;
            se_lds[buf][tid] =
                (g < Skv) ? startend[(long long)b * Skv + g] : 0;
        }
        *reinterpret_cast<short8v*>(swz2<D>(k_lds[buf], s_row0, s_col)) = sk0;
        *reinterpret_cast<short8v*>(swz2<D>(k_lds[buf], s_row0 + 1, s_col)) = sk1;
#pragma unroll
        for (int j = 0; j < 8; j++) {
            unsigned p32 = ((unsigned)(unsigned short)sv0[j]) |
                           (((unsigned)(unsigned short)sv1[j]) << 16);
            *reinterpret_cast<unsigned*>(swz2<FA2_BLKN>(vt_lds[buf], s_col + j, s_row0)) = p32;
        }
    };

    // every wave reduces a tile's 64 bounds itself: returns (min, max)
    // uniform across the wave, no shared-memory round trip
    auto tile_bounds = [&](int kv_base, int& mn_out) -> int {
        int g = kv_base + lane;
        int v = (g < Skv) ? startend[(long long)b * Skv + g] : 0;
        int mx = v, mn = (g < Skv) ? v : 0x7fffffff;
#pragma unroll
        for (int off = 32; off; off >>= 1) {
            mx = max(mx, __shfl_xor(mx, off, 64));
            mn = min(mn, __shfl_xor(mn, off, 64));
        }
        mn_out = mn;
        return mx;
    };

    load_tile(0);
    write_tile(0, 0);
    int se_min_cur = 0, se_max_cur = 0x7fffffff;
    if (MASKED) se_max_cur = tile_bounds(0, se_min_cur);
    __syncthreads();

    for (int kvt = 0; kvt < n_kv_tiles; kvt++) {
        const int kv_base = kvt * FA2_BLKN;
        const int cur = kvt & 1;
        // a wave whose q rows all precede this kv tile contributes nothing:
        // skip its compute but keep it in the staging barriers
        bool wave_skip =
            causal && (kv_base > qw + FA2_QW - 1 + causal_off);
        // FlashMask whole-tile skip: every kv bound in this tile is at or
        // below the wave's first q row -> no visible pair
        if (MASKED && qw >= se_max_cur) wave_skip = true;

        f32x16 st[KVT];
        if (!wave_skip) {
#pragma unroll
            for (int nt = 0; nt < KVT; nt++)
#pragma unroll
                for (int r = 0; r < 16; r++) st[nt][r] = 0.f;
            __builtin_amdgcn_s_setprio(1);
#pragma unroll
            for (int kk = 0; kk < DSTEPS; kk++) {
#pragma unroll
                for (int nt = 0; nt < KVT; nt++) {
                    frag8 ak = *reinterpret_cast<const frag8*>(
                        swz2<D>(k_lds[cur], nt * 32 + l32, kk * 16 + hi * 8));
                    st[nt] = mfma32(ak, aq[kk], st[nt]);
                }
            }
            __builtin_amdgcn_s_setprio(0);
        }

        // issue next tile's global loads now (T14): latency hides under
        // softmax + PV.  FlashMask: when the whole BLOCK's first q row
        // clears the next tile's max bound, nothing will read it — skip
        // the 32 KB K/V stage entirely (the bounds cost 256 B from L2)
        int se_min_next = 0, se_max_next = 0x7fffffff;
        bool stage_skip = false;
        if (kvt + 1 < n_kv_tiles) {
            if (MASKED) {
                se_max_next = tile_bounds(kv_base + FA2_BLKN, se_min_next);
                stage_skip = (q_base >= se_max_next);
            }
            if (!stage_skip) load_tile(kv_base + FA2_BLKN);
        }

        if (!wave_skip) {
            // mask + scale.  A tile is "full" when every (q, kv) pair in this
            // wave's sub-block is visible — the common case away from the
            // diagonal; masking math is skipped entirely.
            const bool full_tile =
                (kv_base + FA2_BLKN <= Skv) &&
                (!causal || (kv_base + FA2_BLKN - 1 <= qw + causal_off)) &&
                (!MASKED || (qw + FA2_QW - 1 < se_min_cur));
            if (full_tile) {
#pragma unroll
                for (int nt = 0; nt < KVT; nt++)
#pragma unroll
                    for (int r = 0; r < 16; r++) st[nt][r] *= scale;
            } else {
#pragma unroll
                for (int nt = 0; nt < KVT; nt++)
#pragma unroll
                    for (int r = 0; r < 16; r++) {
                        int row = nt * 32 + crow32(r, hi);
                        int kvg = kv_base + row;
                        bool vis = (kvg < Skv) &&
                                   (!causal || kvg <= qg + causal_off);
                        if (MASKED) vis = vis && (qg < se_lds[cur][row]);
                        st[nt][r] = vis ? st[nt][r] * scale : -INFINITY;
                    }
            }

            // per-lane row max over the 32 kv values + one half-swap combine
            float pm = -INFINITY;
#pragma unroll
            for (int nt = 0; nt < KVT; nt++)
#pragma unroll
                for (int r = 0; r < 16; r++) pm = fmaxf(pm, st[nt][r]);
            {
                union { float f; unsigned u; } x{pm};
                auto rr = __builtin_amdgcn_permlane32_swap(x.u, x.u, false, false);
                union { unsigned u; float f; } a{rr[0]}, c{rr[1]};
                pm = fmaxf(a.f, c.f);
            }

            // defer-max (T13): only rescale O when the max actually grew
            bool grew = pm > m_run + DEFER_THR ||
                        (m_run == -INFINITY && pm > -INFINITY);
            if (__any(grew)) {
                float m_new = fmaxf(m_run, pm);
                float alpha = (m_run == -INFINITY)
                                  ? 0.f
                                  : __expf(m_run - m_new);
                if (m_new == -INFINITY) alpha = 1.f;
                l_run *= alpha;
                m_run = m_new;
                // O rows are C-layout rows (q = crow32): fetch each row's
                // alpha from the lane that owns that q
#pragma unroll
                for (int r = 0; r < 16; r++) {
                    float ar = __shfl(alpha, crow32(r, hi), 64);
#pragma unroll
                    for (int n = 0; n < NDT; n++) acc_o[n][r] *= ar;
                }
            }

            // P = exp(S - m); per-lane sum + half-swap combine
            float ps = 0.f;
#pragma unroll
            for (int nt = 0; nt < KVT; nt++)
#pragma unroll
                for (int r = 0; r < 16; r++) {
                    float s = st[nt][r];
                    float p = (s == -INFINITY) ? 0.f : __expf(s - m_run);
                    st[nt][r] = p;
                    ps += p;
                }
            {
                union { float f; unsigned u; } x{ps};
                auto rr = __builtin_amdgcn_permlane32_swap(x.u, x.u, false, false);
                union { unsigned u; float f; } a{rr[0]}, c{rr[1]};
                ps = a.f + c.f;
            }
            l_run += ps;

            // P (C-layout, rows = kv) -> PV A-fragments, fully in-register
            // (T12).  For k-step ks the lane needs P[q = l32][kv = ks*16 +
            // hi*8 + j]; swap(pack(p0,p1), pack(p4,p5)) delivers words {0,2}
            // and swap(pack(p2,p3), pack(p6,p7)) words {1,3}.
            __builtin_amdgcn_s_setprio(1);
#pragma unroll
            for (int ks = 0; ks < FA2_BLKN / 16; ks++) {
                const int nt = ks >> 1;
                const int rb = (ks & 1) * 8;   // reg base within the sub-tile
                unsigned w[4];
                {
                    unsigned lo0 = pack_bf16(st[nt][rb + 0], st[nt][rb + 1]);
                    unsigned hi0 = pack_bf16(st[nt][rb + 4], st[nt][rb + 5]);
                    auto rr = __builtin_amdgcn_permlane32_swap(lo0, hi0, false, false);
                    w[0] = rr[0]; w[2] = rr[1];
                }
                {
                    unsigned lo1 = pack_bf16(st[nt][rb + 2], st[nt][rb + 3]);
                    unsigned hi1 = pack_bf16(st[nt][rb + 6], st[nt][rb + 7]);
                    auto rr = __builtin_amdgcn_permlane32_swap(lo1, hi1, false, false);
                    w[1] = rr[0]; w[3] = rr[1];
                }
                frag8 pa;
#pragma unroll
                for (int j = 0; j < 4; j++) {
                    pa[j * 2] = (short)(w[j] & 0xffff);
                    pa[j * 2 + 1] = (short)(w[j] >> 16);
                }
#pragma unroll
                for (int n = 0; n < NDT; n++) {
                    frag8 bv = *reinterpret_cast<const frag8*>(
                        swz2<FA2_BLKN>(vt_lds[cur], n * 32 + l32, ks * 16 + hi * 8));
                    acc_o[n] = mfma32(pa, bv, acc_o[n]);
                }
            }
            __builtin_amdgcn_s_setprio(0);
        }

        // stage t+1 into the OTHER buffer while this tile's compute may
        // still be in flight on other waves, then one barrier: it both
        // publishes buf[cur^1] and guarantees every wave has finished
        // reading buf[cur] before iteration t+1 overwrites it
        if (kvt + 1 < n_kv_tiles && !stage_skip)
            write_tile(cur ^ 1, kv_base + FA2_BLKN);
        if (MASKED) { se_max_cur = se_max_next; se_min_cur = se_min_next; }
        __syncthreads();
    }

    // epilogue: O / l, LSE.  inv_l lives in the lane that owns q; O rows
    // fetch it by shfl like the rescale.
    float inv_l = (l_run > 0.f) ? 1.0f / l_run : 0.f;
#pragma unroll
    for (int r = 0; r < 16; r++) {
        int qrow = qw + crow32(r, hi);
        if (qrow >= Sq) continue;
        float il = __shfl(inv_l, crow32(r, hi), 64);
        ushort_t* orow = o + ((long long)b * Sq + qrow) * q_row_stride + (long long)hq * D;
#pragma unroll
        for (int n = 0; n < NDT; n++)
            orow[n * 32 + l32] = f32_to_bf16(acc_o[n][r] * il);
    }
    if (hi == 0 && qg < Sq) {
        float lv = (l_run > 0.f) ? (m_run + __logf(l_run)) : -INFINITY;
        lse[((long long)b * Hq + hq) * Sq + qg] = lv;
    }
}

// ---------------------------------------------------------------------------
// launchers
// ---------------------------------------------------------------------------
void launch_mfma32_probe(const void* A, const void* B, float* C, hipStream_t stream) {
    hipLaunchKernelGGL(mfma32_layout_probe, dim3(1), dim3(64), 0, stream,
                       (const ushort_t*)A, (const ushort_t*)B, C);
}

void launch_permlane_probe(int* out0, int* out1, hipStream_t stream) {
    hipLaunchKernelGGL(permlane_probe, dim3(1), dim3(64), 0, stream, out0, out1);
}

bool launch_flash_fwd2(const void* q, const void* k, const void* v, void* o,
                       float* lse, int B, int Sq, int Skv, int Hq, int Hk,
                       int D, float scale, bool causal, hipStream_t stream) {
    if (D != 128 || (Hq % Hk) != 0) return false;
    dim3 grid((Sq + FA2_BLKM - 1) / FA2_BLKM, B * Hq);
    hipLaunchKernelGGL((flash_fwd2_kernel<128, false>), grid, dim3(FA2_BLOCK), 0, stream,
                       (const ushort_t*)q, (const ushort_t*)k, (const ushort_t*)v,
                       (ushort_t*)o, lse, (const int*)nullptr,
                       B, Sq, Skv, Hq, Hk, scale, causal ? 1 : 0);
    return true;
}

bool launch_flash_fwd2_mask(const void* q, const void* k, const void* v,
                            void* o, float* lse, const int* startend,
                            int B, int Sq, int Skv, int Hq, int Hk,
                            int D, float scale, hipStream_t stream) {
    if (D != 128 || (Hq % Hk) != 0) return false;
    dim3 grid((Sq + FA2_BLKM - 1) / FA2_BLKM, B * Hq);
    hipLaunchKernelGGL((flash_fwd2_kernel<128, true>), grid, dim3(FA2_BLOCK), 0,
                       stream, (const ushort_t*)q, (const ushort_t*)k,
                       (const ushort_t*)v, (ushort_t*)o, lse, startend,
                       B, Sq, Skv, Hq, Hk, scale, 1);
    return true;
}
