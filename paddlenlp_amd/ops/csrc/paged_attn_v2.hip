// Paged-KV decode attention v2 — MFMA-tiled, gfx950.
//
// Replaces the v1 scalar-dot-product decode kernel (paged_attn.hip) on the
// D=128, G<=16 path.  v1 processed one token per wave-iteration with a
// serial wave_reduce -> exp -> rescale chain per token and 4-byte K/V
// loads; measured 0.9 TB/s of KV traffic at B64/ctx4096 (19 ms of a 26 ms
// decode step).  v2 reuses the flash_fwd2 structure (flash_attn_v2.hip):
//
//   * one workgroup = 4 waves per (batch, kv-head, split); the block
//     cooperatively stages a 128-token K/V tile into LDS with coalesced
//     16-byte loads, XOR-swizzled rows, V transposed via u32 row-pair
//     packing;
//   * each wave runs a 32-token quarter: S^T = K·Q^T on
//     mfma_f32_32x32x16_bf16 (C-layout col = q-head, so the online
//     softmax is per-lane with one permlane32_swap combine), then
//     P -> PV A-fragments fully in-register (T12) and O += P·V;
//   * the G (<=16) query heads of the GQA group ride along as the 32-col
//     MFMA N dimension — decode is bandwidth-bound, the idle columns are
//     free;
//   * the 4 waves' (m, l, acc) merge through LDS (aliased over the K
//     staging buffer), writing either the output or the same fp32
//     partials the v1 split-merge kernel consumes.
//
// Reference behavior: csrc/gpu/append_attn decode path.
#include "common.h"

#define PD2_WAVES 4
#define PD2_BLOCK (PD2_WAVES * 64)
#define PD2_TILE 256            // tokens staged per block iteration
#define PD2_MAXG 16

typedef short8v frag8;

static __device__ __forceinline__ f32x16 mfma32d(frag8 a, frag8 b, f32x16 c) {
    return __builtin_amdgcn_mfma_f32_32x32x16_bf16(a, b, c, 0, 0, 0);
}

static __device__ __forceinline__ int crow32d(int r, int hi) {
    return (r & 3) + 8 * (r >> 2) + 4 * hi;
}

template <int LD>
static __device__ __forceinline__ char* swzd(ushort_t* base, int row, int col_elem) {
    return reinterpret_cast<char*>(base) +
           (((row * LD + col_elem) * 2) ^ ((row & 7) << 4));
}

// MODE: 0 = bf16 cache, 1 = int8, 2 = int4 packed nibbles (q+8 offset)
template <int D, int MODE>
__global__ __launch_bounds__(PD2_BLOCK) void paged_decode_attn2_kernel(
    const ushort_t* __restrict__ q,        // [B, Hq, D]
    const void* __restrict__ k_cache,      // [nblocks, bs, Hk, D] bf16|int8
    const void* __restrict__ v_cache,
    const float* __restrict__ k_scale,     // [nblocks*bs*Hk] (C8)
    const float* __restrict__ v_scale,
    const int* __restrict__ block_table,   // [B, max_blocks]
    const int* __restrict__ seq_lens,      // [B]
    ushort_t* __restrict__ out,            // [B, Hq, D]
    float* __restrict__ partials,          // [B, Hk, nsplit, G*(2+D)]
    int B, int Hq, int Hk, int block_size, int max_blocks, float scale,
    int nsplit) {
    constexpr int DSTEPS = D / 16;
    constexpr int NDT = D / 32;

    // K is read straight from the cache as per-lane A-fragments (the rows
    // stream through L1; no reuse to justify LDS), so LDS holds only the
    // transposed V tile (32 KB) -> 4 blocks/CU, twice the in-flight HBM
    // bytes of the k_lds variant (measured: 4.1k -> see profiles).  The
    // wave-merge scratch (<=33.3 KB for G=16) aliases it after the loop.
    constexpr int SMEM_BYTES = (D * PD2_TILE * 2) > (PD2_WAVES * PD2_MAXG * (2 + D) * 4)
                                   ? (D * PD2_TILE * 2)
                                   : (PD2_WAVES * PD2_MAXG * (2 + D) * 4);
    __shared__ char smem[SMEM_BYTES];
    ushort_t* vt_lds = reinterpret_cast<ushort_t*>(smem);
    float* mrg = reinterpret_cast<float*>(smem);   // [4][G][2+D] after loop

    const int b = blockIdx.x;
    const int hk = blockIdx.y;
    const int split = blockIdx.z;
    const int G = Hq / Hk;
    const int tid = threadIdx.x;
    const int wave = tid >> 6;
    const int lane = tid & 63;
    const int l32 = lane & 31;
    const int hi = lane >> 5;
    const int seq_len = seq_lens[b];
    if (seq_len <= 0) return;
    const int hq0 = hk * G;

    const int chunk = (seq_len + nsplit - 1) / nsplit;
    const int c0 = split * chunk;
    const int c1 = min(seq_len, c0 + chunk);

    const ushort_t* k16 = (const ushort_t*)k_cache;
    const ushort_t* v16 = (const ushort_t*)v_cache;
    const signed char* k8 = (const signed char*)k_cache;
    const signed char* v8 = (const signed char*)v_cache;
    const int* bt = block_table + (long long)b * max_blocks;

    // Q^T B-fragments: lane holds Q[hq0 + l32][kk*16 + hi*8 + j] (zero for
    // lanes beyond the group)
    frag8 qT[DSTEPS];
    if (l32 < G) {
        const ushort_t* qp = q + ((long long)b * Hq + hq0 + l32) * D;
#pragma unroll
        for (int kk = 0; kk < DSTEPS; kk++)
            qT[kk] = *reinterpret_cast<const frag8*>(qp + kk * 16 + hi * 8);
    } else {
#pragma unroll
        for (int kk = 0; kk < DSTEPS; kk++) qT[kk] = frag8{0};
    }

    float m_run = -INFINITY, l_run = 0.f;
    f32x16 acc_o[NDT];
#pragma unroll
    for (int n = 0; n < NDT; n++)
#pragma unroll
        for (int r = 0; r < 16; r++) acc_o[n][r] = 0.f;

    // staging: thread owns token row-pair (tid&63)*2, cols wave*32..+31
    const int s_r0 = (tid & 63) * 4;
    const int s_c = wave * 32;
    short8v vr[4][4];

    auto load_tile = [&](int t0) {
#pragma unroll
        for (int rr = 0; rr < 4; rr++) {
            int tok = t0 + s_r0 + rr;
            bool valid = tok < c1;
            if (valid) {
                int blk = bt[tok / block_size];
                long long rec = ((long long)blk * block_size + (tok % block_size)) * Hk + hk;
                long long base = rec * D + s_c;
                if (MODE == 1) {
                    float vs = v_scale[rec];
#pragma unroll
                    for (int cc = 0; cc < 4; cc++)
#pragma unroll
                        for (int j = 0; j < 8; j++)
                            vr[rr][cc][j] = (short)f32_to_bf16((float)v8[base + cc * 8 + j] * vs);
                } else if (MODE == 2) {
                    float vs = v_scale[rec];
                    const unsigned char* v4 =
                        (const unsigned char*)v_cache + (rec * D + s_c) / 2;
#pragma unroll
                    for (int cc = 0; cc < 4; cc++)
#pragma unroll
                        for (int j = 0; j < 4; j++) {
                            unsigned byte = v4[cc * 4 + j];
                            vr[rr][cc][j * 2] = (short)f32_to_bf16(
                                ((int)(byte >> 4) - 8) * vs);
                            vr[rr][cc][j * 2 + 1] = (short)f32_to_bf16(
                                ((int)(byte & 0xF) - 8) * vs);
                        }
                } else {
#pragma unroll
                    for (int cc = 0; cc < 4; cc++)
                        vr[rr][cc] = *reinterpret_cast<const short8v*>(v16 + base + cc * 8);
                }
            } else {
#pragma unroll
                for (int cc = 0; cc < 4; cc++)
                    vr[rr][cc] = short8v{0,0,0,0,0,0,0,0};
            }
        }
    };
    auto write_tile = [&]() {
#pragma unroll
        for (int pp = 0; pp < 2; pp++)
#pragma unroll
            for (int cc = 0; cc < 4; cc++) {
#pragma unroll
                for (int j = 0; j < 8; j++) {
                    unsigned p32 = ((unsigned)(unsigned short)vr[pp * 2][cc][j]) |
                                   (((unsigned)(unsigned short)vr[pp * 2 + 1][cc][j]) << 16);
                    *reinterpret_cast<unsigned*>(
                        swzd<PD2_TILE>(vt_lds, s_c + cc * 8 + j, s_r0 + pp * 2)) = p32;
                }
            }
    };

    if (c0 < c1) {
        load_tile(c0);
        write_tile();
        __syncthreads();
    }
    for (int t0 = c0; t0 < c1; t0 += PD2_TILE) {
        // issue the NEXT tile's global loads before this tile's compute
        // (T14 async-stage split): HBM latency hides under MFMA + softmax
        if (t0 + PD2_TILE < c1) load_tile(t0 + PD2_TILE);

        // ---- per-wave 64-token quarter (two 32-token sub-iterations) ----
#pragma unroll 1
        for (int sub = 0; sub < 2; sub++) {
        const int tq0 = t0 + (wave * 2 + sub) * 32;
        if (tq0 < c1) {
            f32x16 st;
#pragma unroll
            for (int r = 0; r < 16; r++) st[r] = 0.f;
            // per-lane K row pointer for this quarter (clamped; masked later)
            {
                int tok = tq0 + l32;
                int tokc = min(tok, c1 - 1);
                int blk = bt[tokc / block_size];
                long long rec = ((long long)blk * block_size + (tokc % block_size)) * Hk + hk;
                const long long kbase = rec * D;
                if (MODE == 1) {
                    float ks = k_scale[rec];
                    frag8 ak[DSTEPS];
#pragma unroll
                    for (int kk = 0; kk < DSTEPS; kk++)
#pragma unroll
                        for (int j = 0; j < 8; j++)
                            ak[kk][j] = (short)f32_to_bf16(
                                (float)k8[kbase + kk * 16 + hi * 8 + j] * ks);
                    __builtin_amdgcn_s_setprio(1);
#pragma unroll
                    for (int kk = 0; kk < DSTEPS; kk++)
                        st = mfma32d(ak[kk], qT[kk], st);
                    __builtin_amdgcn_s_setprio(0);
                } else if (MODE == 2) {
                    float ks = k_scale[rec];
                    const unsigned char* k4 = (const unsigned char*)k_cache + kbase / 2;
                    frag8 ak[DSTEPS];
#pragma unroll
                    for (int kk = 0; kk < DSTEPS; kk++)
#pragma unroll
                        for (int j = 0; j < 4; j++) {
                            unsigned byte = k4[(kk * 16 + hi * 8) / 2 + j];
                            ak[kk][j * 2] = (short)f32_to_bf16(
                                ((int)(byte >> 4) - 8) * ks);
                            ak[kk][j * 2 + 1] = (short)f32_to_bf16(
                                ((int)(byte & 0xF) - 8) * ks);
                        }
                    __builtin_amdgcn_s_setprio(1);
#pragma unroll
                    for (int kk = 0; kk < DSTEPS; kk++)
                        st = mfma32d(ak[kk], qT[kk], st);
                    __builtin_amdgcn_s_setprio(0);
                } else {
                    frag8 ak[DSTEPS];
#pragma unroll
                    for (int kk = 0; kk < DSTEPS; kk++)
                        ak[kk] = *reinterpret_cast<const frag8*>(
                            k16 + kbase + kk * 16 + hi * 8);
                    __builtin_amdgcn_s_setprio(1);
#pragma unroll
                    for (int kk = 0; kk < DSTEPS; kk++)
                        st = mfma32d(ak[kk], qT[kk], st);
                    __builtin_amdgcn_s_setprio(0);
                }
            }

            const bool full = (tq0 + 32 <= c1);
            if (full) {
#pragma unroll
                for (int r = 0; r < 16; r++) st[r] *= scale;
            } else {
#pragma unroll
                for (int r = 0; r < 16; r++) {
                    int tok = tq0 + crow32d(r, hi);
                    st[r] = (tok < c1) ? st[r] * scale : -INFINITY;
                }
            }

            float pm = -INFINITY;
#pragma unroll
            for (int r = 0; r < 16; r++) pm = fmaxf(pm, st[r]);
            {
                union { float f; unsigned u; } x{pm};
                auto rr = __builtin_amdgcn_permlane32_swap(x.u, x.u, false, false);
                union { unsigned u; float f; } a{rr[0]}, c{rr[1]};
                pm = fmaxf(a.f, c.f);
            }
            bool grew = pm > m_run + 8.0f ||
                        (m_run == -INFINITY && pm > -INFINITY);
            if (__any(grew)) {
                float m_new = fmaxf(m_run, pm);
                float alpha = (m_run == -INFINITY) ? 0.f : __expf(m_run - m_new);
                if (m_new == -INFINITY) alpha = 1.f;
                l_run *= alpha;
                m_run = m_new;
#pragma unroll
                for (int r = 0; r < 16; r++) {
                    float ar = __shfl(alpha, crow32d(r, hi), 64);
#pragma unroll
                    for (int n = 0; n < NDT; n++) acc_o[n][r] *= ar;
                }
            }
            float ps = 0.f;
#pragma unroll
            for (int r = 0; r < 16; r++) {
                float s = st[r];
                float p = (s == -INFINITY) ? 0.f : __expf(s - m_run);
                st[r] = p;
                ps += p;
            }
            {
                union { float f; unsigned u; } x{ps};
                auto rr = __builtin_amdgcn_permlane32_swap(x.u, x.u, false, false);
                union { unsigned u; float f; } a{rr[0]}, c{rr[1]};
                ps = a.f + c.f;
            }
            l_run += ps;

            // P -> A-fragments (T12) and O += P·V
            __builtin_amdgcn_s_setprio(1);
#pragma unroll
            for (int ks = 0; ks < 2; ks++) {
                const int rb = ks * 8;
                unsigned w[4];
                {
                    unsigned lo0 = (unsigned)f32_to_bf16(st[rb + 0]) |
                                   ((unsigned)f32_to_bf16(st[rb + 1]) << 16);
                    unsigned hi0 = (unsigned)f32_to_bf16(st[rb + 4]) |
                                   ((unsigned)f32_to_bf16(st[rb + 5]) << 16);
                    auto rr = __builtin_amdgcn_permlane32_swap(lo0, hi0, false, false);
                    w[0] = rr[0]; w[2] = rr[1];
                }
                {
                    unsigned lo1 = (unsigned)f32_to_bf16(st[rb + 2]) |
                                   ((unsigned)f32_to_bf16(st[rb + 3]) << 16);
                    unsigned hi1 = (unsigned)f32_to_bf16(st[rb + 6]) |
                                   ((unsigned)f32_to_bf16(st[rb + 7]) << 16);
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
                        swzd<PD2_TILE>(vt_lds, n * 32 + l32,
                                       (wave * 2 + sub) * 32 + ks * 16 + hi * 8));
                    acc_o[n] = mfma32d(pa, bv, acc_o[n]);
                }
            }
            __builtin_amdgcn_s_setprio(0);
        }
        }
        __syncthreads();
        if (t0 + PD2_TILE < c1) {
            write_tile();
            __syncthreads();
        }
    }

    // ---- merge the 4 waves through LDS (aliases k_lds) ----
    // layout: mrg[wave][g][0]=m, [1]=l, [2+d]=acc
    const int rec = 2 + D;
    if (hi == 0 && l32 < G) {
        mrg[(wave * G + l32) * rec + 0] = m_run;
        mrg[(wave * G + l32) * rec + 1] = l_run;
    }
#pragma unroll
    for (int r = 0; r < 16; r++) {
        int g = crow32d(r, hi);
        if (g < G) {
#pragma unroll
            for (int n = 0; n < NDT; n++)
                mrg[(wave * G + g) * rec + 2 + n * 32 + l32] = acc_o[n][r];
        }
    }
    __syncthreads();
    for (int idx = tid; idx < G * D; idx += PD2_BLOCK) {
        int g = idx / D, d = idx % D;
        float gm = -INFINITY;
#pragma unroll
        for (int w = 0; w < PD2_WAVES; w++)
            gm = fmaxf(gm, mrg[(w * G + g) * rec]);
        float gl = 0.f, oa = 0.f;
#pragma unroll
        for (int w = 0; w < PD2_WAVES; w++) {
            float mw = mrg[(w * G + g) * rec];
            float a = (mw == -INFINITY) ? 0.f : __expf(mw - gm);
            gl += mrg[(w * G + g) * rec + 1] * a;
            oa += mrg[(w * G + g) * rec + 2 + d] * a;
        }
        if (nsplit == 1) {
            float inv = (gl > 0.f) ? 1.0f / gl : 0.f;
            out[((long long)b * Hq + hq0 + g) * D + d] = f32_to_bf16(oa * inv);
        } else {
            float* pp = partials +
                (((long long)b * Hk + hk) * nsplit + split) * (long long)G * rec +
                (long long)g * rec;
            if (d == 0) { pp[0] = gm; pp[1] = gl; }
            pp[2 + d] = oa;
        }
    }
}

// ---------------------------------------------------------------------------
bool launch_paged_decode_attn2(const void* q, const void* k_cache, const void* v_cache,
                               const float* k_scale, const float* v_scale,
                               const int* block_table, const int* seq_lens, void* out,
                               float* partials, int nsplit,
                               int B, int Hq, int Hk, int D, int block_size,
                               int max_blocks, float scale, int cache_mode,
                               hipStream_t stream) {
    if (D != 128 || (Hq % Hk) != 0 || Hq / Hk > PD2_MAXG) return false;
    dim3 grid(B, Hk, nsplit);
#define PD2_LAUNCH(MM)                                                             \
    hipLaunchKernelGGL((paged_decode_attn2_kernel<128, MM>), grid, dim3(PD2_BLOCK),\
                       0, stream, (const ushort_t*)q, k_cache, v_cache, k_scale,   \
                       v_scale, block_table, seq_lens, (ushort_t*)out, partials,   \
                       B, Hq, Hk, block_size, max_blocks, scale, nsplit)
    if (cache_mode == 2) PD2_LAUNCH(2);
    else if (cache_mode == 1) PD2_LAUNCH(1);
    else PD2_LAUNCH(0);
#undef PD2_LAUNCH
    return true;
}

// ---------------------------------------------------------------------------
// decode v3: 16x16x32 fragments for occupancy.  The v2 kernel's 32x32
// tiles hold ~254 VGPRs -> 2 waves/SIMD; decode is HBM-latency-bound, so
// trading MFMA width for wave count wins.  16x16 fragments cut the
// accumulator to 32 VGPRs (8 d-tiles x f32x4) and the whole kernel to
// ~130, and the LDS footprint to one 32 KB V^T tile + 4 KB P staging ->
// 3-4 blocks/CU.  Fragment layouts are the HW-verified 16x16x32 maps
// (flash_attn.hip header): A[l%16][(l/16)*8+j], B[(l/16)*8+j][l%16],
// C[(l/16)*4+r][l%16].
// ---------------------------------------------------------------------------
#define PD3_WAVES 4
#define PD3_BLOCK (PD3_WAVES * 64)
#define PD3_TILE 128
#define PD3_MAXG 16

typedef __attribute__((ext_vector_type(4))) float f32x4v;

static __device__ __forceinline__ f32x4v mfma16d(frag8 a, frag8 b, f32x4v c) {
    return __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, c, 0, 0, 0);
}

template <int D, int MODE>
__global__ __launch_bounds__(PD3_BLOCK) void paged_decode_attn3_kernel(
    const ushort_t* __restrict__ q,
    const void* __restrict__ k_cache,
    const void* __restrict__ v_cache,
    const float* __restrict__ k_scale,
    const float* __restrict__ v_scale,
    const int* __restrict__ block_table,
    const int* __restrict__ seq_lens,
    ushort_t* __restrict__ out,
    float* __restrict__ partials,
    int B, int Hq, int Hk, int block_size, int max_blocks, float scale,
    int nsplit) {
    constexpr int KD = D / 32;     // S^T k-steps (4 at D=128)
    constexpr int ND = D / 16;     // O d-tiles (8)
    // vt (32 KB) + per-wave P staging (4 x 1 KB); merge scratch aliases
    constexpr int VT_ELEMS = D * PD3_TILE;
    constexpr int P_ELEMS = PD3_WAVES * 16 * 32;
    constexpr int MERGE_FLOATS = PD3_WAVES * PD3_MAXG * (2 + D);
    constexpr int SMEM = ((VT_ELEMS + P_ELEMS) * 2 > MERGE_FLOATS * 4)
                             ? (VT_ELEMS + P_ELEMS) * 2 : MERGE_FLOATS * 4;
    __shared__ char smem[SMEM];
    ushort_t* vt_lds = reinterpret_cast<ushort_t*>(smem);
    ushort_t* p_lds = vt_lds + VT_ELEMS;   // [wave][16 q][32 kv]
    float* mrg = reinterpret_cast<float*>(smem);

    const int b = blockIdx.x;
    const int hk = blockIdx.y;
    const int split = blockIdx.z;
    const int G = Hq / Hk;
    const int tid = threadIdx.x;
    const int wave = tid >> 6;
    const int lane = tid & 63;
    const int l16 = lane & 15;
    const int lg = lane >> 4;          // lane group 0..3
    const int seq_len = seq_lens[b];
    if (seq_len <= 0) return;
    const int hq0 = hk * G;

    const int chunk = (seq_len + nsplit - 1) / nsplit;
    const int c0 = split * chunk;
    const int c1 = min(seq_len, c0 + chunk);

    const ushort_t* k16 = (const ushort_t*)k_cache;
    const ushort_t* v16 = (const ushort_t*)v_cache;
    const signed char* k8 = (const signed char*)k_cache;
    const signed char* v8 = (const signed char*)v_cache;
    const unsigned char* k4 = (const unsigned char*)k_cache;
    const int* bt = block_table + (long long)b * max_blocks;

    // Q^T B-fragments: lane holds Q[hq0 + l16][kk*32 + lg*8 + j]
    frag8 qT[KD];
    if (l16 < G) {
        const ushort_t* qp = q + ((long long)b * Hq + hq0 + l16) * D;
#pragma unroll
        for (int kk = 0; kk < KD; kk++)
            qT[kk] = *reinterpret_cast<const frag8*>(qp + kk * 32 + lg * 8);
    } else {
#pragma unroll
        for (int kk = 0; kk < KD; kk++) qT[kk] = frag8{0};
    }

    float m_run = -INFINITY, l_run = 0.f;
    f32x4v acc_o[ND];
#pragma unroll
    for (int n = 0; n < ND; n++)
#pragma unroll
        for (int r = 0; r < 4; r++) acc_o[n][r] = 0.f;

    // staging: thread owns V token row-pair (tid&63)*2, cols wave*32..+31
    const int s_r0 = (tid & 63) * 2;
    const int s_c = wave * 32;
    short8v vr[2][4];

    auto load_tile = [&](int t0) {
#pragma unroll
        for (int rr = 0; rr < 2; rr++) {
            int tok = t0 + s_r0 + rr;
            bool valid = tok < c1;
            if (valid) {
                int blk = bt[tok / block_size];
                long long rec = ((long long)blk * block_size + (tok % block_size)) * Hk + hk;
                long long base = rec * D + s_c;
                if (MODE == 1) {
                    float vs = v_scale[rec];
#pragma unroll
                    for (int cc = 0; cc < 4; cc++)
#pragma unroll
                        for (int j = 0; j < 8; j++)
                            vr[rr][cc][j] = (short)f32_to_bf16((float)v8[base + cc * 8 + j] * vs);
                } else if (MODE == 2) {
                    float vs = v_scale[rec];
                    const unsigned char* v4 = (const unsigned char*)v_cache + (rec * D + s_c) / 2;
#pragma unroll
                    for (int cc = 0; cc < 4; cc++)
#pragma unroll
                        for (int j = 0; j < 4; j++) {
                            unsigned byte = v4[cc * 4 + j];
                            vr[rr][cc][j * 2] = (short)f32_to_bf16(((int)(byte >> 4) - 8) * vs);
                            vr[rr][cc][j * 2 + 1] = (short)f32_to_bf16(((int)(byte & 0xF) - 8) * vs);
                        }
                } else {
#pragma unroll
                    for (int cc = 0; cc < 4; cc++)
                        vr[rr][cc] = *reinterpret_cast<const short8v*>(v16 + base + cc * 8);
                }
            } else {
#pragma unroll
                for (int cc = 0; cc < 4; cc++)
                    vr[rr][cc] = short8v{0,0,0,0,0,0,0,0};
            }
        }
    };
    auto write_tile = [&]() {
#pragma unroll
        for (int cc = 0; cc < 4; cc++) {
#pragma unroll
            for (int j = 0; j < 8; j++) {
                unsigned p32 = ((unsigned)(unsigned short)vr[0][cc][j]) |
                               (((unsigned)(unsigned short)vr[1][cc][j]) << 16);
                *reinterpret_cast<unsigned*>(
                    swzd<PD3_TILE>(vt_lds, s_c + cc * 8 + j, s_r0)) = p32;
            }
        }
    };

    if (c0 < c1) {
        load_tile(c0);
        write_tile();
        __syncthreads();
    }
    for (int t0 = c0; t0 < c1; t0 += PD3_TILE) {
        if (t0 + PD3_TILE < c1) load_tile(t0 + PD3_TILE);

        const int tq0 = t0 + wave * 32;
        if (tq0 < c1) {
            // ---- S^T over two 16-token tiles (per-lane K-row fragments) ----
            f32x4v st[2];
#pragma unroll
            for (int s = 0; s < 2; s++)
#pragma unroll
                for (int r = 0; r < 4; r++) st[s][r] = 0.f;
#pragma unroll
            for (int s = 0; s < 2; s++) {
                int tok = tq0 + s * 16 + l16;
                int tokc = min(tok, c1 - 1);
                int blk = bt[tokc / block_size];
                long long rec = ((long long)blk * block_size + (tokc % block_size)) * Hk + hk;
                frag8 ak[KD];
                if (MODE == 1) {
                    float ks = k_scale[rec];
                    const signed char* kb = k8 + rec * D;
#pragma unroll
                    for (int kk = 0; kk < KD; kk++)
#pragma unroll
                        for (int j = 0; j < 8; j++)
                            ak[kk][j] = (short)f32_to_bf16(
                                (float)kb[kk * 32 + lg * 8 + j] * ks);
                } else if (MODE == 2) {
                    float ks = k_scale[rec];
                    const unsigned char* kb = k4 + rec * D / 2;
#pragma unroll
                    for (int kk = 0; kk < KD; kk++)
#pragma unroll
                        for (int j = 0; j < 4; j++) {
                            unsigned byte = kb[(kk * 32 + lg * 8) / 2 + j];
                            ak[kk][j * 2] = (short)f32_to_bf16(((int)(byte >> 4) - 8) * ks);
                            ak[kk][j * 2 + 1] = (short)f32_to_bf16(((int)(byte & 0xF) - 8) * ks);
                        }
                } else {
                    const ushort_t* kb = k16 + rec * D;
#pragma unroll
                    for (int kk = 0; kk < KD; kk++)
                        ak[kk] = *reinterpret_cast<const frag8*>(kb + kk * 32 + lg * 8);
                }
                __builtin_amdgcn_s_setprio(1);
#pragma unroll
                for (int kk = 0; kk < KD; kk++)
                    st[s] = mfma16d(ak[kk], qT[kk], st[s]);
                __builtin_amdgcn_s_setprio(0);
            }

            // mask + online softmax (C row = kv = s*16 + lg*4 + r, col = q)
            const bool full = (tq0 + 32 <= c1);
#pragma unroll
            for (int s = 0; s < 2; s++)
#pragma unroll
                for (int r = 0; r < 4; r++) {
                    int tok = tq0 + s * 16 + lg * 4 + r;
                    st[s][r] = (full || tok < c1) ? st[s][r] * scale : -INFINITY;
                }
            float pm = -INFINITY;
#pragma unroll
            for (int s = 0; s < 2; s++)
#pragma unroll
                for (int r = 0; r < 4; r++) pm = fmaxf(pm, st[s][r]);
            pm = fmaxf(pm, __shfl_xor(pm, 16, 64));
            pm = fmaxf(pm, __shfl_xor(pm, 32, 64));
            bool grew = pm > m_run + 8.0f ||
                        (m_run == -INFINITY && pm > -INFINITY);
            if (__any(grew)) {
                float m_new = fmaxf(m_run, pm);
                float alpha = (m_run == -INFINITY) ? 0.f : __expf(m_run - m_new);
                if (m_new == -INFINITY) alpha = 1.f;
                l_run *= alpha;
                m_run = m_new;
                // O rows are q = lg*4 + r: fetch alpha from the owner lane
#pragma unroll
                for (int r = 0; r < 4; r++) {
                    float ar = __shfl(alpha, lg * 4 + r, 64);
#pragma unroll
                    for (int n = 0; n < ND; n++) acc_o[n][r] *= ar;
                }
            }
            float ps = 0.f;
#pragma unroll
            for (int s = 0; s < 2; s++)
#pragma unroll
                for (int r = 0; r < 4; r++) {
                    float sv = st[s][r];
                    float p = (sv == -INFINITY) ? 0.f : __expf(sv - m_run);
                    st[s][r] = p;
                    ps += p;
                }
            ps += __shfl_xor(ps, 16, 64);
            ps += __shfl_xor(ps, 32, 64);
            l_run += ps;

            // P -> per-wave LDS ([16 q][32 kv], u32-packed pairs), then PV
            ushort_t* pw = p_lds + wave * 16 * 32;
#pragma unroll
            for (int s = 0; s < 2; s++)
#pragma unroll
                for (int r = 0; r < 4; r += 2) {
                    unsigned pk = (unsigned)f32_to_bf16(st[s][r]) |
                                  ((unsigned)f32_to_bf16(st[s][r + 1]) << 16);
                    // row = q = l16, col = kv = s*16 + lg*4 + r
                    *reinterpret_cast<unsigned*>(
                        swzd<32>(pw, l16, s * 16 + lg * 4 + r)) = pk;
                }
            asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
            frag8 pa = *reinterpret_cast<const frag8*>(swzd<32>(pw, l16, lg * 8));
            __builtin_amdgcn_s_setprio(1);
#pragma unroll
            for (int n = 0; n < ND; n++) {
                frag8 bv = *reinterpret_cast<const frag8*>(
                    swzd<PD3_TILE>(vt_lds, n * 16 + l16,
                                   wave * 32 + lg * 8));
                acc_o[n] = mfma16d(pa, bv, acc_o[n]);
            }
            __builtin_amdgcn_s_setprio(0);
        }
        __syncthreads();
        if (t0 + PD3_TILE < c1) {
            write_tile();
            __syncthreads();
        }
    }

    // ---- merge the 4 waves (same partial format as v1/v2) ----
    const int rec = 2 + D;
    if (lg == 0 && l16 < G) {
        mrg[(wave * G + l16) * rec + 0] = m_run;
        mrg[(wave * G + l16) * rec + 1] = l_run;
    }
#pragma unroll
    for (int r = 0; r < 4; r++) {
        int g = lg * 4 + r;    // O row = q
        if (g < G) {
#pragma unroll
            for (int n = 0; n < ND; n++)
                mrg[(wave * G + g) * rec + 2 + n * 16 + l16] = acc_o[n][r];
        }
    }
    __syncthreads();
    for (int idx = tid; idx < G * D; idx += PD3_BLOCK) {
        int g = idx / D, d = idx % D;
        float gm = -INFINITY;
#pragma unroll
        for (int w = 0; w < PD3_WAVES; w++)
            gm = fmaxf(gm, mrg[(w * G + g) * rec]);
        float gl = 0.f, oa = 0.f;
#pragma unroll
        for (int w = 0; w < PD3_WAVES; w++) {
            float mw = mrg[(w * G + g) * rec];
            float a = (mw == -INFINITY) ? 0.f : __expf(mw - gm);
            gl += mrg[(w * G + g) * rec + 1] * a;
            oa += mrg[(w * G + g) * rec + 2 + d] * a;
        }
        if (nsplit == 1) {
            float inv = (gl > 0.f) ? 1.0f / gl : 0.f;
            out[((long long)b * Hq + hq0 + g) * D + d] = f32_to_bf16(oa * inv);
        } else {
            float* pp = partials +
                (((long long)b * Hk + hk) * nsplit + split) * (long long)G * rec +
                (long long)g * rec;
            if (d == 0) { pp[0] = gm; pp[1] = gl; }
            pp[2 + d] = oa;
        }
    }
}

bool launch_paged_decode_attn3(const void* q, const void* k_cache, const void* v_cache,
                               const float* k_scale, const float* v_scale,
                               const int* block_table, const int* seq_lens, void* out,
                               float* partials, int nsplit,
                               int B, int Hq, int Hk, int D, int block_size,
                               int max_blocks, float scale, int cache_mode,
                               hipStream_t stream) {
    if (D != 128 || (Hq % Hk) != 0 || Hq / Hk > PD3_MAXG) return false;
    dim3 grid(B, Hk, nsplit);
#define PD3_LAUNCH(MM)                                                             \
    hipLaunchKernelGGL((paged_decode_attn3_kernel<128, MM>), grid, dim3(PD3_BLOCK),\
                       0, stream, (const ushort_t*)q, k_cache, v_cache, k_scale,   \
                       v_scale, block_table, seq_lens, (ushort_t*)out, partials,   \
                       B, Hq, Hk, block_size, max_blocks, scale, nsplit)
    if (cache_mode == 2) PD3_LAUNCH(2);
    else if (cache_mode == 1) PD3_LAUNCH(1);
    else PD3_LAUNCH(0);
#undef PD3_LAUNCH
    return true;
}
