// Paged-KV decode attention + fused RoPE cache writers — gfx950.
//
// MI355X-native replacement for the reference's append_attn family
// (csrc/gpu/append_attention.cu + encoder/decoder_write_cache_with_rope,
// SURVEY §2.9) re-designed for CDNA4 instead of ported: decode attention is
// memory-bound, so the kernel is organized around GQA kv-reuse — one
// workgroup per (batch, kv-head) computes ALL q-heads of the group while
// streaming each K/V block from HBM exactly once, with vectorized short8
// loads and fp32 online-softmax state in registers.
//
// Cache layout: [num_blocks, block_size, Hk, D] bf16 per layer.
#include "common.h"
#include <cstdlib>

#define PA_WAVES 4
#define PA_BLOCK (PA_WAVES * 64)
#define MAX_GQA 16  // max q-heads per kv head

// ---------------------------------------------------------------------------
// decode attention: q [B, Hq, D] (current token, post-RoPE),
// block_table [B, max_blocks] int32, seq_lens [B] int32 (kv length incl.
// the current token), out [B, Hq, D].
// ---------------------------------------------------------------------------
// With NSPLIT > 1 the kv range is chunked across blockIdx.z so small-batch
// decode fills the chip; each split writes fp32 partials
// partial[b, hk, split] = {m[G], l[G], acc[G][D]} merged by the kernel below.
template <int D, bool C8>
__global__ __launch_bounds__(PA_BLOCK) void paged_decode_attn_kernel(
    const ushort_t* __restrict__ q,
    const void* __restrict__ k_cache,      // [nblocks, bs, Hk, D] bf16|int8
    const void* __restrict__ v_cache,
    const float* __restrict__ k_scale,     // [nblocks*bs*Hk] (C8 only)
    const float* __restrict__ v_scale,
    const int* __restrict__ block_table,   // [B, max_blocks]
    const int* __restrict__ seq_lens,      // [B]
    ushort_t* __restrict__ out,
    float* __restrict__ partials,          // [B, Hk, NSPLIT, G*(2+D)] or null
    int B, int Hq, int Hk, int block_size, int max_blocks, float scale,
    int nsplit) {
    const ushort_t* k16 = (const ushort_t*)k_cache;
    const ushort_t* v16 = (const ushort_t*)v_cache;
    const signed char* k8 = (const signed char*)k_cache;
    const signed char* v8 = (const signed char*)v_cache;
    const int b = blockIdx.x;
    const int hk = blockIdx.y;
    const int split = blockIdx.z;
    const int G = Hq / Hk;
    const int wave = threadIdx.x >> 6;
    const int lane = threadIdx.x & 63;
    const int seq_len = seq_lens[b];
    if (seq_len <= 0) return;

    constexpr int EPL = D / 64;  // elements per lane (2 for D=128)

    // q for the G heads of this group -> registers (fp32), pre-scaled
    float qreg[MAX_GQA][EPL];
    const int hq0 = hk * G;
#pragma unroll
    for (int g = 0; g < MAX_GQA; g++) {
        if (g >= G) break;
        const ushort_t* qp = q + ((long long)b * Hq + hq0 + g) * D;
#pragma unroll
        for (int e = 0; e < EPL; e++)
            qreg[g][e] = bf16_to_f32(qp[lane * EPL + e]) * scale;
    }

    float m[MAX_GQA], l[MAX_GQA], acc[MAX_GQA][EPL];
#pragma unroll
    for (int g = 0; g < MAX_GQA; g++) {
        m[g] = -INFINITY;
        l[g] = 0.f;
#pragma unroll
        for (int e = 0; e < EPL; e++) acc[g][e] = 0.f;
    }

    // split chunk, then waves split the chunk into contiguous sub-ranges
    const int chunk = (seq_len + nsplit - 1) / nsplit;
    const int c0 = split * chunk;
    const int c1 = min(seq_len, c0 + chunk);
    const int per_wave = (c1 - c0 + PA_WAVES - 1) / PA_WAVES;
    const int j0 = c0 + wave * per_wave;
    const int j1 = min(c1, j0 + per_wave);
    const int* bt = block_table + (long long)b * max_blocks;

    // The per-token online-softmax update is a serial dependency chain
    // (wave_reduce -> exp -> acc scale).  Unrolling U tokens per iteration
    // pipelines the K/V loads and the U wave-reductions (independent), and
    // folds U softmax updates into ONE rescale (single max over the tile) —
    // long-context decode is latency-bound on this chain, not on HBM bytes.
    constexpr int U = 8;
    int j = j0;
    for (; j + U <= j1; j += U) {
        float kk[U][EPL], vv[U][EPL];
#pragma unroll
        for (int u = 0; u < U; u++) {
            const int jj = j + u;
            const int blk = bt[jj / block_size];
            const long long tok =
                ((long long)blk * block_size + (jj % block_size)) * Hk + hk;
            const long long base = tok * D;
            if (C8) {
                const float ks = k_scale[tok];
                const float vs = v_scale[tok];
#pragma unroll
                for (int e = 0; e < EPL; e++) {
                    kk[u][e] = (float)k8[base + lane * EPL + e] * ks;
                    vv[u][e] = (float)v8[base + lane * EPL + e] * vs;
                }
            } else if (EPL == 2) {
                short2v k2 = *reinterpret_cast<const short2v*>(k16 + base + lane * 2);
                short2v v2 = *reinterpret_cast<const short2v*>(v16 + base + lane * 2);
                kk[u][0] = bf16_to_f32((ushort_t)k2[0]);
                kk[u][1] = bf16_to_f32((ushort_t)k2[1]);
                vv[u][0] = bf16_to_f32((ushort_t)v2[0]);
                vv[u][1] = bf16_to_f32((ushort_t)v2[1]);
            } else {
#pragma unroll
                for (int e = 0; e < EPL; e++) {
                    kk[u][e] = bf16_to_f32(k16[base + lane * EPL + e]);
                    vv[u][e] = bf16_to_f32(v16[base + lane * EPL + e]);
                }
            }
        }
#pragma unroll
        for (int g = 0; g < MAX_GQA; g++) {
            if (g >= G) break;
            float sc[U];
#pragma unroll
            for (int u = 0; u < U; u++) {
                float s = 0.f;
#pragma unroll
                for (int e = 0; e < EPL; e++) s += qreg[g][e] * kk[u][e];
                sc[u] = s;
            }
            // U independent reductions pipeline their shuffle latency
#pragma unroll
            for (int u = 0; u < U; u++) sc[u] = wave_reduce_sum(sc[u]);
            float tile_max = sc[0];
#pragma unroll
            for (int u = 1; u < U; u++) tile_max = fmaxf(tile_max, sc[u]);
            const float m_new = fmaxf(m[g], tile_max);
            const float alpha = (m[g] == -INFINITY) ? 0.f : __expf(m[g] - m_new);
            float p[U];
            float psum = 0.f;
#pragma unroll
            for (int u = 0; u < U; u++) { p[u] = __expf(sc[u] - m_new); psum += p[u]; }
            l[g] = l[g] * alpha + psum;
#pragma unroll
            for (int e = 0; e < EPL; e++) {
                float a = acc[g][e] * alpha;
#pragma unroll
                for (int u = 0; u < U; u++) a += p[u] * vv[u][e];
                acc[g][e] = a;
            }
            m[g] = m_new;
        }
    }
    for (; j < j1; j++) {
        const int blk = bt[j / block_size];
        const long long tok = ((long long)blk * block_size + (j % block_size)) * Hk + hk;
        const long long base = tok * D;
        float kv_k[EPL], kv_v[EPL];
        if (C8) {
            const float ks = k_scale[tok];
            const float vs = v_scale[tok];
#pragma unroll
            for (int e = 0; e < EPL; e++) {
                kv_k[e] = (float)k8[base + lane * EPL + e] * ks;
                kv_v[e] = (float)v8[base + lane * EPL + e] * vs;
            }
        } else {
#pragma unroll
            for (int e = 0; e < EPL; e++) {
                kv_k[e] = bf16_to_f32(k16[base + lane * EPL + e]);
                kv_v[e] = bf16_to_f32(v16[base + lane * EPL + e]);
            }
        }
#pragma unroll
        for (int g = 0; g < MAX_GQA; g++) {
            if (g >= G) break;
            float s = 0.f;
#pragma unroll
            for (int e = 0; e < EPL; e++) s += qreg[g][e] * kv_k[e];
            s = wave_reduce_sum(s);
            float m_new = fmaxf(m[g], s);
            float alpha = (m[g] == -INFINITY) ? 0.f : __expf(m[g] - m_new);
            float p = __expf(s - m_new);
            l[g] = l[g] * alpha + p;
#pragma unroll
            for (int e = 0; e < EPL; e++) acc[g][e] = acc[g][e] * alpha + p * kv_v[e];
            m[g] = m_new;
        }
    }

    // merge the 4 waves' partial (m, l, acc) through LDS
    __shared__ float s_m[PA_WAVES][MAX_GQA];
    __shared__ float s_l[PA_WAVES][MAX_GQA];
    __shared__ float s_acc[PA_WAVES][MAX_GQA][64 * EPL];
#pragma unroll
    for (int g = 0; g < MAX_GQA; g++) {
        if (g >= G) break;
        if (lane == 0) { s_m[wave][g] = m[g]; s_l[wave][g] = l[g]; }
#pragma unroll
        for (int e = 0; e < EPL; e++) s_acc[wave][g][lane * EPL + e] = acc[g][e];
    }
    __syncthreads();
    if (wave == 0) {
#pragma unroll
        for (int g = 0; g < MAX_GQA; g++) {
            if (g >= G) break;
            float gm = -INFINITY;
            for (int w = 0; w < PA_WAVES; w++) gm = fmaxf(gm, s_m[w][g]);
            float gl = 0.f;
            float oacc[EPL] = {0.f};
            for (int w = 0; w < PA_WAVES; w++) {
                float mw = s_m[w][g];
                float a = (mw == -INFINITY) ? 0.f : __expf(mw - gm);
                gl += s_l[w][g] * a;
#pragma unroll
                for (int e = 0; e < EPL; e++)
                    oacc[e] += s_acc[w][g][lane * EPL + e] * a;
            }
            if (nsplit == 1) {
                float inv = (gl > 0.f) ? 1.0f / gl : 0.f;
                ushort_t* op = out + ((long long)b * Hq + hq0 + g) * D;
#pragma unroll
                for (int e = 0; e < EPL; e++)
                    op[lane * EPL + e] = f32_to_bf16(oacc[e] * inv);
            } else {
                float* pp = partials +
                    (((long long)b * Hk + hk) * nsplit + split) * (long long)G * (2 + D)
                    + (long long)g * (2 + D);
                if (lane == 0) { pp[0] = gm; pp[1] = gl; }
#pragma unroll
                for (int e = 0; e < EPL; e++)
                    pp[2 + lane * EPL + e] = oacc[e];
            }
        }
    }
}

// merge the per-split partials: one wave per (b, hk, g)
template <int D>
__global__ void paged_decode_merge_kernel(
    const float* __restrict__ partials, ushort_t* __restrict__ out,
    const int* __restrict__ seq_lens,
    int B, int Hq, int Hk, int nsplit) {
    const int b = blockIdx.x;
    const int hk = blockIdx.y;
    const int g = blockIdx.z;
    const int G = Hq / Hk;
    const int lane = threadIdx.x & 63;
    constexpr int EPL = D / 64;
    const int seq_len = seq_lens[b];
    const int chunk = (seq_len + nsplit - 1) / nsplit;
    const float* base = partials +
        (((long long)b * Hk + hk) * nsplit) * (long long)G * (2 + D)
        + (long long)g * (2 + D);
    float gm = -INFINITY;
    for (int sp = 0; sp < nsplit; sp++) {
        if (sp * chunk >= seq_len) break;
        gm = fmaxf(gm, base[(long long)sp * G * (2 + D)]);
    }
    float gl = 0.f;
    float oacc[EPL] = {0.f};
    for (int sp = 0; sp < nsplit; sp++) {
        if (sp * chunk >= seq_len) break;
        const float* pp = base + (long long)sp * G * (2 + D);
        float a = (pp[0] == -INFINITY) ? 0.f : __expf(pp[0] - gm);
        gl += pp[1] * a;
#pragma unroll
        for (int e = 0; e < EPL; e++)
            oacc[e] += pp[2 + lane * EPL + e] * a;
    }
    float inv = (gl > 0.f) ? 1.0f / gl : 0.f;
    ushort_t* op = out + ((long long)b * Hq + hk * G + g) * D;
#pragma unroll
    for (int e = 0; e < EPL; e++)
        op[lane * EPL + e] = f32_to_bf16(oacc[e] * inv);
}

// ---------------------------------------------------------------------------
// fused RoPE + paged cache append for a batch of new tokens.
// qkv: [B, T, (Hq + 2*Hk) * D] (fused projection output; T tokens per seq
// this step — 1 for decode, prompt length for prefill).
// positions p = seq_lens_before[b] + t; writes roped q to q_out [B, T, Hq, D]
// and roped k / raw v into the paged cache.
// ---------------------------------------------------------------------------
// MODE: 0 = bf16, 1 = int8 (absmax/127), 2 = int4 packed nibbles
// (offset-binary q+8, absmax/7) — reference append_attention_c8/_c4_impl
template <int D, int MODE>
__global__ void rope_cache_append_kernel(
    const ushort_t* __restrict__ qkv,
    ushort_t* __restrict__ q_out,
    void* __restrict__ k_cache, void* __restrict__ v_cache,
    float* __restrict__ k_scale, float* __restrict__ v_scale,
    const int* __restrict__ block_table, const int* __restrict__ seq_lens_before,
    const float* __restrict__ cos_t, const float* __restrict__ sin_t,  // [max_pos, D]
    int B, int T, int Hq, int Hk, int block_size, int max_blocks,
    const int* __restrict__ token_counts  // [B] tokens this step (<= T), or null
) {
    // grid: (B*T, Hq + 2*Hk); one wave (64 lanes) per head
    const int bt_idx = blockIdx.x;
    const int head = blockIdx.y;
    const int b = bt_idx / T;
    const int t = bt_idx % T;
    const int n_tok = token_counts ? token_counts[b] : T;
    if (t >= n_tok) return;
    const int lane = threadIdx.x & 63;
    const int H_all = Hq + 2 * Hk;
    const int pos = seq_lens_before[b] + t;

    const ushort_t* src = qkv + (((long long)b * T + t) * H_all + head) * D;
    constexpr int half = D / 2;
    constexpr int EPL = half / 64;  // rotation pairs per lane

    bool is_q = head < Hq;
    bool is_k = head >= Hq && head < Hq + Hk;

    if (is_q || is_k) {
        const float* cr = cos_t + (long long)pos * D;
        const float* sr = sin_t + (long long)pos * D;
        float o1[EPL > 0 ? EPL : 1], o2[EPL > 0 ? EPL : 1];
#pragma unroll
        for (int e = 0; e < EPL; e++) {
            int i = lane * EPL + e;
            float a = bf16_to_f32(src[i]);
            float bb = bf16_to_f32(src[i + half]);
            o1[e] = a * cr[i] - bb * sr[i];
            o2[e] = bb * cr[i + half] + a * sr[i + half];
        }
        if (is_q) {
            ushort_t* dst = q_out + (((long long)b * T + t) * Hq + head) * D;
#pragma unroll
            for (int e = 0; e < EPL; e++) {
                int i = lane * EPL + e;
                dst[i] = f32_to_bf16(o1[e]);
                dst[i + half] = f32_to_bf16(o2[e]);
            }
        } else {
            int hk = head - Hq;
            int blk = block_table[(long long)b * max_blocks + pos / block_size];
            if (blk < 0) return;  // inactive slot (device scheduler): no cache write
            const long long tok =
                ((long long)blk * block_size + pos % block_size) * Hk + hk;
            if (MODE == 1) {
                float amax = 0.f;
#pragma unroll
                for (int e = 0; e < EPL; e++)
                    amax = fmaxf(amax, fmaxf(fabsf(o1[e]), fabsf(o2[e])));
                amax = wave_reduce_max(amax);
                const float sc = fmaxf(amax, 1e-8f) / 127.f;
                signed char* dst = (signed char*)k_cache + tok * D;
#pragma unroll
                for (int e = 0; e < EPL; e++) {
                    int i = lane * EPL + e;
                    dst[i] = (signed char)lrintf(fminf(fmaxf(o1[e] / sc, -127.f), 127.f));
                    dst[i + half] = (signed char)lrintf(fminf(fmaxf(o2[e] / sc, -127.f), 127.f));
                }
                if (lane == 0) k_scale[tok] = sc;
            } else if (MODE == 2) {
                // int4: EPL(=2 at D=128) adjacent elements per lane pack
                // into one byte; nibbles are offset-binary (q+8)
                static_assert(MODE != 2 || D % 128 == 0 || D == 64, "");
                float amax = 0.f;
#pragma unroll
                for (int e = 0; e < EPL; e++)
                    amax = fmaxf(amax, fmaxf(fabsf(o1[e]), fabsf(o2[e])));
                amax = wave_reduce_max(amax);
                const float sc = fmaxf(amax, 1e-8f) / 7.f;
                unsigned char* dst = (unsigned char*)k_cache + tok * (D / 2);
                auto nib = [&](float v) {
                    int q = (int)lrintf(fminf(fmaxf(v / sc, -7.f), 7.f)) + 8;
                    return (unsigned)q;
                };
#pragma unroll
                for (int e = 0; e < EPL; e += 2) {
                    int i = lane * EPL + e;
                    dst[i / 2] = (unsigned char)((nib(o1[e]) << 4) | nib(o1[e + 1]));
                    dst[(i + half) / 2] =
                        (unsigned char)((nib(o2[e]) << 4) | nib(o2[e + 1]));
                }
                if (lane == 0) k_scale[tok] = sc;
            } else {
                ushort_t* dst = (ushort_t*)k_cache + tok * D;
#pragma unroll
                for (int e = 0; e < EPL; e++) {
                    int i = lane * EPL + e;
                    dst[i] = f32_to_bf16(o1[e]);
                    dst[i + half] = f32_to_bf16(o2[e]);
                }
            }
        }
    } else {
        // V: copy (or quantize) into the cache
        int hv = head - Hq - Hk;
        int blk = block_table[(long long)b * max_blocks + pos / block_size];
        if (blk < 0) return;  // inactive slot (device scheduler): no cache write
        const long long tok =
            ((long long)blk * block_size + pos % block_size) * Hk + hv;
        constexpr int VE = D / 64;
        if (MODE == 1) {
            float vals[VE];
            float amax = 0.f;
#pragma unroll
            for (int e = 0; e < VE; e++) {
                vals[e] = bf16_to_f32(src[lane * VE + e]);
                amax = fmaxf(amax, fabsf(vals[e]));
            }
            amax = wave_reduce_max(amax);
            const float sc = fmaxf(amax, 1e-8f) / 127.f;
            signed char* dst = (signed char*)v_cache + tok * D;
#pragma unroll
            for (int e = 0; e < VE; e++)
                dst[lane * VE + e] =
                    (signed char)lrintf(fminf(fmaxf(vals[e] / sc, -127.f), 127.f));
            if (lane == 0) v_scale[tok] = sc;
        } else if (MODE == 2) {
            float vals[VE];
            float amax = 0.f;
#pragma unroll
            for (int e = 0; e < VE; e++) {
                vals[e] = bf16_to_f32(src[lane * VE + e]);
                amax = fmaxf(amax, fabsf(vals[e]));
            }
            amax = wave_reduce_max(amax);
            const float sc = fmaxf(amax, 1e-8f) / 7.f;
            unsigned char* dst = (unsigned char*)v_cache + tok * (D / 2);
#pragma unroll
            for (int e = 0; e < VE; e += 2) {
                int q0 = (int)lrintf(fminf(fmaxf(vals[e] / sc, -7.f), 7.f)) + 8;
                int q1 = (int)lrintf(fminf(fmaxf(vals[e + 1] / sc, -7.f), 7.f)) + 8;
                dst[(lane * VE + e) / 2] = (unsigned char)((q0 << 4) | q1);
            }
            if (lane == 0) v_scale[tok] = sc;
        } else {
            ushort_t* dst = (ushort_t*)v_cache + tok * D;
#pragma unroll
            for (int e = 0; e < VE; e++) dst[lane * VE + e] = src[lane * VE + e];
        }
    }
}

// ---------------------------------------------------------------------------
// launchers
// ---------------------------------------------------------------------------
int paged_decode_nsplit(int B, int Hk) {
    // fill ~2x the 256 CUs; cap so chunks stay useful (4x measured worse:
    // split-merge partial traffic outweighs the extra block parallelism)
    int target = 512;
    int base = B * Hk;
    int nsplit = (target + base - 1) / base;
    if (nsplit < 1) nsplit = 1;
    if (nsplit > 64) nsplit = 64;
    return nsplit;
}

// v2/v3 (paged_attn_v2.hip): MFMA-tiled cooperative-staging decode kernels
bool launch_paged_decode_attn3(const void* q, const void* k_cache, const void* v_cache,
                               const float* k_scale, const float* v_scale,
                               const int* block_table, const int* seq_lens, void* out,
                               float* partials, int nsplit,
                               int B, int Hq, int Hk, int D, int block_size,
                               int max_blocks, float scale, int cache_mode,
                               hipStream_t stream);
bool launch_paged_decode_attn2(const void* q, const void* k_cache, const void* v_cache,
                               const float* k_scale, const float* v_scale,
                               const int* block_table, const int* seq_lens, void* out,
                               float* partials, int nsplit,
                               int B, int Hq, int Hk, int D, int block_size,
                               int max_blocks, float scale, int cache_mode,
                               hipStream_t stream);

void launch_paged_decode_attn(const void* q, const void* k_cache, const void* v_cache,
                              const float* k_scale, const float* v_scale,
                              const int* block_table, const int* seq_lens, void* out,
                              float* partials, int nsplit,
                              int B, int Hq, int Hk, int D, int block_size,
                              int max_blocks, float scale, int cache_mode,
                              hipStream_t stream) {
    static const char* kver = getenv("PNLP_DECODE_KERNEL");  // "2" forces v2
    const bool want_v2 = (kver != nullptr && kver[0] == '2');
    if ((!want_v2 &&
         launch_paged_decode_attn3(q, k_cache, v_cache, k_scale, v_scale,
                                   block_table, seq_lens, out, partials, nsplit,
                                   B, Hq, Hk, D, block_size, max_blocks, scale,
                                   cache_mode, stream)) ||
        launch_paged_decode_attn2(q, k_cache, v_cache, k_scale, v_scale,
                                  block_table, seq_lens, out, partials, nsplit,
                                  B, Hq, Hk, D, block_size, max_blocks, scale,
                                  cache_mode, stream)) {
        if (nsplit > 1) {
            if (D == 128)
                hipLaunchKernelGGL(paged_decode_merge_kernel<128>, dim3(B, Hk, Hq / Hk),
                                   dim3(64), 0, stream, partials, (ushort_t*)out,
                                   seq_lens, B, Hq, Hk, nsplit);
        }
        return;
    }
    dim3 grid(B, Hk, nsplit);
    const bool c8 = (k_scale != nullptr);
#define PA_LAUNCH(DD, CC)                                                          \
    hipLaunchKernelGGL((paged_decode_attn_kernel<DD, CC>), grid, dim3(PA_BLOCK),   \
                       0, stream, (const ushort_t*)q, k_cache, v_cache,            \
                       k_scale, v_scale, block_table, seq_lens, (ushort_t*)out,    \
                       partials, B, Hq, Hk, block_size, max_blocks, scale, nsplit)
    if (D == 128) {
        if (c8) PA_LAUNCH(128, true); else PA_LAUNCH(128, false);
        if (nsplit > 1)
            hipLaunchKernelGGL(paged_decode_merge_kernel<128>, dim3(B, Hk, Hq / Hk), dim3(64),
                               0, stream, partials, (ushort_t*)out, seq_lens, B, Hq, Hk, nsplit);
    } else if (D == 64) {
        if (c8) PA_LAUNCH(64, true); else PA_LAUNCH(64, false);
        if (nsplit > 1)
            hipLaunchKernelGGL(paged_decode_merge_kernel<64>, dim3(B, Hk, Hq / Hk), dim3(64),
                               0, stream, partials, (ushort_t*)out, seq_lens, B, Hq, Hk, nsplit);
    }
#undef PA_LAUNCH
}

void launch_rope_cache_append(const void* qkv, void* q_out, void* k_cache, void* v_cache,
                              float* k_scale, float* v_scale,
                              const int* block_table, const int* seq_lens_before,
                              const float* cos_t, const float* sin_t,
                              int B, int T, int Hq, int Hk, int D, int block_size,
                              int max_blocks, const int* token_counts,
                              int cache_mode, hipStream_t stream) {
    dim3 grid(B * T, Hq + 2 * Hk);
#define RC_LAUNCH(DD, MM)                                                         \
    hipLaunchKernelGGL((rope_cache_append_kernel<DD, MM>), grid, dim3(64), 0,     \
                       stream, (const ushort_t*)qkv, (ushort_t*)q_out,            \
                       k_cache, v_cache, k_scale, v_scale,                        \
                       block_table, seq_lens_before, cos_t, sin_t,                \
                       B, T, Hq, Hk, block_size, max_blocks, token_counts)
    if (D == 128) {
        if (cache_mode == 2) RC_LAUNCH(128, 2);
        else if (cache_mode == 1) RC_LAUNCH(128, 1);
        else RC_LAUNCH(128, 0);
    } else if (D == 64) {
        if (cache_mode == 1) RC_LAUNCH(64, 1); else RC_LAUNCH(64, 0);
    }
#undef RC_LAUNCH
}
