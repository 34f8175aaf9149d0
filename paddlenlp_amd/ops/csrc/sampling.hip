// Fused decode sampling pipeline + device-side block scheduler — gfx950.
//
// MI355X-native equivalents of the reference generation-runtime ops
// (SURVEY §2.9): get_token_penalty_multi_scores_v2
// (csrc/gpu/token_penalty_multi_scores_v2.cu:251), top_p_sampling_reject
// (csrc/gpu/sample_kernels/top_p_sampling_reject.cu:73),
// set_value_by_flags_and_idx_v2 / set_stop_value_multi_ends_v2 /
// update_inputs (set_value_by_flags.cu, stop_generation_multi_ends.cu,
// update_inputs.cu) and the paged-KV block scheduler step_paddle
// (csrc/gpu/step.cu:316).  Design notes:
//
//   * top-p is implemented threshold-style: an exponent-bucket histogram
//     of the unnormalized softmax masses (one pass) plus one linear
//     refinement pass locates the probability threshold tau such that
//     {p >= tau} carries ~top_p of the mass (within 1/64 of a power of
//     two); the sample is then drawn by CDF inversion over that set.
//     Same semantics as sort-and-truncate up to boundary ties, without
//     sorting 128k logits.
//   * the block scheduler runs as ONE workgroup with a device-resident
//     free list, so a decode loop makes no host round-trips: stopped
//     sequences release blocks, growing sequences allocate, and when the
//     free list is empty the longest running sequence is preempted
//     (flagged in is_block_step for host-side recovery/refill).
#include "common.h"
#include <hip/hip_runtime.h>

// ---------------------------------------------------------------------------
// block-wide float prefix sum (inclusive) for 512 threads
// ---------------------------------------------------------------------------
__device__ __forceinline__ float block_prefix_sum_512(float x, float* lds8) {
    const int lane = threadIdx.x & 63;
    const int wave = threadIdx.x >> 6;
    // wave-inclusive scan
#pragma unroll
    for (int off = 1; off < 64; off <<= 1) {
        float y = __shfl_up(x, off, 64);
        if (lane >= off) x += y;
    }
    if (lane == 63) lds8[wave] = x;
    __syncthreads();
    if (wave == 0 && lane < 8) {
        float v = lds8[lane];
#pragma unroll
        for (int off = 1; off < 8; off <<= 1) {
            float y = __shfl_up(v, off, 64);
            if (lane >= off) v += y;
        }
        lds8[lane] = v;
    }
    __syncthreads();
    float base = (wave > 0) ? lds8[wave - 1] : 0.f;
    __syncthreads();
    return x + base;
}

// ---------------------------------------------------------------------------
// repetition penalty scatter (in-place on bf16 logits)
// ---------------------------------------------------------------------------
__global__ void repetition_penalty_kernel(
    ushort_t* __restrict__ logits,          // [B, V]
    const long long* __restrict__ pre_ids,  // [B, max_len]
    const int* __restrict__ pre_lens,       // [B]
    const float* __restrict__ rep_pen,      // [B]
    int B, long long V, int max_len) {
    const int b = blockIdx.x;
    const float rp = rep_pen[b];
    if (rp == 1.0f) return;
    const int n = min(pre_lens[b], max_len);
    const long long* row = pre_ids + (long long)b * max_len;
    ushort_t* lrow = logits + (long long)b * V;
    for (int i = threadIdx.x; i < n; i += blockDim.x) {
        long long t = row[i];
        if (t < 0 || t >= V) continue;
        float l = bf16_to_f32(lrow[t]);
        lrow[t] = f32_to_bf16(l < 0 ? l * rp : l / rp);
    }
}

// ---------------------------------------------------------------------------
// fused temperature + top-p threshold sampling (or greedy when top_p <= 0)
// one 512-thread block per sequence
// ---------------------------------------------------------------------------
#define SMP_BLOCK 512
#define NBIN 64

__global__ __launch_bounds__(SMP_BLOCK) void topp_sample_kernel(
    const ushort_t* __restrict__ logits,   // [B, V] bf16
    const float* __restrict__ temperature, // [B]
    const float* __restrict__ top_p,       // [B]; <=0 -> greedy
    const float* __restrict__ uniform,     // [B] U(0,1)
    const long long* __restrict__ ban_eos_mask, // eos ids to ban, or null
    int n_ban,
    const int* __restrict__ cur_lens,      // [B] generated-so-far (min_length)
    const int* __restrict__ min_lens,      // [B] or null
    long long* __restrict__ out,           // [B]
    int B, long long V) {
    const int b = blockIdx.x;
    const int tid = threadIdx.x;
    const ushort_t* lrow = logits + (long long)b * V;
    const float invT = 1.0f / fmaxf(temperature[b], 1e-6f);
    const float tp = top_p[b];
    const bool ban_eos = (ban_eos_mask != nullptr) && (min_lens != nullptr) &&
                         (cur_lens[b] < min_lens[b]);

    __shared__ float s_hist[NBIN];
    __shared__ float s_red[8];
    __shared__ float s_scalar[4];
    __shared__ long long s_tok;

    auto banned = [&](long long v) {
        if (!ban_eos) return false;
        for (int e = 0; e < n_ban; e++)
            if (ban_eos_mask[e] == v) return true;
        return false;
    };

    // ---- pass 1: max (+ argmax for greedy) ----
    float m = -INFINITY;
    long long am = 0;
    for (long long v = tid; v < V; v += SMP_BLOCK) {
        if (banned(v)) continue;
        float s = bf16_to_f32(lrow[v]) * invT;
        if (s > m) { m = s; am = v; }
    }
    {   // block argmax via LDS ping-pong
        const int lane = tid & 63, wave = tid >> 6;
#pragma unroll
        for (int off = 32; off > 0; off >>= 1) {
            float om = __shfl_xor(m, off, 64);
            long long oa = __shfl_xor(am, off, 64);
            if (om > m || (om == m && oa < am)) { m = om; am = oa; }
        }
        __shared__ float wm[8];
        __shared__ long long wa[8];
        if (lane == 0) { wm[wave] = m; wa[wave] = am; }
        __syncthreads();
        if (tid == 0) {
            for (int w = 1; w < 8; w++)
                if (wm[w] > wm[0] || (wm[w] == wm[0] && wa[w] < wa[0])) {
                    wm[0] = wm[w]; wa[0] = wa[w];
                }
        }
        __syncthreads();
        m = wm[0]; am = wa[0];
    }
    if (tp <= 0.f || m == -INFINITY) {       // greedy (or fully banned row)
        if (tid == 0) out[b] = am;
        return;
    }

    // ---- pass 2: Z + exponent histogram of q = exp(s - m) ----
    for (int i = tid; i < NBIN; i += SMP_BLOCK) s_hist[i] = 0.f;
    __syncthreads();
    float zpart = 0.f;
    for (long long v = tid; v < V; v += SMP_BLOCK) {
        if (banned(v)) continue;
        float q = __expf(bf16_to_f32(lrow[v]) * invT - m);
        zpart += q;
        if (q > 0.f) {
            int bin = min(NBIN - 1, max(0, -(int)ilogbf(q)));
            atomicAdd(&s_hist[bin], q);
        }
    }
    // block sum of zpart
    {
        const int lane = tid & 63, wave = tid >> 6;
#pragma unroll
        for (int off = 32; off > 0; off >>= 1) zpart += __shfl_xor(zpart, off, 64);
        if (lane == 0) s_red[wave] = zpart;
        __syncthreads();
        if (tid == 0) {
            float z = 0.f;
            for (int w = 0; w < 8; w++) z += s_red[w];
            s_scalar[0] = z;
        }
        __syncthreads();
    }
    const float Z = s_scalar[0];
    const float target_mass = tp * Z;

    // scan bins high-probability-first to locate the threshold bin
    __syncthreads();
    if (tid == 0) {
        float cum = 0.f;
        int beta = NBIN - 1;
        for (int i = 0; i < NBIN; i++) {
            if (cum + s_hist[i] >= target_mass) { beta = i; break; }
            cum += s_hist[i];
        }
        s_scalar[1] = (float)beta;
        s_scalar[2] = cum;      // mass above bin beta
    }
    __syncthreads();
    const int beta = (int)s_scalar[1];
    const float cum_above = s_scalar[2];

    // ---- pass 2b: linear sub-histogram inside bin beta -> tau ----
    // bin k holds q in [2^-k, 2^(1-k)) (ilogbf bucketing), capped at 1
    const float qlo = exp2f(-(float)beta);
    const float qhi = (beta == 0) ? 1.0001f : exp2f(1.0f - (float)beta);
    for (int i = tid; i < NBIN; i += SMP_BLOCK) s_hist[i] = 0.f;
    __syncthreads();
    for (long long v = tid; v < V; v += SMP_BLOCK) {
        if (banned(v)) continue;
        float q = __expf(bf16_to_f32(lrow[v]) * invT - m);
        if (q >= qlo && q < qhi) {
            int sb = min(NBIN - 1, (int)((qhi - q) / (qhi - qlo) * NBIN));
            atomicAdd(&s_hist[sb], q);
        }
    }
    __syncthreads();
    if (tid == 0) {
        float cum = cum_above;
        int sb = NBIN - 1;
        for (int i = 0; i < NBIN; i++) {
            cum += s_hist[i];
            if (cum >= target_mass) { sb = i; break; }
        }
        s_scalar[1] = qhi - (float)(sb + 1) / NBIN * (qhi - qlo);  // tau
        s_scalar[2] = cum;                                          // kept mass
        s_tok = -1;
    }
    __syncthreads();
    const float tau = s_scalar[1];
    const float kept = fmaxf(s_scalar[2], 1e-30f);

    // ---- pass 3: CDF inversion over {q >= tau} ----
    const float target = uniform[b] * kept;
    float running = 0.f;
    for (long long base = 0; base < V; base += SMP_BLOCK) {
        long long v = base + tid;
        float q = 0.f;
        if (v < V && !banned(v)) {
            float qq = __expf(bf16_to_f32(lrow[v]) * invT - m);
            if (qq >= tau) q = qq;
        }
        float pre = block_prefix_sum_512(q, s_red);
        // total of the whole chunk lives in the last thread; broadcast
        __shared__ float s_chunk;
        if (tid == SMP_BLOCK - 1) s_chunk = pre;
        __syncthreads();
        float chunk_total = s_chunk;
        if (q > 0.f && s_tok < 0) {
            float lo = running + pre - q, hi = running + pre;
            if (target > lo && target <= hi) s_tok = v;
        }
        __syncthreads();
        if (s_tok >= 0) break;
        running += chunk_total;
        __syncthreads();
    }
    if (tid == 0) out[b] = (s_tok >= 0) ? s_tok : am;
}

// ---------------------------------------------------------------------------
// post-sample update: eos detection, history append, length bump,
// not_need_stop recompute (reference update_inputs + set_value_by_flags +
// set_stop_value_multi_ends)
// ---------------------------------------------------------------------------
__global__ void decode_update_kernel(
    long long* __restrict__ tokens,       // [B] sampled this step (in/out)
    signed char* __restrict__ stop_flags, // [B]
    const signed char* __restrict__ active,  // [B] slot occupied
    long long* __restrict__ pre_ids,      // [B, max_len]
    int* __restrict__ pre_lens,           // [B]
    int* __restrict__ seq_lens,           // [B]
    const long long* __restrict__ eos_ids, int n_eos,
    int* __restrict__ not_need_stop,      // [1]
    const int* __restrict__ max_new, long long pad_id,
    int B, int max_len) {
    __shared__ int s_any;
    if (threadIdx.x == 0) s_any = 0;
    __syncthreads();
    for (int b = threadIdx.x; b < B; b += blockDim.x) {
        if (!active[b]) continue;
        if (stop_flags[b]) { tokens[b] = pad_id; continue; }
        long long t = tokens[b];
        bool stop = false;
        for (int e = 0; e < n_eos; e++) stop |= (t == eos_ids[e]);
        int n = pre_lens[b];
        if (n < max_len) { pre_ids[(long long)b * max_len + n] = t; pre_lens[b] = n + 1; }
        seq_lens[b] += 1;
        if (max_new && pre_lens[b] >= max_new[b]) stop = true;
        if (stop) stop_flags[b] = 1;
        else atomicOr(&s_any, 1);
    }
    __syncthreads();
    if (threadIdx.x == 0 && blockIdx.x == 0) *not_need_stop = s_any;
}

// ---------------------------------------------------------------------------
// device block scheduler (reference step.cu free_and_dispatch_block :19):
// ONE workgroup.  Frees stopped sequences' blocks, allocates a block to
// any sequence whose next token crosses a block boundary, preempts the
// longest sequence when the free list runs dry.
// ---------------------------------------------------------------------------
__global__ __launch_bounds__(512) void block_step_kernel(
    int* __restrict__ block_table,     // [B, max_blocks]
    int* __restrict__ seq_lens,        // [B]
    signed char* __restrict__ stop_flags,
    signed char* __restrict__ active,
    int* __restrict__ free_list,       // [num_blocks]
    int* __restrict__ free_count,      // [1]
    signed char* __restrict__ is_block_step,  // [B] preempted flag
    int B, int block_size, int max_blocks) {
    const int tid = threadIdx.x;
    __shared__ int s_need[512];
    __shared__ int s_longest;
    __shared__ int s_longlen;

    // 1) release blocks of freshly stopped sequences
    for (int b = tid; b < B; b += 512) {
        if (active[b] && stop_flags[b] && seq_lens[b] > 0) {
            int n = (seq_lens[b] + block_size - 1) / block_size;
            int* row = block_table + (long long)b * max_blocks;
            for (int j = 0; j < n; j++) {
                if (row[j] >= 0) {
                    int idx = atomicAdd(free_count, 1);
                    free_list[idx] = row[j];
                    row[j] = -1;
                }
            }
            seq_lens[b] = 0;
            active[b] = 0;
        }
    }
    __syncthreads();

    // 2) allocate for sequences crossing a block boundary next step
    s_need[tid] = 0;
    for (int b = tid; b < B; b += 512) {
        int need = 0;
        if (active[b] && !stop_flags[b] && !is_block_step[b]) {
            int len = seq_lens[b];
            if (len > 0 && (len % block_size) == 0) {
                int pos = len / block_size;
                if (pos < max_blocks &&
                    block_table[(long long)b * max_blocks + pos] < 0)
                    need = 1;
            }
        }
        if (need) {
            int idx = atomicSub(free_count, 1) - 1;
            if (idx >= 0) {
                int pos = seq_lens[b] / block_size;
                block_table[(long long)b * max_blocks + pos] = free_list[idx];
            } else {
                atomicAdd(free_count, 1);
                s_need[tid] = b + 1;   // unsatisfied: remember who
            }
        }
    }
    __syncthreads();

    // 3) preemption loop: while someone is starving, evict the longest
    //    running sequence (flag it for host recovery) and retry
    for (int round = 0; round < 8; round++) {
        int starving = 0;
        for (int t = 0; t < 512; t++) starving |= s_need[t];
        if (!starving) break;
        if (tid == 0) { s_longest = -1; s_longlen = -1; }
        __syncthreads();
        for (int b = tid; b < B; b += 512) {
            if (active[b] && !stop_flags[b] && !is_block_step[b]) {
                atomicMax(&s_longlen, seq_lens[b]);
            }
        }
        __syncthreads();
        for (int b = tid; b < B; b += 512) {
            if (active[b] && !stop_flags[b] && !is_block_step[b] &&
                seq_lens[b] == s_longlen)
                atomicMax(&s_longest, b);
        }
        __syncthreads();
        if (s_longest < 0) break;
        // evict on thread 0
        if (tid == 0) {
            int b = s_longest;
            int n = (seq_lens[b] + block_size - 1) / block_size;
            int* row = block_table + (long long)b * max_blocks;
            for (int j = 0; j < n; j++) {
                if (row[j] >= 0) {
                    int idx = atomicAdd(free_count, 1);
                    free_list[idx] = row[j];
                    row[j] = -1;
                }
            }
            is_block_step[b] = 1;     // host re-prefills this sequence later
        }
        __syncthreads();
        // retry unsatisfied allocations
        if (s_need[tid]) {
            int b = s_need[tid] - 1;
            if (b != s_longest) {
                int idx = atomicSub(free_count, 1) - 1;
                if (idx >= 0) {
                    int pos = seq_lens[b] / block_size;
                    block_table[(long long)b * max_blocks + pos] = free_list[idx];
                    s_need[tid] = 0;
                }else {
                    atomicAdd(free_count, 1);
                }
            } else {
                s_need[tid] = 0;
            }
        }
        __syncthreads();
    }
}

// ---------------------------------------------------------------------------
// launchers
// ---------------------------------------------------------------------------
void launch_repetition_penalty(void* logits, const long long* pre_ids,
                               const int* pre_lens, const float* rep_pen,
                               int B, long long V, int max_len, hipStream_t s) {
    hipLaunchKernelGGL(repetition_penalty_kernel, dim3(B), dim3(256), 0, s,
                       (ushort_t*)logits, pre_ids, pre_lens, rep_pen, B, V, max_len);
}

void launch_topp_sample(const void* logits, const float* temperature,
                        const float* top_p, const float* uniform,
                        const long long* ban_eos, int n_ban,
                        const int* cur_lens, const int* min_lens,
                        long long* out, int B, long long V, hipStream_t s) {
    hipLaunchKernelGGL(topp_sample_kernel, dim3(B), dim3(SMP_BLOCK), 0, s,
                       (const ushort_t*)logits, temperature, top_p, uniform,
                       ban_eos, n_ban, cur_lens, min_lens, out, B, V);
}

void launch_decode_update(long long* tokens, signed char* stop_flags,
                          const signed char* active, long long* pre_ids,
                          int* pre_lens, int* seq_lens,
                          const long long* eos_ids, int n_eos,
                          int* not_need_stop, const int* max_new,
                          long long pad_id, int B, int max_len, hipStream_t s) {
    hipLaunchKernelGGL(decode_update_kernel, dim3(1), dim3(512), 0, s,
                       tokens, stop_flags, active, pre_ids, pre_lens, seq_lens,
                       eos_ids, n_eos, not_need_stop, max_new, pad_id, B, max_len);
}

void launch_block_step(int* block_table, int* seq_lens, signed char* stop_flags,
                       signed char* active, int* free_list, int* free_count,
                       signed char* is_block_step, int B, int block_size,
                       int max_blocks, hipStream_t s) {
    hipLaunchKernelGGL(block_step_kernel, dim3(1), dim3(512), 0, s,
                       block_table, seq_lens, stop_flags, active, free_list,
                       free_count, is_block_step, B, block_size, max_blocks);
}
