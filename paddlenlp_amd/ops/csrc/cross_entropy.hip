// Fused softmax cross-entropy — gfx950, bf16 logits, fp32 math.
//
// Equivalent of the reference's pretraining criterion loss path (llama
// ParallelCrossEntropy local form + fused CE).  The fwd saves
// per-row (max, log-sum-exp) so the bwd recomputes softmax without a second
// reduction pass.
//
// fwd:  loss[n] = lse[n] - (logit[n, y_n] - max[n]),  lse = log(sum exp(l - max))
// bwd:  dlogits[n, v] = dloss[n] * (softmax[n, v] - 1{v == y_n})
#include "common.h"

#define CE_BLOCK 512

// one block per row (V is large: 32k-128k)
__global__ void ce_fwd_kernel(
    const ushort_t* __restrict__ logits, const long long* __restrict__ labels,
    float* __restrict__ loss, float* __restrict__ maxlse,  // [N, 2]: (max, lse)
    long long N, int V, long long ignore_index) {
    __shared__ float scratch[16];
    for (long long row = blockIdx.x; row < N; row += gridDim.x) {
        const ushort_t* lr = logits + row * V;
        long long y = labels[row];
        float m = -INFINITY;
        for (int i = threadIdx.x * 8; i < V; i += blockDim.x * 8) {
            short8v v = *reinterpret_cast<const short8v*>(lr + i);
#pragma unroll
            for (int j = 0; j < 8; j++) m = fmaxf(m, bf16_to_f32((ushort_t)v[j]));
        }
        m = block_reduce_max(m, scratch);
        float se = 0.f;
        for (int i = threadIdx.x * 8; i < V; i += blockDim.x * 8) {
            short8v v = *reinterpret_cast<const short8v*>(lr + i);
#pragma unroll
            for (int j = 0; j < 8; j++) se += __expf(bf16_to_f32((ushort_t)v[j]) - m);
        }
        se = block_reduce_sum(se, scratch);
        if (threadIdx.x == 0) {
            float lse = __logf(se);
            maxlse[row * 2] = m;
            maxlse[row * 2 + 1] = lse;
            if (y == ignore_index) {
                loss[row] = 0.f;
            } else {
                float tgt = bf16_to_f32(lr[y]) - m;
                loss[row] = lse - tgt;
            }
        }
        __syncthreads();
    }
}

__global__ void ce_bwd_kernel(
    const float* __restrict__ dloss, const ushort_t* __restrict__ logits,
    const long long* __restrict__ labels, const float* __restrict__ maxlse,
    ushort_t* __restrict__ dlogits, long long N, int V, long long ignore_index) {
    for (long long row = blockIdx.x; row < N; row += gridDim.x) {
        const ushort_t* lr = logits + row * V;
        ushort_t* dr = dlogits + row * V;
        long long y = labels[row];
        float dl = (y == ignore_index) ? 0.f : dloss[row];
        float m = maxlse[row * 2];
        float inv_se = __expf(-maxlse[row * 2 + 1]);
        for (int i = threadIdx.x * 8; i < V; i += blockDim.x * 8) {
            short8v v = *reinterpret_cast<const short8v*>(lr + i);
            short8v out;
#pragma unroll
            for (int j = 0; j < 8; j++) {
                float p = __expf(bf16_to_f32((ushort_t)v[j]) - m) * inv_se;
                float g = dl * (p - ((i + j) == y ? 1.0f : 0.0f));
                out[j] = (short)f32_to_bf16(g);
            }
            *reinterpret_cast<short8v*>(dr + i) = out;
        }
    }
}

void launch_ce_fwd(const void* logits, const long long* labels, float* loss,
                   float* maxlse, long long N, int V, long long ignore_index,
                   hipStream_t stream) {
    int grid = (int)(N < 4096 ? N : 4096);
    hipLaunchKernelGGL(ce_fwd_kernel, dim3(grid), dim3(CE_BLOCK), 0, stream,
                       (const ushort_t*)logits, labels, loss, maxlse, N, V, ignore_index);
}

void launch_ce_bwd(const float* dloss, const void* logits, const long long* labels,
                   const float* maxlse, void* dlogits, long long N, int V,
                   long long ignore_index, hipStream_t stream) {
    int grid = (int)(N < 4096 ? N : 4096);
    hipLaunchKernelGGL(ce_bwd_kernel, dim3(grid), dim3(CE_BLOCK), 0, stream,
                       dloss, (const ushort_t*)logits, labels, maxlse,
                       (ushort_t*)dlogits, N, V, ignore_index);
}
