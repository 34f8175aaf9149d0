// Fused AdamW — gfx950 multi-tensor update with fp32 master weights.
//
// Replaces the reference's multi-tensor adamw (SURVEY §2.9 training-side
// fused ops; reference trainer.py:1817 optimizer).  bf16 params + grads,
// fp32 moments + master.  Chunked multi-tensor: host packs (pointers,
// numels) into a device metadata buffer; each block grid-strides over one
// chunk so a single launch covers every parameter.
#include "common.h"

#define ADAMW_BLOCK 256
#define ADAMW_CHUNK (1 << 20)  // 1M elements per chunk
#define MAX_TENSORS_PER_LAUNCH 320

struct AdamWChunk {
    void* param;       // bf16 or fp32
    const void* grad;  // matches param dtype
    float* m;
    float* v;
    float* master;     // nullptr when param is fp32
    long long offset;  // element offset of this chunk within the tensor
    long long n;       // elements in this chunk
    int is_bf16;
};

__global__ void adamw_kernel(
    const AdamWChunk* __restrict__ chunks, int n_chunks,
    float lr, float beta1, float beta2, float eps, float wd,
    float bias1, float bias2) {
    int c = blockIdx.x;
    if (c >= n_chunks) return;
    AdamWChunk ch = chunks[c];
    float inv_b1 = 1.0f / bias1;
    float inv_b2 = 1.0f / bias2;
    float decay = 1.0f - lr * wd;
    if (ch.is_bf16) {
        ushort_t* p = (ushort_t*)ch.param + ch.offset;
        const ushort_t* g = (const ushort_t*)ch.grad + ch.offset;
        float* m = ch.m + ch.offset;
        float* v = ch.v + ch.offset;
        float* w = ch.master + ch.offset;
        for (long long i = threadIdx.x * 4; i + 3 < ch.n; i += (long long)blockDim.x * 4) {
            short4v gv = *reinterpret_cast<const short4v*>(g + i);
            f32x4 mv = *reinterpret_cast<const f32x4*>(m + i);
            f32x4 vv = *reinterpret_cast<const f32x4*>(v + i);
            f32x4 wv = *reinterpret_cast<const f32x4*>(w + i);
            short4v pv;
#pragma unroll
            for (int j = 0; j < 4; j++) {
                float gf = bf16_to_f32((ushort_t)gv[j]);
                mv[j] = beta1 * mv[j] + (1.0f - beta1) * gf;
                vv[j] = beta2 * vv[j] + (1.0f - beta2) * gf * gf;
                float denom = __fsqrt_rn(vv[j] * inv_b2) + eps;
                wv[j] = wv[j] * decay - lr * (mv[j] * inv_b1) / denom;
                pv[j] = (short)f32_to_bf16(wv[j]);
            }
            *reinterpret_cast<f32x4*>(m + i) = mv;
            *reinterpret_cast<f32x4*>(v + i) = vv;
            *reinterpret_cast<f32x4*>(w + i) = wv;
            *reinterpret_cast<short4v*>(p + i) = pv;
        }
        // tail (n % 4)
        long long tail_start = (ch.n / 4) * 4;
        for (long long i = tail_start + threadIdx.x; i < ch.n; i += blockDim.x) {
            float gf = bf16_to_f32(g[i]);
            float mf = beta1 * m[i] + (1.0f - beta1) * gf;
            float vf = beta2 * v[i] + (1.0f - beta2) * gf * gf;
            float denom = __fsqrt_rn(vf * inv_b2) + eps;
            float wf = w[i] * decay - lr * (mf * inv_b1) / denom;
            m[i] = mf; v[i] = vf; w[i] = wf;
            p[i] = f32_to_bf16(wf);
        }
    } else {
        float* p = (float*)ch.param + ch.offset;
        const float* g = (const float*)ch.grad + ch.offset;
        float* m = ch.m + ch.offset;
        float* v = ch.v + ch.offset;
        for (long long i = threadIdx.x; i < ch.n; i += blockDim.x) {
            float gf = g[i];
            float mf = beta1 * m[i] + (1.0f - beta1) * gf;
            float vf = beta2 * v[i] + (1.0f - beta2) * gf * gf;
            float denom = __fsqrt_rn(vf * inv_b2) + eps;
            float wf = p[i] * decay - lr * (mf * inv_b1) / denom;
            m[i] = mf; v[i] = vf; p[i] = wf;
        }
    }
}

void launch_adamw(const AdamWChunk* dev_chunks, int n_chunks,
                  float lr, float beta1, float beta2, float eps, float wd,
                  float bias1, float bias2, hipStream_t stream) {
    hipLaunchKernelGGL(adamw_kernel, dim3(n_chunks), dim3(ADAMW_BLOCK), 0, stream,
                       dev_chunks, n_chunks, lr, beta1, beta2, eps, wd, bias1, bias2);
}
