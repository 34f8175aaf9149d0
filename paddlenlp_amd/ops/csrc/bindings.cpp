// Python bindings for the paddlenlp_amd gfx950 kernel pack.
// Built by setup.py via torch.utils.cpp_extension (hipcc, PYTORCH_ROCM_ARCH=gfx950).
#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>
#include <vector>

// ---- launcher decls (implemented in the .hip files) ----
void launch_rms_norm_fwd(const void*, const void*, void*, float*, long long, int, float, hipStream_t);
void launch_rms_norm_bwd(const void*, const void*, const void*, const float*, void*, float*, float*, void*, bool, long long, int, int, hipStream_t);
void launch_rope(const void*, void*, const float*, const float*, long long, int, int, int, bool, hipStream_t);
void launch_swiglu_fwd(const void*, void*, long long, int, hipStream_t);
void launch_swiglu_bwd(const void*, const void*, void*, long long, int, hipStream_t);
void launch_ce_fwd(const void*, const long long*, float*, float*, long long, int, long long, hipStream_t);
void launch_ce_bwd(const float*, const void*, const long long*, const float*, void*, long long, int, long long, hipStream_t);
void launch_mfma_probe(const void*, const void*, float*, hipStream_t);
void launch_flash_fwd(const void*, const void*, const void*, void*, float*, int, int, int, int, int, int, float, bool, hipStream_t);
void launch_flash_fwd_variant(const void*, const void*, const void*, void*, float*, int, int, int, int, int, int, float, bool, int, hipStream_t);
void launch_flash_bwd(const void*, const void*, const void*, const void*, const void*, const float*, float*, void*, void*, void*, int, int, int, int, int, int, float, bool, hipStream_t);
void launch_flash_bwd_variant(const void*, const void*, const void*, const void*, const void*, const float*, float*, void*, void*, void*, int, int, int, int, int, int, float, bool, int, hipStream_t);
void launch_flash_fwd_mask(const void*, const void*, const void*, void*, float*, const int*, int, int, int, int, int, int, float, hipStream_t);
void launch_flash_bwd_mask(const void*, const void*, const void*, const void*, const void*, const float*, float*, void*, void*, void*, const int*, int, int, int, int, int, int, float, hipStream_t);

struct AdamWChunk {
    void* param;
    const void* grad;
    float* m;
    float* v;
    float* master;
    long long offset;
    long long n;
    int is_bf16;
};
void launch_adamw(const AdamWChunk*, int, float, float, float, float, float, float, float, hipStream_t);
void launch_paged_decode_attn(const void*, const void*, const void*, const float*, const float*, const int*, const int*, void*, float*, int, int, int, int, int, int, int, float, int, hipStream_t);
int paged_decode_nsplit(int, int);
void launch_wint8_gemv(const void*, const void*, const float*, void*, int, int, int, hipStream_t);
void launch_fp8_rowwise_quant(const void*, void*, float*, long long, int, hipStream_t);
void launch_skinny_gemm(const void*, const void*, float*, void*, int, int, int, int, hipStream_t);
int skinny_gemm_ksplit(int, int, int);
void launch_rope_cache_append(const void*, void*, void*, void*, float*, float*, const int*, const int*, const float*, const float*, int, int, int, int, int, int, int, const int*, int, hipStream_t);

#define CHECK_GPU(x) TORCH_CHECK(x.is_cuda(), #x " must be on GPU")
#define CHECK_CONTIG(x) TORCH_CHECK(x.is_contiguous(), #x " must be contiguous")
#define CHECK_BF16(x) TORCH_CHECK(x.scalar_type() == torch::kBFloat16, #x " must be bf16")

static hipStream_t cur_stream() {
    return at::hip::getCurrentHIPStream().stream();
}

// ---------------------------------------------------------------------------
std::vector<torch::Tensor> rms_norm_fwd(torch::Tensor x, torch::Tensor w, double eps) {
    CHECK_GPU(x); CHECK_CONTIG(x); CHECK_BF16(x); CHECK_BF16(w);
    int H = x.size(-1);
    TORCH_CHECK(H % 8 == 0, "hidden size must be a multiple of 8");
    long long rows = x.numel() / H;
    auto y = torch::empty_like(x);
    auto invrms = torch::empty({rows}, x.options().dtype(torch::kFloat32));
    launch_rms_norm_fwd(x.data_ptr(), w.data_ptr(), y.data_ptr(),
                        invrms.data_ptr<float>(), rows, H, (float)eps, cur_stream());
    return {y, invrms};
}

std::vector<torch::Tensor> rms_norm_bwd(torch::Tensor dy, torch::Tensor x, torch::Tensor w,
                                        torch::Tensor invrms) {
    CHECK_GPU(dy); CHECK_CONTIG(dy); CHECK_BF16(dy);
    int H = x.size(-1);
    long long rows = x.numel() / H;
    int P = (int)std::min<long long>(rows, 2048);
    auto dx = torch::empty_like(x);
    auto dw_partial = torch::zeros({P, H}, x.options().dtype(torch::kFloat32));
    bool w_bf16 = w.scalar_type() == torch::kBFloat16;
    auto dw32 = torch::zeros({H}, x.options().dtype(torch::kFloat32));
    auto dw = w_bf16 ? torch::empty_like(w) : dw32;
    launch_rms_norm_bwd(dy.data_ptr(), x.data_ptr(), w.data_ptr(),
                        invrms.data_ptr<float>(), dx.data_ptr(),
                        dw_partial.data_ptr<float>(), dw32.data_ptr<float>(),
                        dw.data_ptr(), w_bf16, rows, H, P, cur_stream());
    return {dx, dw};
}

// ---------------------------------------------------------------------------
std::vector<torch::Tensor> rope_fwd(torch::Tensor q, torch::Tensor k,
                                    torch::Tensor cos_t, torch::Tensor sin_t,
                                    bool backward) {
    CHECK_GPU(q); CHECK_CONTIG(q); CHECK_BF16(q); CHECK_CONTIG(k); CHECK_BF16(k);
    TORCH_CHECK(q.dim() == 4 && k.dim() == 4, "q/k must be [B,S,H,D]");
    int B = q.size(0), S = q.size(1), Hq = q.size(2), D = q.size(3);
    int Hk = k.size(2);
    TORCH_CHECK(D % 8 == 0, "head dim must be a multiple of 8");
    auto cf = cos_t.to(torch::kFloat32).contiguous();
    auto sf = sin_t.to(torch::kFloat32).contiguous();
    TORCH_CHECK(cf.size(0) == S && cf.size(-1) == D, "cos table must be [S, D]");
    auto q_out = torch::empty_like(q);
    auto k_out = torch::empty_like(k);
    launch_rope(q.data_ptr(), q_out.data_ptr(), cf.data_ptr<float>(), sf.data_ptr<float>(),
                (long long)B * S * Hq, Hq, D, S, backward, cur_stream());
    launch_rope(k.data_ptr(), k_out.data_ptr(), cf.data_ptr<float>(), sf.data_ptr<float>(),
                (long long)B * S * Hk, Hk, D, S, backward, cur_stream());
    return {q_out, k_out};
}

// ---------------------------------------------------------------------------
torch::Tensor swiglu_fwd(torch::Tensor x) {
    CHECK_GPU(x); CHECK_CONTIG(x); CHECK_BF16(x);
    int twoI = x.size(-1);
    TORCH_CHECK(twoI % 16 == 0, "last dim must be a multiple of 16");
    int I = twoI / 2;
    long long N = x.numel() / twoI;
    auto sizes = x.sizes().vec();
    sizes.back() = I;
    auto y = torch::empty(sizes, x.options());
    launch_swiglu_fwd(x.data_ptr(), y.data_ptr(), N, I, cur_stream());
    return y;
}

torch::Tensor swiglu_bwd(torch::Tensor dy, torch::Tensor x) {
    CHECK_GPU(dy); CHECK_CONTIG(dy); CHECK_BF16(dy);
    int twoI = x.size(-1);
    int I = twoI / 2;
    long long N = x.numel() / twoI;
    auto dx = torch::empty_like(x);
    launch_swiglu_bwd(dy.data_ptr(), x.data_ptr(), dx.data_ptr(), N, I, cur_stream());
    return dx;
}

// ---------------------------------------------------------------------------
std::vector<torch::Tensor> cross_entropy_fwd(torch::Tensor logits, torch::Tensor labels,
                                             int64_t ignore_index) {
    CHECK_GPU(logits); CHECK_CONTIG(logits); CHECK_BF16(logits);
    TORCH_CHECK(logits.dim() == 2, "logits must be [N, V]");
    long long N = logits.size(0);
    int V = logits.size(1);
    TORCH_CHECK(V % 8 == 0, "vocab must be a multiple of 8");
    auto labels64 = labels.to(torch::kInt64).contiguous();
    auto loss = torch::empty({N}, logits.options().dtype(torch::kFloat32));
    auto maxlse = torch::empty({N, 2}, logits.options().dtype(torch::kFloat32));
    launch_ce_fwd(logits.data_ptr(), reinterpret_cast<const long long*>(labels64.data_ptr<int64_t>()),
                  loss.data_ptr<float>(), maxlse.data_ptr<float>(), N, V,
                  ignore_index, cur_stream());
    return {loss, maxlse};
}

torch::Tensor cross_entropy_bwd(torch::Tensor dloss, torch::Tensor logits,
                                torch::Tensor labels, torch::Tensor maxlse,
                                int64_t ignore_index) {
    CHECK_GPU(dloss);
    long long N = logits.size(0);
    int V = logits.size(1);
    auto labels64 = labels.to(torch::kInt64).contiguous();
    auto dl = dloss.to(torch::kFloat32).contiguous();
    auto dlogits = torch::empty_like(logits);
    launch_ce_bwd(dl.data_ptr<float>(), logits.data_ptr(),
                  reinterpret_cast<const long long*>(labels64.data_ptr<int64_t>()), maxlse.data_ptr<float>(),
                  dlogits.data_ptr(), N, V, ignore_index, cur_stream());
    return dlogits;
}

// ---------------------------------------------------------------------------
torch::Tensor mfma_probe(torch::Tensor A, torch::Tensor B) {
    CHECK_GPU(A); CHECK_BF16(A);
    auto C = torch::empty({16, 16}, A.options().dtype(torch::kFloat32));
    launch_mfma_probe(A.contiguous().data_ptr(), B.contiguous().data_ptr(),
                      C.data_ptr<float>(), cur_stream());
    return C;
}

void launch_mfma32_probe(const void*, const void*, float*, hipStream_t);
void launch_permlane_probe(int*, int*, hipStream_t);
void launch_repetition_penalty(void*, const long long*, const int*, const float*,
                               int, long long, int, hipStream_t);
void launch_topp_sample(const void*, const float*, const float*, const float*,
                        const long long*, int, const int*, const int*,
                        long long*, int, long long, hipStream_t);
void launch_decode_update(long long*, signed char*, const signed char*, long long*,
                          int*, int*, const long long*, int, int*, const int*,
                          long long, int, int, hipStream_t);
void launch_block_step(int*, int*, signed char*, signed char*, int*, int*,
                       signed char*, int, int, int, hipStream_t);

void apply_repetition_penalty(torch::Tensor logits, torch::Tensor pre_ids,
                              torch::Tensor pre_lens, torch::Tensor rep_pen) {
    CHECK_GPU(logits); CHECK_CONTIG(logits); CHECK_BF16(logits);
    int B = logits.size(0);
    long long V = logits.size(1);
    launch_repetition_penalty(logits.data_ptr(),
                              reinterpret_cast<const long long*>(pre_ids.data_ptr<int64_t>()),
                              pre_lens.data_ptr<int>(), rep_pen.data_ptr<float>(),
                              B, V, pre_ids.size(1), cur_stream());
}

torch::Tensor topp_sample(torch::Tensor logits, torch::Tensor temperature,
                          torch::Tensor top_p, torch::Tensor uniform,
                          c10::optional<torch::Tensor> ban_eos,
                          c10::optional<torch::Tensor> cur_lens,
                          c10::optional<torch::Tensor> min_lens) {
    CHECK_GPU(logits); CHECK_CONTIG(logits); CHECK_BF16(logits);
    int B = logits.size(0);
    long long V = logits.size(1);
    auto out = torch::empty({B}, logits.options().dtype(torch::kInt64));
    launch_topp_sample(
        logits.data_ptr(), temperature.data_ptr<float>(), top_p.data_ptr<float>(),
        uniform.data_ptr<float>(),
        ban_eos ? reinterpret_cast<const long long*>(ban_eos->data_ptr<int64_t>()) : nullptr,
        ban_eos ? (int)ban_eos->numel() : 0,
        cur_lens ? cur_lens->data_ptr<int>() : nullptr,
        min_lens ? min_lens->data_ptr<int>() : nullptr,
        reinterpret_cast<long long*>(out.data_ptr<int64_t>()), B, V, cur_stream());
    return out;
}

void decode_update(torch::Tensor tokens, torch::Tensor stop_flags, torch::Tensor active,
                   torch::Tensor pre_ids, torch::Tensor pre_lens, torch::Tensor seq_lens,
                   torch::Tensor eos_ids, torch::Tensor not_need_stop,
                   c10::optional<torch::Tensor> max_new, int64_t pad_id) {
    int B = tokens.size(0);
    launch_decode_update(
        reinterpret_cast<long long*>(tokens.data_ptr<int64_t>()),
        stop_flags.data_ptr<signed char>(), active.data_ptr<signed char>(),
        reinterpret_cast<long long*>(pre_ids.data_ptr<int64_t>()),
        pre_lens.data_ptr<int>(), seq_lens.data_ptr<int>(),
        reinterpret_cast<const long long*>(eos_ids.data_ptr<int64_t>()),
        (int)eos_ids.numel(), not_need_stop.data_ptr<int>(),
        max_new ? max_new->data_ptr<int>() : nullptr,
        (long long)pad_id, B, (int)pre_ids.size(1), cur_stream());
}

void block_step(torch::Tensor block_table, torch::Tensor seq_lens,
                torch::Tensor stop_flags, torch::Tensor active,
                torch::Tensor free_list, torch::Tensor free_count,
                torch::Tensor is_block_step, int64_t block_size) {
    int B = block_table.size(0);
    launch_block_step(block_table.data_ptr<int>(), seq_lens.data_ptr<int>(),
                      stop_flags.data_ptr<signed char>(), active.data_ptr<signed char>(),
                      free_list.data_ptr<int>(), free_count.data_ptr<int>(),
                      is_block_step.data_ptr<signed char>(), B, (int)block_size,
                      (int)block_table.size(1), cur_stream());
}

torch::Tensor mfma32_probe(torch::Tensor A, torch::Tensor B) {
    CHECK_GPU(A); CHECK_BF16(A);
    auto C = torch::empty({32, 32}, A.options().dtype(torch::kFloat32));
    launch_mfma32_probe(A.contiguous().data_ptr(), B.contiguous().data_ptr(),
                        C.data_ptr<float>(), cur_stream());
    return C;
}

std::vector<torch::Tensor> permlane_probe() {
    auto opt = torch::TensorOptions().dtype(torch::kInt32).device(torch::kCUDA);
    auto o0 = torch::empty({64}, opt);
    auto o1 = torch::empty({64}, opt);
    launch_permlane_probe(o0.data_ptr<int>(), o1.data_ptr<int>(), cur_stream());
    return {o0, o1};
}

std::vector<torch::Tensor> flash_attn_fwd_ex(torch::Tensor q, torch::Tensor k, torch::Tensor v,
                                             bool causal, int64_t variant) {
    int B = q.size(0), Sq = q.size(1), Hq = q.size(2), D = q.size(3);
    int Skv = k.size(1), Hk = k.size(2);
    auto o = torch::empty_like(q);
    auto lse = torch::empty({B, Hq, Sq}, q.options().dtype(torch::kFloat32));
    float scale = 1.0f / std::sqrt((float)D);
    launch_flash_fwd_variant(q.data_ptr(), k.data_ptr(), v.data_ptr(), o.data_ptr(),
                             lse.data_ptr<float>(), B, Sq, Skv, Hq, Hk, D, scale,
                             causal, (int)variant, cur_stream());
    return {o, lse};
}

std::vector<torch::Tensor> flash_attn_fwd(torch::Tensor q, torch::Tensor k, torch::Tensor v,
                                          bool causal) {
    CHECK_GPU(q); CHECK_CONTIG(q); CHECK_BF16(q);
    CHECK_CONTIG(k); CHECK_CONTIG(v);
    TORCH_CHECK(q.dim() == 4, "q must be [B,S,H,D]");
    int B = q.size(0), Sq = q.size(1), Hq = q.size(2), D = q.size(3);
    int Skv = k.size(1), Hk = k.size(2);
    TORCH_CHECK(D == 128 || D == 64 || D == 32, "head dim must be 32/64/128");
    TORCH_CHECK(Hq % Hk == 0, "GQA requires Hq % Hk == 0");
    auto o = torch::empty_like(q);
    auto lse = torch::empty({B, Hq, Sq}, q.options().dtype(torch::kFloat32));
    float scale = 1.0f / std::sqrt((float)D);
    launch_flash_fwd(q.data_ptr(), k.data_ptr(), v.data_ptr(), o.data_ptr(),
                     lse.data_ptr<float>(), B, Sq, Skv, Hq, Hk, D, scale, causal,
                     cur_stream());
    return {o, lse};
}

std::vector<torch::Tensor> flash_attn_bwd_ex(torch::Tensor dout, torch::Tensor q,
                                             torch::Tensor k, torch::Tensor v,
                                             torch::Tensor o, torch::Tensor lse,
                                             bool causal, int64_t variant) {
    int B = q.size(0), Sq = q.size(1), Hq = q.size(2), D = q.size(3);
    int Skv = k.size(1), Hk = k.size(2);
    auto dq = torch::empty_like(q);
    auto dk_h = torch::empty({B, Skv, Hk, D}, k.options());
    auto dv_h = torch::empty({B, Skv, Hk, D}, v.options());
    auto delta = torch::empty({B, Hq, Sq}, q.options().dtype(torch::kFloat32));
    float scale = 1.0f / std::sqrt((float)D);
    launch_flash_bwd_variant(dout.data_ptr(), q.data_ptr(), k.data_ptr(), v.data_ptr(),
                             o.data_ptr(), lse.data_ptr<float>(), delta.data_ptr<float>(),
                             dq.data_ptr(), dk_h.data_ptr(), dv_h.data_ptr(),
                             B, Sq, Skv, Hq, Hk, D, scale, causal, (int)variant, cur_stream());
    return {dq, dk_h, dv_h};
}

std::vector<torch::Tensor> flash_attn_bwd(torch::Tensor dout, torch::Tensor q,
                                          torch::Tensor k, torch::Tensor v,
                                          torch::Tensor o, torch::Tensor lse,
                                          bool causal) {
    CHECK_GPU(dout); CHECK_CONTIG(dout); CHECK_BF16(dout);
    int B = q.size(0), Sq = q.size(1), Hq = q.size(2), D = q.size(3);
    int Skv = k.size(1), Hk = k.size(2);
    auto dq = torch::empty_like(q);
    auto dk_h = torch::empty({B, Skv, Hk, D}, k.options());
    auto dv_h = torch::empty({B, Skv, Hk, D}, v.options());
    auto delta = torch::empty({B, Hq, Sq}, q.options().dtype(torch::kFloat32));
    float scale = 1.0f / std::sqrt((float)D);
    launch_flash_bwd(dout.data_ptr(), q.data_ptr(), k.data_ptr(), v.data_ptr(),
                     o.data_ptr(), lse.data_ptr<float>(), delta.data_ptr<float>(),
                     dq.data_ptr(), dk_h.data_ptr(), dv_h.data_ptr(),
                     B, Sq, Skv, Hq, Hk, D, scale, causal, cur_stream());
    return {dq, dk_h, dv_h};
}

// ---------------------------------------------------------------------------
void fused_adamw(std::vector<torch::Tensor> params, std::vector<torch::Tensor> grads,
                 std::vector<torch::Tensor> exp_avgs, std::vector<torch::Tensor> exp_avg_sqs,
                 std::vector<torch::Tensor> masters,
                 double lr, double beta1, double beta2, double eps, double wd,
                 int64_t step) {
    TORCH_CHECK(!params.empty());
    bool has_master = !masters.empty();
    const long long CHUNK = 1 << 22;  // 4M elements per chunk
    std::vector<AdamWChunk> chunks;
    for (size_t i = 0; i < params.size(); i++) {
        auto& p = params[i];
        CHECK_GPU(p); CHECK_CONTIG(p);
        bool is_bf16 = p.scalar_type() == torch::kBFloat16;
        TORCH_CHECK(!is_bf16 || has_master, "bf16 params need master weights");
        long long n = p.numel();
        torch::Tensor g = grads[i];
        if (g.scalar_type() != p.scalar_type()) g = g.to(p.scalar_type());
        for (long long off = 0; off < n; off += CHUNK) {
            AdamWChunk c;
            c.param = p.data_ptr();
            c.grad = g.data_ptr();
            c.m = exp_avgs[i].data_ptr<float>();
            c.v = exp_avg_sqs[i].data_ptr<float>();
            c.master = has_master ? masters[i].data_ptr<float>() : nullptr;
            c.offset = off;
            c.n = std::min(CHUNK, n - off);
            c.is_bf16 = is_bf16 ? 1 : 0;
            chunks.push_back(c);
        }
    }
    auto meta = torch::from_blob(chunks.data(), {(long long)(chunks.size() * sizeof(AdamWChunk))},
                                 torch::kUInt8).clone();
    auto dev_meta = meta.to(params[0].device(), /*non_blocking=*/true);
    float bias1 = 1.0f - std::pow((float)beta1, (float)step);
    float bias2 = 1.0f - std::pow((float)beta2, (float)step);
    launch_adamw(reinterpret_cast<const AdamWChunk*>(dev_meta.data_ptr()),
                 (int)chunks.size(), (float)lr, (float)beta1, (float)beta2,
                 (float)eps, (float)wd, bias1, bias2, cur_stream());
}

std::vector<torch::Tensor> flashmask_attn_fwd(torch::Tensor q, torch::Tensor k, torch::Tensor v,
                                              torch::Tensor startend) {
    CHECK_GPU(q); CHECK_CONTIG(q); CHECK_BF16(q);
    int B = q.size(0), Sq = q.size(1), Hq = q.size(2), D = q.size(3);
    int Skv = k.size(1), Hk = k.size(2);
    TORCH_CHECK(startend.numel() == (long long)B * Skv, "startend must be [B, Skv]");
    auto se = startend.to(torch::kInt32).contiguous();
    auto o = torch::empty_like(q);
    auto lse = torch::empty({B, Hq, Sq}, q.options().dtype(torch::kFloat32));
    float scale = 1.0f / std::sqrt((float)D);
    launch_flash_fwd_mask(q.data_ptr(), k.data_ptr(), v.data_ptr(), o.data_ptr(),
                          lse.data_ptr<float>(), se.data_ptr<int>(),
                          B, Sq, Skv, Hq, Hk, D, scale, cur_stream());
    return {o, lse};
}

std::vector<torch::Tensor> flashmask_attn_bwd(torch::Tensor dout, torch::Tensor q,
                                              torch::Tensor k, torch::Tensor v,
                                              torch::Tensor o, torch::Tensor lse,
                                              torch::Tensor startend) {
    int B = q.size(0), Sq = q.size(1), Hq = q.size(2), D = q.size(3);
    int Skv = k.size(1), Hk = k.size(2);
    auto se = startend.to(torch::kInt32).contiguous();
    auto dq = torch::empty_like(q);
    auto dk_h = torch::empty({B, Skv, Hk, D}, k.options());
    auto dv_h = torch::empty({B, Skv, Hk, D}, v.options());
    auto delta = torch::empty({B, Hq, Sq}, q.options().dtype(torch::kFloat32));
    float scale = 1.0f / std::sqrt((float)D);
    launch_flash_bwd_mask(dout.data_ptr(), q.data_ptr(), k.data_ptr(), v.data_ptr(),
                          o.data_ptr(), lse.data_ptr<float>(), delta.data_ptr<float>(),
                          dq.data_ptr(), dk_h.data_ptr(), dv_h.data_ptr(),
                          se.data_ptr<int>(), B, Sq, Skv, Hq, Hk, D, scale, cur_stream());
    return {dq, dk_h, dv_h};
}

// ---------------------------------------------------------------------------
// paged-KV inference ops
// ---------------------------------------------------------------------------
torch::Tensor paged_decode_attn(torch::Tensor q, torch::Tensor k_cache, torch::Tensor v_cache,
                                torch::Tensor block_table, torch::Tensor seq_lens,
                                c10::optional<torch::Tensor> k_scale,
                                c10::optional<torch::Tensor> v_scale) {
    CHECK_GPU(q); CHECK_CONTIG(q); CHECK_BF16(q);
    TORCH_CHECK(q.dim() == 3, "q must be [B, Hq, D] (single decode token)");
    int B = q.size(0), Hq = q.size(1), D = q.size(2);
    int block_size = k_cache.size(1), Hk = k_cache.size(2);
    int max_blocks = block_table.size(1);
    int cache_mode = 0;
    if (k_cache.scalar_type() == torch::kChar) cache_mode = 1;
    else if (k_cache.scalar_type() == torch::kByte) cache_mode = 2;
    TORCH_CHECK(cache_mode == 0 || (k_scale.has_value() && v_scale.has_value()),
                "quantized KV cache needs k_scale/v_scale");
    TORCH_CHECK(cache_mode != 2 || (D == 128 && Hq / Hk <= 16),
                "int4 KV cache requires head dim 128 and GQA group <= 16");
    auto out = torch::empty_like(q);
    float scale = 1.0f / std::sqrt((float)D);
    int G = Hq / Hk;
    int nsplit = paged_decode_nsplit(B, Hk);
    torch::Tensor partials;
    float* pptr = nullptr;
    if (nsplit > 1) {
        partials = torch::empty({(long)B, (long)Hk, (long)nsplit, (long)G * (2 + D)},
                                q.options().dtype(torch::kFloat32));
        pptr = partials.data_ptr<float>();
    }
    launch_paged_decode_attn(q.data_ptr(), k_cache.data_ptr(), v_cache.data_ptr(),
                             cache_mode ? k_scale->data_ptr<float>() : nullptr,
                             cache_mode ? v_scale->data_ptr<float>() : nullptr,
                             block_table.data_ptr<int>(), seq_lens.data_ptr<int>(),
                             out.data_ptr(), pptr, nsplit, B, Hq, Hk, D, block_size,
                             max_blocks, scale, cache_mode, cur_stream());
    return out;
}

torch::Tensor rope_cache_append(torch::Tensor qkv, torch::Tensor k_cache, torch::Tensor v_cache,
                                torch::Tensor block_table, torch::Tensor seq_lens_before,
                                torch::Tensor cos_t, torch::Tensor sin_t,
                                int64_t Hq, int64_t Hk,
                                c10::optional<torch::Tensor> token_counts,
                                c10::optional<torch::Tensor> k_scale,
                                c10::optional<torch::Tensor> v_scale) {
    CHECK_GPU(qkv); CHECK_CONTIG(qkv); CHECK_BF16(qkv);
    TORCH_CHECK(qkv.dim() == 3, "qkv must be [B, T, (Hq+2Hk)*D]");
    int B = qkv.size(0), T = qkv.size(1);
    int cache_mode = 0;
    if (k_cache.scalar_type() == torch::kChar) cache_mode = 1;
    else if (k_cache.scalar_type() == torch::kByte) cache_mode = 2;
    int D = (int)k_cache.size(3) * (cache_mode == 2 ? 2 : 1);
    int block_size = k_cache.size(1);
    int max_blocks = block_table.size(1);
    // zeros (not empty): tokens beyond token_counts[b] are skipped by the
    // kernel and must compare deterministically
    auto q_out = torch::zeros({B, T, Hq, (long)D}, qkv.options());
    auto cf = cos_t.to(torch::kFloat32).contiguous();
    auto sf = sin_t.to(torch::kFloat32).contiguous();
    const int* tc = nullptr;
    if (token_counts.has_value()) tc = token_counts->data_ptr<int>();
    TORCH_CHECK(cache_mode == 0 || (k_scale.has_value() && v_scale.has_value()),
                "quantized KV cache needs k_scale/v_scale");
    TORCH_CHECK(cache_mode != 2 || D == 128, "int4 KV cache requires head dim 128");
    launch_rope_cache_append(qkv.data_ptr(), q_out.data_ptr(),
                             k_cache.data_ptr(), v_cache.data_ptr(),
                             cache_mode ? k_scale->data_ptr<float>() : nullptr,
                             cache_mode ? v_scale->data_ptr<float>() : nullptr,
                             block_table.data_ptr<int>(), seq_lens_before.data_ptr<int>(),
                             cf.data_ptr<float>(), sf.data_ptr<float>(),
                             B, T, (int)Hq, (int)Hk, D, block_size, max_blocks, tc,
                             cache_mode, cur_stream());
    return q_out;
}

torch::Tensor skinny_gemm(torch::Tensor x, torch::Tensor w, int64_t ksplit) {
    CHECK_GPU(x); CHECK_BF16(x); CHECK_BF16(w);
    auto x2 = x.contiguous().view({-1, x.size(-1)});
    int M = x2.size(0), K = x2.size(1), N = w.size(0);
    TORCH_CHECK(M <= 64, "skinny_gemm is for decode batches (M <= 64)");
    TORCH_CHECK(K % 8 == 0, "K must be a multiple of 8");
    int ks = skinny_gemm_ksplit(N, K, (int)ksplit);
    auto partial = torch::empty({(long)ks, (long)M, (long)N},
                                x.options().dtype(torch::kFloat32));
    auto sizes = x.sizes().vec();
    sizes.back() = N;
    auto y = torch::empty(sizes, x.options());
    launch_skinny_gemm(x2.data_ptr(), w.contiguous().data_ptr(),
                       partial.data_ptr<float>(), y.data_ptr(),
                       M, N, K, (int)ksplit, cur_stream());
    return y;
}

std::vector<torch::Tensor> fp8_rowwise_quant(torch::Tensor x) {
    CHECK_GPU(x); CHECK_BF16(x);
    auto x2 = x.contiguous().view({-1, x.size(-1)});
    long long R = x2.size(0);
    int K = x2.size(1);
    TORCH_CHECK(K % 8 == 0, "K must be a multiple of 8");
    auto y = torch::empty({R, (long)K}, x.options().dtype(torch::kFloat8_e4m3fn));
    auto scales = torch::empty({R}, x.options().dtype(torch::kFloat32));
    launch_fp8_rowwise_quant(x2.data_ptr(), y.data_ptr(), scales.data_ptr<float>(),
                             R, K, cur_stream());
    return {y, scales};
}

torch::Tensor wint8_gemv(torch::Tensor x, torch::Tensor wq, torch::Tensor scale) {
    CHECK_GPU(x); CHECK_BF16(x);
    auto x2 = x.contiguous().view({-1, x.size(-1)});
    int M = x2.size(0), K = x2.size(1), N = wq.size(0);
    TORCH_CHECK(M <= 16, "wint8_gemv is for decode shapes (M <= 16)");
    TORCH_CHECK(K % 128 == 0, "K must be a multiple of 128");
    auto sizes = x.sizes().vec();
    sizes.back() = N;
    auto y = torch::empty(sizes, x.options());
    launch_wint8_gemv(x2.data_ptr(), wq.data_ptr(), scale.data_ptr<float>(),
                      y.data_ptr(), M, N, K, cur_stream());
    return y;
}

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
    m.def("rms_norm_fwd", &rms_norm_fwd);
    m.def("rms_norm_bwd", &rms_norm_bwd);
    m.def("rope_fwd", &rope_fwd);
    m.def("swiglu_fwd", &swiglu_fwd);
    m.def("swiglu_bwd", &swiglu_bwd);
    m.def("cross_entropy_fwd", &cross_entropy_fwd);
    m.def("cross_entropy_bwd", &cross_entropy_bwd);
    m.def("mfma_probe", &mfma_probe);
    m.def("mfma32_probe", &mfma32_probe);
    m.def("permlane_probe", &permlane_probe);
    m.def("apply_repetition_penalty", &apply_repetition_penalty);
    m.def("topp_sample", &topp_sample,
          py::arg("logits"), py::arg("temperature"), py::arg("top_p"),
          py::arg("uniform"), py::arg("ban_eos") = py::none(),
          py::arg("cur_lens") = py::none(), py::arg("min_lens") = py::none());
    m.def("decode_update", &decode_update,
          py::arg("tokens"), py::arg("stop_flags"), py::arg("active"),
          py::arg("pre_ids"), py::arg("pre_lens"), py::arg("seq_lens"),
          py::arg("eos_ids"), py::arg("not_need_stop"),
          py::arg("max_new") = py::none(), py::arg("pad_id") = 0);
    m.def("block_step", &block_step);
    m.def("flash_attn_fwd", &flash_attn_fwd);
    m.def("flash_attn_fwd_ex", &flash_attn_fwd_ex);
    m.def("flash_attn_bwd", &flash_attn_bwd);
    m.def("flash_attn_bwd_ex", &flash_attn_bwd_ex);
    m.def("flashmask_attn_fwd", &flashmask_attn_fwd);
    m.def("flashmask_attn_bwd", &flashmask_attn_bwd);
    m.def("fused_adamw", &fused_adamw);
    m.def("paged_decode_attn", &paged_decode_attn,
          py::arg("q"), py::arg("k_cache"), py::arg("v_cache"),
          py::arg("block_table"), py::arg("seq_lens"),
          py::arg("k_scale") = c10::nullopt, py::arg("v_scale") = c10::nullopt);
    m.def("rope_cache_append", &rope_cache_append,
          py::arg("qkv"), py::arg("k_cache"), py::arg("v_cache"),
          py::arg("block_table"), py::arg("seq_lens_before"),
          py::arg("cos_t"), py::arg("sin_t"), py::arg("Hq"), py::arg("Hk"),
          py::arg("token_counts") = c10::nullopt,
          py::arg("k_scale") = c10::nullopt, py::arg("v_scale") = c10::nullopt);
    m.def("wint8_gemv", &wint8_gemv);
    m.def("fp8_rowwise_quant", &fp8_rowwise_quant);
    m.def("skinny_gemm", &skinny_gemm, py::arg("x"), py::arg("w"),
          py::arg("ksplit") = 8);
}
