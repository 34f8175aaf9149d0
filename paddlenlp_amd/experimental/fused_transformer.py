"""FusedMultiTransformer: the decode-optimized inference engine.

Reference behavior: paddlenlp/experimental/transformers/
fused_transformer_layers.py (FusedMultiTransformerBase :348 /
FusedBlockMultiTransformer :2192 — one module holding ALL layers' weights as
lists, per-layer micro-ops compute_layernorm_before_qkv / compute_qkv /
compute_attn / compute_ffn, TP all-reduce at :1145/:1168) and the per-model
InferenceModel wrappers (experimental/transformers/llama/modeling.py:1595).

MI355X design: decode steps run hand-written gfx950 kernels end-to-end —
rms_norm, fused rope+paged-cache append, GQA paged decode attention (one
K/V stream per kv-head group) — with hipBLASLt GEMMs for the projections.
Prefill reuses the training flash-attention kernel on the contiguous prompt
and back-fills the paged cache.  CPU fallbacks mirror the math for tests.
"""
from __future__ import annotations

import math
from dataclasses import dataclass

import torch
import torch.nn as nn
import torch.nn.functional as F

from .. import ops
from ..ops import reference


@dataclass
class FusedMultiTransformerConfig:
    hidden_size: int
    num_heads: int
    num_kv_heads: int
    intermediate_size: int
    num_layers: int
    vocab_size: int
    rms_norm_eps: float = 1e-5
    rope_theta: float = 10000.0
    block_size: int = 64
    max_seq_len: int = 8192
    dtype: torch.dtype = torch.bfloat16
    # explicit head_dim for TP shards (local num_heads != hidden/head_dim)
    head_dim_override: int = 0
    qkv_bias: bool = False          # qwen2-style attention bias
    moe_num_experts: int = 0        # >0: FFN is a routed MoE (mixtral)
    moe_top_k: int = 2

    @property
    def head_dim(self):
        return self.head_dim_override or self.hidden_size // self.num_heads


# ---------------------------------------------------------------------------
# CPU reference versions of the paged kernels (test oracle)
# ---------------------------------------------------------------------------
def paged_decode_attn_ref(q, k_cache, v_cache, block_table, seq_lens,
                          k_scale=None, v_scale=None):
    """q [B, Hq, D] -> out [B, Hq, D] (fp32 math; int8 caches dequant with
    the per-(token, head) scales)."""
    B, Hq, D = q.shape
    bs = k_cache.shape[1]
    Hk = k_cache.shape[2]
    out = torch.empty_like(q)

    def _unpack4(t):
        # packed uint8 nibbles (offset-binary q+8) -> float
        hi = (t >> 4).to(torch.int8) - 8
        lo = (t & 0xF).to(torch.int8) - 8
        return torch.stack([hi, lo], dim=-1).reshape(*t.shape[:-1], D).float()

    is_int4 = k_cache.dtype == torch.uint8
    for b in range(B):
        L = int(seq_lens[b])
        nb = (L + bs - 1) // bs
        blocks = block_table[b, :nb].long()
        if is_int4:
            k = _unpack4(k_cache[blocks]).reshape(-1, Hk, D)[:L]
            v = _unpack4(v_cache[blocks]).reshape(-1, Hk, D)[:L]
        else:
            k = k_cache[blocks].reshape(-1, Hk, D)[:L].float()  # [L, Hk, D]
            v = v_cache[blocks].reshape(-1, Hk, D)[:L].float()
        if k_scale is not None:
            ks = k_scale[blocks].reshape(-1, Hk)[:L].float()
            vs = v_scale[blocks].reshape(-1, Hk)[:L].float()
            k = k * ks[:, :, None]
            v = v * vs[:, :, None]
        rep = Hq // Hk
        k = k.repeat_interleave(rep, dim=1)
        v = v.repeat_interleave(rep, dim=1)
        qb = q[b].float()  # [Hq, D]
        s = torch.einsum("hd,lhd->hl", qb, k) / math.sqrt(D)
        p = s.softmax(-1)
        out[b] = torch.einsum("hl,lhd->hd", p, v).to(q.dtype)
    return out


def rope_cache_append_ref(qkv, k_cache, v_cache, block_table, seq_lens_before,
                          cos_t, sin_t, Hq, Hk, token_counts=None,
                          k_scale=None, v_scale=None):
    """CPU mirror of the fused kernel; returns roped q [B, T, Hq, D]."""
    B, T, _ = qkv.shape
    D = k_cache.shape[3] * (2 if k_cache.dtype == torch.uint8 else 1)
    bs = k_cache.shape[1]
    qkv = qkv.view(B, T, Hq + 2 * Hk, D)
    q = qkv[:, :, :Hq]
    k = qkv[:, :, Hq:Hq + Hk]
    v = qkv[:, :, Hq + Hk:]
    q_out = torch.zeros_like(q)
    for b in range(B):
        n_tok = int(token_counts[b]) if token_counts is not None else T
        start = int(seq_lens_before[b])
        for t in range(n_tok):
            pos = start + t
            cos = cos_t[pos].to(torch.float32)
            sin = sin_t[pos].to(torch.float32)
            qr, kr = reference.apply_rope(
                q[b, t][None, None].float(), k[b, t][None, None].float(),
                cos[None], sin[None])
            q_out[b, t] = qr[0, 0].to(q.dtype)
            blk = int(block_table[b, pos // bs])
            if k_scale is not None and k_cache.dtype == torch.uint8:
                # int4: per-(token, head) absmax/7, offset-binary nibbles
                kv32 = kr[0, 0].float()
                vv32 = v[b, t].float()
                ks = kv32.abs().amax(-1).clamp(min=1e-8) / 7.0
                vs = vv32.abs().amax(-1).clamp(min=1e-8) / 7.0

                def _pack4(x, sc):
                    qq = torch.round(x / sc[:, None]).clamp(-7, 7).to(torch.int8) + 8
                    u = qq.to(torch.uint8)
                    return (u[:, 0::2] << 4) | u[:, 1::2]

                k_cache[blk, pos % bs] = _pack4(kv32, ks)
                v_cache[blk, pos % bs] = _pack4(vv32, vs)
                k_scale[blk, pos % bs] = ks
                v_scale[blk, pos % bs] = vs
            elif k_scale is not None:  # int8 cache: per-(token, head) absmax
                kv32 = kr[0, 0].float()
                vv32 = v[b, t].float()
                ks = kv32.abs().amax(-1).clamp(min=1e-8) / 127.0
                vs = vv32.abs().amax(-1).clamp(min=1e-8) / 127.0
                k_cache[blk, pos % bs] = torch.round(
                    kv32 / ks[:, None]).clamp(-127, 127).to(torch.int8)
                v_cache[blk, pos % bs] = torch.round(
                    vv32 / vs[:, None]).clamp(-127, 127).to(torch.int8)
                k_scale[blk, pos % bs] = ks
                v_scale[blk, pos % bs] = vs
            else:
                k_cache[blk, pos % bs] = kr[0, 0].to(k_cache.dtype)
                v_cache[blk, pos % bs] = v[b, t].to(v_cache.dtype)
    return q_out


class FusedMultiTransformer(nn.Module):
    """All decoder layers in one module; weights as per-layer lists."""

    k_scales = None  # set by allocate_caches(cachekv_dtype="int8")
    v_scales = None
    tp_group = None  # set by from_llama(tp_degree > 1): row-parallel reduce

    def __init__(self, config: FusedMultiTransformerConfig):
        super().__init__()
        self.config = config
        c = config
        mk = lambda *shape: nn.Parameter(
            torch.empty(*shape, dtype=c.dtype), requires_grad=False)
        L = c.num_layers
        h, hd = c.hidden_size, c.head_dim
        qkv_out = (c.num_heads + 2 * c.num_kv_heads) * hd
        self.ln_scales = nn.ParameterList([mk(h) for _ in range(L)])
        self.qkv_weights = nn.ParameterList([mk(qkv_out, h) for _ in range(L)])
        self.out_proj_weights = nn.ParameterList([mk(h, c.num_heads * hd) for _ in range(L)])
        self.ffn_ln_scales = nn.ParameterList([mk(h) for _ in range(L)])
        if c.qkv_bias:
            self.qkv_biases = nn.ParameterList([mk(qkv_out) for _ in range(L)])
        if c.moe_num_experts > 0:
            E, I = c.moe_num_experts, c.intermediate_size
            self.moe_gates = nn.ParameterList([mk(E, h) for _ in range(L)])
            self.moe_w1 = nn.ParameterList([mk(E, h, I) for _ in range(L)])
            self.moe_w3 = nn.ParameterList([mk(E, h, I) for _ in range(L)])
            self.moe_w2 = nn.ParameterList([mk(E, I, h) for _ in range(L)])
            self.gate_up_weights = nn.ParameterList([])
            self.down_weights = nn.ParameterList([])
        else:
            self.gate_up_weights = nn.ParameterList([mk(2 * c.intermediate_size, h) for _ in range(L)])
            self.down_weights = nn.ParameterList([mk(h, c.intermediate_size) for _ in range(L)])
        self.embed_tokens = mk(c.vocab_size, h)
        self.final_norm = mk(h)
        self.lm_head = mk(c.vocab_size, h)

        cos, sin = ops.build_rope_cache(c.max_seq_len, hd, base=c.rope_theta)
        self.register_buffer("rope_cos", cos, persistent=False)
        self.register_buffer("rope_sin", sin, persistent=False)
        self.quant_algo = None  # None | "fp8" | "weight_only_int8"
        self._qw = {}

    # ------------------------------------------------------------------
    @classmethod
    @torch.no_grad()
    def from_llama(cls, model, block_size: int = 64, max_seq_len: int = 8192,
                   tp_degree: int = 1, tp_rank: int = 0, tp_group=None):
        """Import weights from a LlamaForCausalLM (fused qkv / gate_up).

        With tp_degree > 1 this builds the tensor-parallel SHARD for
        tp_rank: qkv/gate_up are head/channel column-sharded, o/down are
        row-sharded and their partial outputs all-reduce over tp_group
        (Megatron inference split; embeddings/head replicated)."""
        mc = model.config
        assert mc.num_attention_heads % tp_degree == 0
        assert mc.num_key_value_heads % tp_degree == 0
        assert mc.intermediate_size % tp_degree == 0
        Hq, Hk = mc.num_attention_heads, mc.num_key_value_heads
        I = mc.intermediate_size
        hd = mc.hidden_size // Hq
        cfg = FusedMultiTransformerConfig(
            hidden_size=mc.hidden_size, num_heads=Hq // tp_degree,
            num_kv_heads=Hk // tp_degree, intermediate_size=I // tp_degree,
            num_layers=mc.num_hidden_layers, vocab_size=mc.vocab_size,
            rms_norm_eps=mc.rms_norm_eps, rope_theta=mc.rope_theta,
            block_size=block_size, max_seq_len=max_seq_len,
            dtype=next(model.parameters()).dtype, head_dim_override=hd,
        )
        eng = cls(cfg)
        eng.tp_group = tp_group if tp_degree > 1 else None
        r = tp_rank

        def qsl(w):   # rows of this rank's q heads
            return w[r * (Hq // tp_degree) * hd:(r + 1) * (Hq // tp_degree) * hd]

        def kvsl(w):  # rows of this rank's kv heads
            return w[r * (Hk // tp_degree) * hd:(r + 1) * (Hk // tp_degree) * hd]

        base = model.llama
        for i, layer in enumerate(base.layers):
            eng.ln_scales[i].copy_(layer.input_layernorm.weight)
            a = layer.self_attn
            if mc.fuse_attention_qkv:
                w = a.qkv_proj.weight
                qw, kw, vw = w[:Hq * hd], w[Hq * hd:(Hq + Hk) * hd], w[(Hq + Hk) * hd:]
            else:
                qw, kw, vw = a.q_proj.weight, a.k_proj.weight, a.v_proj.weight
            eng.qkv_weights[i].copy_(torch.cat([qsl(qw), kvsl(kw), kvsl(vw)], dim=0))
            eng.out_proj_weights[i].copy_(
                a.o_proj.weight[:, r * (Hq // tp_degree) * hd:(r + 1) * (Hq // tp_degree) * hd])
            eng.ffn_ln_scales[i].copy_(layer.post_attention_layernorm.weight)
            m = layer.mlp
            if mc.fuse_attention_ffn:
                w = m.gate_up_fused_proj.weight
                gw, uw = w[:I], w[I:]
            else:
                gw, uw = m.gate_proj.weight, m.up_proj.weight
            Il = I // tp_degree
            eng.gate_up_weights[i].copy_(torch.cat(
                [gw[r * Il:(r + 1) * Il], uw[r * Il:(r + 1) * Il]], dim=0))
            eng.down_weights[i].copy_(m.down_proj.weight[:, r * Il:(r + 1) * Il])
        eng.embed_tokens.copy_(base.embed_tokens.weight)
        eng.final_norm.copy_(base.norm.weight)
        eng.lm_head.copy_(model.lm_head.weight)
        return eng

    @classmethod
    @torch.no_grad()
    def from_qwen2(cls, model, block_size: int = 64, max_seq_len: int = 8192):
        """Import a Qwen2ForCausalLM: llama layout + attention bias
        (reference experimental/transformers/qwen2/modeling.py)."""
        mc = model.config
        Hq, Hk = mc.num_attention_heads, mc.num_key_value_heads
        I = mc.intermediate_size
        hd = mc.hidden_size // Hq
        has_bias = getattr(mc, "attention_bias", True)
        cfg = FusedMultiTransformerConfig(
            hidden_size=mc.hidden_size, num_heads=Hq, num_kv_heads=Hk,
            intermediate_size=I, num_layers=mc.num_hidden_layers,
            vocab_size=mc.vocab_size, rms_norm_eps=mc.rms_norm_eps,
            rope_theta=mc.rope_theta, block_size=block_size,
            max_seq_len=max_seq_len, dtype=next(model.parameters()).dtype,
            head_dim_override=hd, qkv_bias=has_bias,
        )
        eng = cls(cfg)
        base = model.qwen2
        for i, layer in enumerate(base.layers):
            eng.ln_scales[i].copy_(layer.input_layernorm.weight)
            a = layer.self_attn
            if hasattr(a, "qkv_proj"):
                w = a.qkv_proj.weight
                qw, kw, vw = w[:Hq * hd], w[Hq * hd:(Hq + Hk) * hd], w[(Hq + Hk) * hd:]
                if has_bias:
                    b = a.qkv_proj.bias
                    eng.qkv_biases[i].copy_(b)
            else:
                qw, kw, vw = a.q_proj.weight, a.k_proj.weight, a.v_proj.weight
                if has_bias:
                    eng.qkv_biases[i].copy_(torch.cat(
                        [a.q_proj.bias, a.k_proj.bias, a.v_proj.bias]))
            eng.qkv_weights[i].copy_(torch.cat([qw, kw, vw], dim=0))
            eng.out_proj_weights[i].copy_(a.o_proj.weight)
            eng.ffn_ln_scales[i].copy_(layer.post_attention_layernorm.weight)
            m = layer.mlp
            if hasattr(m, "gate_up_fused_proj"):
                eng.gate_up_weights[i].copy_(m.gate_up_fused_proj.weight)
            else:
                eng.gate_up_weights[i].copy_(torch.cat(
                    [m.gate_proj.weight, m.up_proj.weight], dim=0))
            eng.down_weights[i].copy_(m.down_proj.weight)
        eng.embed_tokens.copy_(base.embed_tokens.weight)
        eng.final_norm.copy_(base.norm.weight)
        eng.lm_head.copy_(model.lm_head.weight)
        return eng

    @classmethod
    @torch.no_grad()
    def from_mixtral(cls, model, block_size: int = 64, max_seq_len: int = 8192):
        """Import a MixtralForCausalLM: llama attention + routed-MoE FFN
        (reference experimental/transformers/mixtral/modeling.py)."""
        mc = model.config
        Hq, Hk = mc.num_attention_heads, mc.num_key_value_heads
        hd = mc.hidden_size // Hq
        cfg = FusedMultiTransformerConfig(
            hidden_size=mc.hidden_size, num_heads=Hq, num_kv_heads=Hk,
            intermediate_size=mc.intermediate_size,
            num_layers=mc.num_hidden_layers, vocab_size=mc.vocab_size,
            rms_norm_eps=mc.rms_norm_eps, rope_theta=mc.rope_theta,
            block_size=block_size, max_seq_len=max_seq_len,
            dtype=next(model.parameters()).dtype, head_dim_override=hd,
            moe_num_experts=mc.num_local_experts,
            moe_top_k=mc.num_experts_per_tok,
        )
        eng = cls(cfg)
        base = model.mixtral
        for i, layer in enumerate(base.layers):
            eng.ln_scales[i].copy_(layer.input_layernorm.weight)
            a = layer.self_attn
            if hasattr(a, "qkv_proj"):
                eng.qkv_weights[i].copy_(a.qkv_proj.weight)
            else:
                eng.qkv_weights[i].copy_(torch.cat(
                    [a.q_proj.weight, a.k_proj.weight, a.v_proj.weight], dim=0))
            eng.out_proj_weights[i].copy_(a.o_proj.weight)
            eng.ffn_ln_scales[i].copy_(layer.post_attention_layernorm.weight)
            moe = layer.block_sparse_moe
            eng.moe_gates[i].copy_(moe.gate.weight)
            eng.moe_w1[i].copy_(moe.experts.w1)
            eng.moe_w3[i].copy_(moe.experts.w3)
            eng.moe_w2[i].copy_(moe.experts.w2)
        eng.embed_tokens.copy_(base.embed_tokens.weight)
        eng.final_norm.copy_(base.norm.weight)
        eng.lm_head.copy_(model.lm_head.weight)
        return eng

    @classmethod
    def from_model(cls, model, **kw):
        """Generic importer: dispatch on the model family."""
        name = type(model).__name__
        if "Qwen2" in name:
            return cls.from_qwen2(model, **kw)
        if "Mixtral" in name:
            return cls.from_mixtral(model, **kw)
        return cls.from_llama(model, **kw)

    def _tp_reduce(self, t):
        if self.tp_group is not None:
            import torch.distributed as dist

            dist.all_reduce(t, group=self.tp_group)
        return t

    @torch.no_grad()
    def quantize(self, algo: str = "fp8", names=None):
        """Weight-only quantization of the projection weights (fp8 e4m3fn via
        the gfx950 fp8 MFMA path, or per-channel int8).  Norm scales and
        embeddings stay in the compute dtype.

        `names` restricts quantization, e.g. the bandwidth-bound subset
        ("gate_up_weights", "down_weights", "lm_head") — at decode batch
        sizes the skinny qkv/o GEMMs are latency-bound and fp8 does not
        pay there (see profiles/r02_decode_throughput.md)."""
        from ..quantization import quantize_fp8, quantize_int8

        qfn = quantize_fp8 if algo == "fp8" else quantize_int8
        self._qw = {}
        all_names = ("qkv_weights", "out_proj_weights", "gate_up_weights",
                     "down_weights")
        names = tuple(names) if names is not None else all_names
        for name in names:
            if name == "lm_head":
                q, sc = qfn(self.lm_head.data)
                self._qw["lm_head"] = [(q, sc)]
                self.lm_head.data = self.lm_head.data.new_zeros(1)
                continue
            plist = getattr(self, name)
            qs = []
            for w in plist:
                q, sc = qfn(w.data)
                qs.append((q, sc))
                w.data = w.data.new_zeros(1)  # free the bf16 copy
            self._qw[name] = qs
        self.quant_algo = algo
        return self

    def _mm(self, x, name: str, i: int):
        if self.quant_algo is None or name not in self._qw:
            return x @ getattr(self, name)[i].t()
        from ..quantization import weight_only_linear

        q, sc = self._qw[name][i]
        algo = self.quant_algo if self.quant_algo == "fp8" else "weight_only_int8"
        return weight_only_linear(x, q, sc, None, algo)

    def _moe_ffn(self, h, i):
        """Routed MoE FFN (mixtral): top-k router + capacity-padded batched
        GEMMs over the stacked expert weights (same grouped-GEMM scheme as
        parallel.expert_parallel.GroupedExperts; reference fused_moe,
        fused_transformer_layers.py:951-1008)."""
        c = self.config
        shp = h.shape
        x = h.reshape(-1, shp[-1])
        T = x.shape[0]
        E, K = c.moe_num_experts, c.moe_top_k
        router = x @ self.moe_gates[i].t()
        probs = router.float().softmax(-1)
        topw, tope = probs.topk(K, dim=-1)
        topw = (topw / topw.sum(-1, keepdim=True)).to(x.dtype)
        flat_x = x.repeat_interleave(K, dim=0)
        flat_e = tope.reshape(-1)
        sort_idx = torch.argsort(flat_e, stable=True)
        sorted_x = flat_x[sort_idx]
        counts = torch.bincount(flat_e, minlength=E)
        C = int(counts.max().item())
        out_sorted = torch.zeros_like(sorted_x)
        if C > 0:
            dev = x.device
            offs = torch.cumsum(counts, 0) - counts
            tok_e = torch.repeat_interleave(torch.arange(E, device=dev), counts)
            pos = torch.arange(sorted_x.shape[0], device=dev) - offs[tok_e]
            idx = tok_e * C + pos
            xp = sorted_x.new_zeros(E * C, shp[-1]).index_copy(0, idx, sorted_x)
            xp = xp.view(E, C, shp[-1])
            gu = torch.cat([torch.bmm(xp, self.moe_w1[i]),
                            torch.bmm(xp, self.moe_w3[i])], dim=-1)
            act = ops.swiglu(gu) if gu.is_cuda else reference.swiglu(gu)
            yp = torch.bmm(act, self.moe_w2[i])
            out_sorted = yp.reshape(E * C, shp[-1]).index_select(0, idx)
        out_flat = torch.empty_like(out_sorted)
        out_flat[sort_idx] = out_sorted
        out = (out_flat.reshape(T, K, shp[-1]) * topw[..., None]).sum(1)
        return out.reshape(shp)

    def allocate_caches(self, num_blocks: int, device, cachekv_dtype: str = "bf16"):
        """cachekv_dtype="int8" halves KV memory: caches store int8 with one
        fp32 absmax scale per cached (token, kv-head) vector; the decode and
        append kernels (de)quantize in-register (reference cachekv int8,
        write_int8_cache_kv / append_attention_c8)."""
        c = self.config
        shape = (num_blocks, c.block_size, c.num_kv_heads, c.head_dim)
        self.cachekv_dtype = cachekv_dtype
        if cachekv_dtype == "int4":
            # packed nibbles: 4x capacity vs bf16 (reference
            # append_attention_c4_impl); per-(token, head) absmax scale
            assert c.head_dim % 2 == 0
            p4 = (num_blocks, c.block_size, c.num_kv_heads, c.head_dim // 2)
            self.k_caches = [torch.zeros(p4, dtype=torch.uint8, device=device)
                             for _ in range(c.num_layers)]
            self.v_caches = [torch.zeros(p4, dtype=torch.uint8, device=device)
                             for _ in range(c.num_layers)]
            sshape = (num_blocks, c.block_size, c.num_kv_heads)
            self.k_scales = [torch.zeros(sshape, dtype=torch.float32, device=device)
                             for _ in range(c.num_layers)]
            self.v_scales = [torch.zeros(sshape, dtype=torch.float32, device=device)
                             for _ in range(c.num_layers)]
        elif cachekv_dtype == "int8":
            self.k_caches = [torch.zeros(shape, dtype=torch.int8, device=device)
                             for _ in range(c.num_layers)]
            self.v_caches = [torch.zeros(shape, dtype=torch.int8, device=device)
                             for _ in range(c.num_layers)]
            sshape = (num_blocks, c.block_size, c.num_kv_heads)
            self.k_scales = [torch.zeros(sshape, dtype=torch.float32, device=device)
                             for _ in range(c.num_layers)]
            self.v_scales = [torch.zeros(sshape, dtype=torch.float32, device=device)
                             for _ in range(c.num_layers)]
        else:
            self.k_caches = [torch.zeros(shape, dtype=c.dtype, device=device)
                             for _ in range(c.num_layers)]
            self.v_caches = [torch.zeros(shape, dtype=c.dtype, device=device)
                             for _ in range(c.num_layers)]
            self.k_scales = self.v_scales = None
        return self.k_caches, self.v_caches

    # ------------------------------------------------------------------
    def _rms(self, x, w):
        if x.is_cuda:
            return ops.rms_norm(x, w, self.config.rms_norm_eps)
        return reference.rms_norm(x, w, self.config.rms_norm_eps)

    def _rope_append(self, i, qkv, block_table, lens_before, token_counts=None):
        c = self.config
        if qkv.is_cuda:
            from ..ops.functional import _load_extension

            C = _load_extension()
            return C.rope_cache_append(
                qkv, self.k_caches[i], self.v_caches[i], block_table,
                lens_before, self.rope_cos, self.rope_sin,
                c.num_heads, c.num_kv_heads, token_counts,
                self.k_scales[i] if self.k_scales is not None else None,
                self.v_scales[i] if self.v_scales is not None else None)
        return rope_cache_append_ref(
            qkv, self.k_caches[i], self.v_caches[i], block_table, lens_before,
            self.rope_cos, self.rope_sin, c.num_heads, c.num_kv_heads, token_counts,
            self.k_scales[i] if self.k_scales is not None else None,
            self.v_scales[i] if self.v_scales is not None else None)

    def _paged_attn(self, i, q, block_table, seq_lens):
        if q.is_cuda:
            from ..ops.functional import _load_extension

            C = _load_extension()
            return C.paged_decode_attn(
                q, self.k_caches[i], self.v_caches[i], block_table, seq_lens,
                self.k_scales[i] if self.k_scales is not None else None,
                self.v_scales[i] if self.v_scales is not None else None)
        return paged_decode_attn_ref(
            q, self.k_caches[i], self.v_caches[i], block_table, seq_lens,
            self.k_scales[i] if self.k_scales is not None else None,
            self.v_scales[i] if self.v_scales is not None else None)

    # ------------------------------------------------------------------
    @torch.no_grad()
    def decode_step(self, input_ids, block_table, seq_lens_before) -> torch.Tensor:
        """One token per sequence.  input_ids [B, 1]; returns logits [B, V].

        seq_lens_before[b] = tokens already in the cache (the new token's
        position); the caller must have extended the block table first."""
        c = self.config
        B = input_ids.shape[0]
        x = F.embedding(input_ids, self.embed_tokens)  # [B, 1, H]
        seq_lens_after = seq_lens_before + 1
        for i in range(c.num_layers):
            h = self._rms(x, self.ln_scales[i])
            qkv = self._mm(h, "qkv_weights", i)          # [B, 1, qkv_out]
            if c.qkv_bias:
                qkv = qkv + self.qkv_biases[i]
            q = self._rope_append(i, qkv, block_table, seq_lens_before)
            attn = self._paged_attn(i, q[:, 0], block_table, seq_lens_after)
            x = x + self._tp_reduce(self._mm(attn.reshape(B, 1, -1), "out_proj_weights", i))
            h = self._rms(x, self.ffn_ln_scales[i])
            if c.moe_num_experts > 0:
                x = x + self._tp_reduce(self._moe_ffn(h, i))
            else:
                gu = self._mm(h, "gate_up_weights", i)
                act = ops.swiglu(gu) if gu.is_cuda else reference.swiglu(gu)
                x = x + self._tp_reduce(self._mm(act, "down_weights", i))
        x = self._rms(x, self.final_norm)
        if "lm_head" in self._qw:
            from ..quantization import weight_only_linear

            q, sc = self._qw["lm_head"][0]
            logits = weight_only_linear(x[:, 0], q, sc, None, self.quant_algo)
        else:
            logits = x[:, 0] @ self.lm_head.t()
        return logits.float()

    @torch.no_grad()
    def prefill(self, input_ids, block_table, prompt_lens) -> torch.Tensor:
        """Prefill a batch of prompts (right-padded to a common T).
        input_ids [B, T]; prompt_lens [B]; returns last-token logits [B, V].

        Uses the training flash kernel on the contiguous prompt; the fused
        rope+append kernel back-fills the paged cache in the same pass."""
        c = self.config
        B, T = input_ids.shape
        x = F.embedding(input_ids, self.embed_tokens)
        zeros = torch.zeros_like(prompt_lens)
        for i in range(c.num_layers):
            h = self._rms(x, self.ln_scales[i])
            qkv = self._mm(h, "qkv_weights", i)          # [B, T, *]
            if c.qkv_bias:
                qkv = qkv + self.qkv_biases[i]
            q = self._rope_append(i, qkv, block_table, zeros, token_counts=prompt_lens)
            # contiguous roped K + raw V for the flash kernel
            qkv_v = qkv.view(B, T, c.num_heads + 2 * c.num_kv_heads, c.head_dim)
            k_raw = qkv_v[:, :, c.num_heads:c.num_heads + c.num_kv_heads].contiguous()
            v = qkv_v[:, :, c.num_heads + c.num_kv_heads:].contiguous()
            if x.is_cuda:
                _, k_roped = ops.fused_rope(k_raw, k_raw, self.rope_cos[:T], self.rope_sin[:T])
            else:
                _, k_roped = reference.apply_rope(k_raw, k_raw, self.rope_cos[:T], self.rope_sin[:T])
            if x.is_cuda:
                attn = ops.flash_attention(q, k_roped, v, causal=True)
            else:
                attn = reference.flash_attention(q, k_roped, v, causal=True)
            x = x + self._tp_reduce(self._mm(attn.reshape(B, T, -1), "out_proj_weights", i))
            h = self._rms(x, self.ffn_ln_scales[i])
            if c.moe_num_experts > 0:
                x = x + self._tp_reduce(self._moe_ffn(h, i))
            else:
                gu = self._mm(h, "gate_up_weights", i)
                act = ops.swiglu(gu) if gu.is_cuda else reference.swiglu(gu)
                x = x + self._tp_reduce(self._mm(act, "down_weights", i))
        x = self._rms(x, self.final_norm)
        idx = (prompt_lens.long() - 1).clamp(min=0)
        last = x[torch.arange(B, device=x.device), idx]
        if "lm_head" in self._qw:
            from ..quantization import weight_only_linear

            q, sc = self._qw["lm_head"][0]
            return weight_only_linear(last, q, sc, None, self.quant_algo).float()
        return (last @ self.lm_head.t()).float()


    @torch.no_grad()
    def mixed_step(self, decode_ids, decode_bt, decode_lens,
                   prefill_ids=None, prefill_bt=None, prefill_lens=None):
        """One engine call serving a mixed batch: the running decode batch
        plus a freshly admitted prompt batch.  The reference runs encoder
        and decoder work on dual streams with event hand-off
        (csrc/gpu/append_attention.cu:120-160); here the prefill runs on a
        side HIP stream overlapping the decode pass — weights are
        read-only and the cache blocks are disjoint, so no ordering is
        needed beyond the join.  Continuous batching therefore admits new
        prompts without stalling the decode batch.

        Returns (decode_logits, prefill_logits_or_None)."""
        if prefill_ids is None:
            return self.decode_step(decode_ids, decode_bt, decode_lens), None
        if not decode_ids.is_cuda:
            d = self.decode_step(decode_ids, decode_bt, decode_lens)
            p = self.prefill(prefill_ids, prefill_bt, prefill_lens)
            return d, p
        if not hasattr(self, "_side_stream") or self._side_stream is None:
            self._side_stream = torch.cuda.Stream()
        cur = torch.cuda.current_stream()
        self._side_stream.wait_stream(cur)
        with torch.cuda.stream(self._side_stream):
            p_logits = self.prefill(prefill_ids, prefill_bt, prefill_lens)
        d_logits = self.decode_step(decode_ids, decode_bt, decode_lens)
        cur.wait_stream(self._side_stream)
        return d_logits, p_logits


class GraphDecodeRunner:
    """hipGraph-captured decode_step.

    Captures one graph per (batch_size, block_table_width) with static
    input/output buffers; replay swaps tensor contents only.  Seq lens and
    block tables are DEVICE tensors read by the kernels, so their values can
    change between replays — only shapes are baked in.  Falls back to eager
    decode off-GPU.  Biggest win on small models / small batches where the
    ~30-40 launches per step dominate; measured neutral at 8B batch 64
    (tools/bench_infer.py --graph).
    """

    def __init__(self, engine: "FusedMultiTransformer"):
        self.engine = engine
        self._graphs = {}

    def __call__(self, input_ids, block_table, seq_lens_before):
        if not input_ids.is_cuda:
            return self.engine.decode_step(input_ids, block_table, seq_lens_before)
        key = (input_ids.shape[0], block_table.shape[1])
        entry = self._graphs.get(key)
        if entry is None:
            static = {
                "tok": input_ids.clone(),
                "bt": block_table.clone(),
                "lens": seq_lens_before.clone(),
            }
            # one eager pass to warm allocator state before capture
            self.engine.decode_step(static["tok"], static["bt"], static["lens"])
            torch.cuda.synchronize()
            graph = torch.cuda.CUDAGraph()
            with torch.cuda.graph(graph):
                static["logits"] = self.engine.decode_step(
                    static["tok"], static["bt"], static["lens"])
            entry = self._graphs[key] = (graph, static)
        graph, static = entry
        static["tok"].copy_(input_ids)
        static["bt"].copy_(block_table)
        static["lens"].copy_(seq_lens_before)
        graph.replay()
        return static["logits"]
