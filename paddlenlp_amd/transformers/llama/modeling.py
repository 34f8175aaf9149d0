"""Llama model family — MI355X-native implementation.

Behavior parity with paddlenlp/transformers/llama/modeling.py (GQA at
:690-705, RoPE variants :442-556, decoder layer :1138, LlamaModel :1588,
LlamaForCausalLM :2013, LlamaPretrainingCriterion :1799), re-designed for
PyTorch-ROCm: [B, S, H, D] tensor layout end-to-end (no transposes around
attention), fused gfx950 HIP ops behind the paddlenlp_amd.ops seam
(flash_attention, rms_norm, fused_rope, swiglu), and Megatron-style TP
layers from paddlenlp_amd.parallel when tensor_parallel_degree > 1.
"""
from __future__ import annotations

import math
from functools import partial
from typing import Optional, Tuple

import torch
import torch.nn as nn
from torch.utils.checkpoint import checkpoint

from ... import ops
from ...parallel.tensor_parallel import ColumnParallelLinear, ParallelCrossEntropy, RowParallelLinear, VocabParallelEmbedding
from ...parallel.topology import get_topology
from ...generation import GenerationConfig, GenerationMixin
from ..model_utils import PretrainedModel
from .configuration import LlamaConfig

__all__ = [
    "LlamaRMSNorm",
    "LlamaAttention",
    "LlamaMLP",
    "LlamaDecoderLayer",
    "LlamaModel",
    "LlamaForCausalLM",
    "LlamaPretrainingCriterion",
]


class LlamaRMSNorm(nn.Module):
    def __init__(self, config: LlamaConfig, hidden_size: Optional[int] = None):
        super().__init__()
        hidden_size = hidden_size or config.hidden_size
        self.weight = nn.Parameter(torch.ones(hidden_size))
        self.eps = config.rms_norm_eps
        if config.sequence_parallel and config.tensor_parallel_degree > 1:
            # sees seq-sharded activations: grad needs an mp all-reduce
            # (reference register_sequence_parallel_allreduce_hooks)
            self.weight.sequence_parallel = True

    def forward(self, x):
        out = ops.rms_norm(x, self.weight, self.eps)
        # outlier-suppression shift (llm/utils/quant.py apply_shift): the
        # following linears see shifted activations and compensate in bias
        shift = getattr(self, "shift_bias", None)
        if shift is not None:
            out = out + shift
        return out


class LlamaRotaryEmbedding(nn.Module):
    """Precomputed cos/sin cache; supports linear / ntk / llama3 scaling
    (reference: llama/modeling.py:442-556)."""

    def __init__(self, config: LlamaConfig):
        super().__init__()
        self.config = config
        self.head_dim = config.head_dim
        self.base = config.rope_theta
        self.max_seq_len_cached = 0
        self.register_buffer("cos_cached", torch.empty(0), persistent=False)
        self.register_buffer("sin_cached", torch.empty(0), persistent=False)

    def _inv_freq(self, device):
        dim = self.head_dim
        base = self.base
        stype = self.config.rope_scaling_type
        factor = self.config.rope_scaling_factor
        if stype == "ntk":
            base = base * factor ** (dim / (dim - 2))
        inv_freq = 1.0 / (base ** (torch.arange(0, dim, 2, dtype=torch.float32, device=device) / dim))
        if stype == "llama3":
            # reference llama/modeling.py:520 Llama3 rope: frequency-dependent scaling
            low_freq_factor, high_freq_factor, orig_ctx = 1.0, 4.0, 8192
            wavelen = 2 * math.pi / inv_freq
            low_wl = orig_ctx / low_freq_factor
            high_wl = orig_ctx / high_freq_factor
            scaled = inv_freq / factor
            smooth = (orig_ctx / wavelen - low_freq_factor) / (high_freq_factor - low_freq_factor)
            mid = (1 - smooth) * scaled + smooth * inv_freq
            inv_freq = torch.where(wavelen > low_wl, scaled, torch.where(wavelen < high_wl, inv_freq, mid))
        return inv_freq

    def _build(self, seq_len, device):
        inv_freq = self._inv_freq(device)
        t = torch.arange(seq_len, dtype=torch.float32, device=device)
        if self.config.rope_scaling_type == "linear":
            t = t / self.config.rope_scaling_factor
        freqs = torch.outer(t, inv_freq)
        emb = torch.cat((freqs, freqs), dim=-1)
        self.cos_cached = emb.cos()
        self.sin_cached = emb.sin()
        self.max_seq_len_cached = seq_len

    def forward(self, seq_len: int, device, position_offset: int = 0):
        total = seq_len + position_offset
        if total > self.max_seq_len_cached or self.cos_cached.device != torch.device(device):
            self._build(max(total, self.config.max_position_embeddings), device)
        return (
            self.cos_cached[position_offset:position_offset + seq_len],
            self.sin_cached[position_offset:position_offset + seq_len],
        )


class LlamaAttention(nn.Module):
    """GQA attention in [B, S, H, D] layout (no transposes: the CDNA4 flash
    kernel consumes bshd directly)."""

    def __init__(self, config: LlamaConfig, layer_idx: int = 0):
        super().__init__()
        self.config = config
        self.layer_idx = layer_idx
        self.hidden_size = config.hidden_size
        self.num_heads = config.num_attention_heads
        self.num_kv_heads = config.num_key_value_heads
        self.head_dim = config.head_dim

        tp = config.tensor_parallel_degree
        if tp > 1:
            assert self.num_heads % tp == 0 and self.num_kv_heads % tp == 0
            self.num_heads_local = self.num_heads // tp
            self.num_kv_heads_local = self.num_kv_heads // tp
        else:
            self.num_heads_local = self.num_heads
            self.num_kv_heads_local = self.num_kv_heads

        q_out = self.num_heads * self.head_dim
        kv_out = self.num_kv_heads * self.head_dim
        Column, Row = _linear_classes(config)

        if config.fuse_attention_qkv:
            self.qkv_proj = Column(self.hidden_size, q_out + 2 * kv_out, bias=False)
        else:
            self.q_proj = Column(self.hidden_size, q_out, bias=False)
            self.k_proj = Column(self.hidden_size, kv_out, bias=False)
            self.v_proj = Column(self.hidden_size, kv_out, bias=False)
        self.o_proj = Row(q_out, self.hidden_size, bias=False)
        self.rotary_emb = LlamaRotaryEmbedding(config)
        # Ulysses reshard around the attention core (reference
        # segment_parallel_utils.py ReshardLayer, llama/modeling.py:811-815)
        self.sep_degree = max(config.sep_parallel_degree, 1)
        self.cp_degree = max(config.context_parallel_degree, 1)
        if self.sep_degree > 1:
            from ...parallel.segment_parallel import ReshardLayer

            assert self.num_heads_local % self.sep_degree == 0
            assert self.num_kv_heads_local % self.sep_degree == 0
            self.reshard = ReshardLayer()

    def forward(
        self,
        hidden_states: torch.Tensor,  # [B, S, H]
        attn_mask: Optional[torch.Tensor] = None,
        startend_row_indices: Optional[torch.Tensor] = None,
        past_key_value: Optional[Tuple[torch.Tensor, torch.Tensor]] = None,
        use_cache: bool = False,
        position_offset: int = 0,
    ):
        B = hidden_states.shape[0]
        hl, kl, d = self.num_heads_local, self.num_kv_heads_local, self.head_dim

        if self.config.fuse_attention_qkv:
            qkv = self.qkv_proj(hidden_states)
            q, k, v = qkv.split([hl * d, kl * d, kl * d], dim=-1)
        else:
            q = self.q_proj(hidden_states)
            k = self.k_proj(hidden_states)
            v = self.v_proj(hidden_states)
        # with sequence parallel the column linear all-gathered the seq dim,
        # so infer S from the projection output, not the input
        S = q.shape[1]
        q = q.view(B, S, hl, d)
        k = k.view(B, S, kl, d)
        v = v.view(B, S, kl, d)

        # sep/cp: this rank holds a seq chunk; rope positions are offset
        balanced_cp = (self.cp_degree > 1 and
                       getattr(self.config, "context_parallel_balanced", False))
        if balanced_cp:
            # zigzag shard: local seq = [chunk r | chunk 2w-1-r] of 2w chunks
            sep_rank = get_topology().get_rank_in("sep")
            w = self.cp_degree
            C = S // 2
            dev = hidden_states.device
            cos1, sin1 = self.rotary_emb(C, dev, position_offset + sep_rank * C)
            cos2, sin2 = self.rotary_emb(
                C, dev, position_offset + (2 * w - 1 - sep_rank) * C)
            cos = torch.cat([cos1, cos2], dim=0)
            sin = torch.cat([sin1, sin2], dim=0)
        else:
            if self.sep_degree > 1 or self.cp_degree > 1:
                sep_rank = get_topology().get_rank_in("sep")
                position_offset = position_offset + sep_rank * S
            cos, sin = self.rotary_emb(S, hidden_states.device, position_offset)
        if self.config.use_fused_rope:
            q, k = ops.fused_rope(q, k, cos, sin)
        else:
            q, k = ops.reference.apply_rope(q, k, cos, sin)

        if past_key_value is not None:
            k = torch.cat([past_key_value[0], k], dim=1)
            v = torch.cat([past_key_value[1], v], dim=1)
        present = (k, v) if use_cache else None

        if self.cp_degree > 1:
            from ...parallel.ring_attention import ring_flash_attention

            attn_out = ring_flash_attention(
                q, k, v, causal=True,
                balanced=getattr(self.config, "context_parallel_balanced", False))
        elif self.sep_degree > 1:
            # Ulysses: [B, S/sep, H, D] -> [B, S, H/sep, D] around the core
            q = self.reshard.seq_to_head(q)
            k = self.reshard.seq_to_head(k)
            v = self.reshard.seq_to_head(v)
            attn_out = ops.flash_attention(q, k, v, causal=True)
            attn_out = self.reshard.head_to_seq(attn_out)
        else:
            attn_out = ops.flash_attention(
                q, k, v, causal=True, attn_mask=attn_mask,
                startend_row_indices=startend_row_indices,
            )
        attn_out = attn_out.reshape(B, attn_out.shape[1], hl * d)
        out = self.o_proj(attn_out)
        if use_cache:
            return out, present
        return out


class _Linear(nn.Linear):
    """nn.Linear with the (in, out, bias=) ctor order of the parallel layers."""

    def __init__(self, in_features, out_features, bias=False, **kwargs):
        super().__init__(in_features, out_features, bias=bias)


def _linear_classes(config):
    """(Column, Row) linear classes for this parallel config.

    Reference: paddlenlp/transformers/linear_utils.py:33-38 aliasing."""
    if config.tensor_parallel_degree > 1:
        if config.sequence_parallel:
            from ...parallel.sequence_parallel import (
                ColumnSequenceParallelLinear,
                RowSequenceParallelLinear,
            )

            return ColumnSequenceParallelLinear, RowSequenceParallelLinear
        return ColumnParallelLinear, RowParallelLinear
    return _Linear, _Linear


class LlamaMLP(nn.Module):
    def __init__(self, config: LlamaConfig):
        super().__init__()
        Column, Row = _linear_classes(config)
        self.config = config
        h, i = config.hidden_size, config.intermediate_size
        if config.fuse_attention_ffn:
            self.gate_up_fused_proj = Column(h, 2 * i, bias=False)
        else:
            self.gate_proj = Column(h, i, bias=False)
            self.up_proj = Column(h, i, bias=False)
        self.down_proj = Row(i, h, bias=False)

    def forward(self, x):
        if self.config.fuse_attention_ffn:
            gate_up = self.gate_up_fused_proj(x)
            if self.config.tensor_parallel_degree > 1:
                # column split interleaves [gate|up] per rank — layout is
                # [gate_local, up_local] because we shard the fused weight
                # with the paired split in _get_tensor_parallel_mappings
                act = ops.swiglu(gate_up)
            else:
                act = ops.swiglu(gate_up)
        else:
            act = ops.swiglu(torch.cat([self.gate_proj(x), self.up_proj(x)], dim=-1))
        return self.down_proj(act)


class LlamaDecoderLayer(nn.Module):
    def __init__(self, config: LlamaConfig, layer_idx: int = 0):
        super().__init__()
        self.self_attn = LlamaAttention(config, layer_idx)
        self.mlp = LlamaMLP(config)
        self.input_layernorm = LlamaRMSNorm(config)
        self.post_attention_layernorm = LlamaRMSNorm(config)

    def forward(
        self,
        hidden_states,
        attn_mask=None,
        startend_row_indices=None,
        past_key_value=None,
        use_cache=False,
        position_offset=0,
    ):
        residual = hidden_states
        hidden_states = self.input_layernorm(hidden_states)
        attn_out = self.self_attn(
            hidden_states, attn_mask, startend_row_indices,
            past_key_value, use_cache, position_offset,
        )
        if use_cache:
            attn_out, present = attn_out
        hidden_states = residual + attn_out

        residual = hidden_states
        hidden_states = self.post_attention_layernorm(hidden_states)
        hidden_states = residual + self.mlp(hidden_states)
        if use_cache:
            return hidden_states, present
        return hidden_states


class LlamaPretrainedModel(PretrainedModel):
    config_class = LlamaConfig
    base_model_prefix = "llama"

    def _init_weights(self, module):
        std = self.config.initializer_range
        if isinstance(module, (nn.Linear, _Linear, ColumnParallelLinear, RowParallelLinear)):
            module.weight.data.normal_(mean=0.0, std=std)
            if getattr(module, "bias", None) is not None:
                module.bias.data.zero_()
        elif isinstance(module, (nn.Embedding, VocabParallelEmbedding)):
            module.weight.data.normal_(mean=0.0, std=std)

    @classmethod
    def _get_tensor_parallel_mappings(cls, config: LlamaConfig, is_split=True):
        """Weight-name -> split/merge action for TP sharded load/save.

        Reference: llama/modeling.py:1277 _get_tensor_parallel_mappings.
        """
        from ...parallel.tp_mappings import (
            merge_column, merge_fused_column, merge_row,
            split_column, split_fused_column, split_row,
        )

        topo = get_topology()
        tp, rank = topo.mp_degree, topo.get_rank_in("mp")
        actions = {}

        def col(name, fuse_parts=1):
            if fuse_parts > 1:
                if is_split:
                    actions[name] = partial(split_fused_column, tp=tp, rank=rank, parts=fuse_parts)
                else:
                    actions[name] = partial(merge_fused_column, parts=fuse_parts)
            else:
                actions[name] = partial(split_column, tp=tp, rank=rank) if is_split else merge_column

        def row(name):
            actions[name] = partial(split_row, tp=tp, rank=rank) if is_split else merge_row

        col("llama.embed_tokens.weight")
        col("lm_head.weight")
        for i in range(config.num_hidden_layers):
            p = f"llama.layers.{i}."
            if config.fuse_attention_qkv:
                # q/k/v have different widths -> split each part separately
                if is_split:
                    actions[p + "self_attn.qkv_proj.weight"] = partial(
                        _split_qkv, config=config, tp=tp, rank=rank
                    )
                else:
                    actions[p + "self_attn.qkv_proj.weight"] = partial(_merge_qkv, config=config)
            else:
                col(p + "self_attn.q_proj.weight")
                col(p + "self_attn.k_proj.weight")
                col(p + "self_attn.v_proj.weight")
            row(p + "self_attn.o_proj.weight")
            if config.fuse_attention_ffn:
                col(p + "mlp.gate_up_fused_proj.weight", fuse_parts=2)
            else:
                col(p + "mlp.gate_proj.weight")
                col(p + "mlp.up_proj.weight")
            row(p + "mlp.down_proj.weight")
        return actions


def _split_qkv(w, config, tp, rank):
    d = config.head_dim
    q = config.num_attention_heads * d
    kv = config.num_key_value_heads * d
    wq, wk, wv = w.split([q, kv, kv], dim=0)
    return torch.cat([
        wq.chunk(tp, dim=0)[rank], wk.chunk(tp, dim=0)[rank], wv.chunk(tp, dim=0)[rank]
    ], dim=0)


def _merge_qkv(shards, config):
    d = config.head_dim
    tp = len(shards)
    q = config.num_attention_heads * d // tp
    kv = config.num_key_value_heads * d // tp
    qs, ks, vs = [], [], []
    for s in shards:
        wq, wk, wv = s.split([q, kv, kv], dim=0)
        qs.append(wq); ks.append(wk); vs.append(wv)
    return torch.cat(qs + ks + vs, dim=0)


class LlamaModel(LlamaPretrainedModel):
    def __init__(self, config: LlamaConfig):
        super().__init__(config)
        tp = config.tensor_parallel_degree
        if tp > 1:
            self.embed_tokens = VocabParallelEmbedding(config.vocab_size, config.hidden_size)
        else:
            self.embed_tokens = nn.Embedding(config.vocab_size, config.hidden_size)
        self.layers = nn.ModuleList(
            [LlamaDecoderLayer(config, i) for i in range(config.num_hidden_layers)]
        )
        self.norm = LlamaRMSNorm(config)

    def get_input_embeddings(self):
        return self.embed_tokens

    def forward(
        self,
        input_ids=None,
        attn_mask=None,
        startend_row_indices=None,
        past_key_values=None,
        use_cache=False,
        inputs_embeds=None,
    ):
        if inputs_embeds is None:
            inputs_embeds = self.embed_tokens(input_ids)
        hidden_states = inputs_embeds
        if self.config.sequence_parallel and self.config.tensor_parallel_degree > 1:
            # shard activations on the seq dim over the mp group
            # (reference llama/modeling.py ScatterOp after embedding)
            from ...parallel.sequence_parallel import ScatterOp

            hidden_states = ScatterOp(hidden_states)

        position_offset = 0
        if past_key_values is not None and past_key_values[0] is not None:
            position_offset = past_key_values[0][0].shape[1]

        presents = [] if use_cache else None
        for i, layer in enumerate(self.layers):
            past = past_key_values[i] if past_key_values is not None else None
            if self.config.recompute and self.training and past is None:
                hidden_states = checkpoint(
                    layer, hidden_states, attn_mask, startend_row_indices,
                    None, False, position_offset, use_reentrant=False,
                )
            else:
                out = layer(
                    hidden_states, attn_mask, startend_row_indices,
                    past, use_cache, position_offset,
                )
                if use_cache:
                    hidden_states, present = out
                    presents.append(present)
                else:
                    hidden_states = out
        hidden_states = self.norm(hidden_states)
        if use_cache:
            return hidden_states, presents
        return hidden_states


class LlamaPretrainingCriterion(nn.Module):
    """Shift-free CE over pre-shifted labels (reference llama/modeling.py:1799).

    With tensor_parallel_output the logits stay vocab-sharded and we use
    ParallelCrossEntropy; otherwise the fused single-GPU cross-entropy.
    """

    def __init__(self, config: LlamaConfig):
        super().__init__()
        self.config = config
        self.ignore_index = -100
        if config.tensor_parallel_degree > 1 and config.tensor_parallel_output:
            self.loss_fn = ParallelCrossEntropy(ignore_index=self.ignore_index)
        else:
            self.loss_fn = None

    def forward(self, logits, labels):
        if self.loss_fn is not None:
            loss = self.loss_fn(logits, labels)
            loss = loss.reshape(-1)
            valid = (labels.reshape(-1) != self.ignore_index)
            return loss.sum() / valid.sum().clamp(min=1)
        logits = logits.reshape(-1, logits.shape[-1])
        labels = labels.reshape(-1)
        return ops.cross_entropy(logits, labels, self.ignore_index, reduction="mean")


class LlamaForCausalLM(LlamaPretrainedModel, GenerationMixin):
    _tied_weights_keys = []  # lm_head untied by default for llama

    def __init__(self, config: LlamaConfig):
        super().__init__(config)
        self.llama = LlamaModel(config)
        tp = config.tensor_parallel_degree
        if tp > 1:
            self.lm_head = ColumnParallelLinear(
                config.hidden_size, config.vocab_size, bias=False,
                gather_output=not config.tensor_parallel_output,
            )
        else:
            self.lm_head = _Linear(config.hidden_size, config.vocab_size, bias=False)
        self.criterion = LlamaPretrainingCriterion(config)
        self.generation_config = GenerationConfig.from_model_config(config)

    def get_output_embeddings(self):
        return self.lm_head

    def get_input_embeddings(self):
        return self.llama.embed_tokens

    def forward(
        self,
        input_ids=None,
        labels=None,
        attn_mask=None,
        attention_mask=None,  # accepted and ignored when causal-only (parity arg)
        startend_row_indices=None,
        attn_mask_startend_row_indices=None,  # reference kw name (FlashMask)
        past_key_values=None,
        use_cache=False,
        inputs_embeds=None,
        position_ids=None,  # parity arg; rope uses position_offset from cache
        **kwargs,
    ):
        if startend_row_indices is None:
            startend_row_indices = attn_mask_startend_row_indices
        out = self.llama(
            input_ids=input_ids,
            attn_mask=attn_mask,
            startend_row_indices=startend_row_indices,
            past_key_values=past_key_values,
            use_cache=use_cache,
            inputs_embeds=inputs_embeds,
        )
        if use_cache:
            hidden_states, presents = out
        else:
            hidden_states, presents = out, None
        if self.config.sequence_parallel and self.config.tensor_parallel_degree > 1:
            # gather the full sequence before the LM head
            # (reference llama/modeling.py:1896 GatherOp)
            from ...parallel.sequence_parallel import GatherOp

            hidden_states = GatherOp(hidden_states)
        if (labels is not None and not use_cache
                and self.config.use_fused_linear_cross_entropy
                and self.config.tensor_parallel_degree <= 1):
            # chunked head+CE: never materializes [tokens, vocab] logits
            # (reference fused_head_and_loss_fn, tensor_parallel_utils.py:112)
            from ..tensor_parallel_utils import fused_head_and_loss_fn

            loss = fused_head_and_loss_fn(
                hidden_states, self.lm_head.weight, labels)
            return loss, None
        logits = self.lm_head(hidden_states)
        if labels is not None:
            loss = self.criterion(logits, labels)
            return (loss, logits) if not use_cache else (loss, logits, presents)
        return logits if not use_cache else (logits, presents)
