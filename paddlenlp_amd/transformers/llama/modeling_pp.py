"""Pipeline-parallel Llama: flattened LayerDesc stack.

Reference behavior: paddlenlp/transformers/llama/modeling_pp.py:296
(LlamaForCausalLMPipe — add_sequential_layer list Embedding -> N x
DecoderLayer -> RMSNorm -> LMHead :360-387, SharedLayerDesc for tied
embeddings :359-385, seg_method "layer:LlamaDecoderLayer" :391-393).
"""
from __future__ import annotations


import torch.nn as nn

from ...parallel.pipeline import LayerDesc, PipelineModule
from ...parallel.topology import get_topology
from .configuration import LlamaConfig
from .modeling import LlamaDecoderLayer, LlamaPretrainingCriterion, LlamaRMSNorm, _Linear
from ...parallel.tensor_parallel import ColumnParallelLinear, VocabParallelEmbedding

__all__ = ["LlamaForCausalLMPipe", "Qwen2ForCausalLMPipe", "MistralForCausalLMPipe"]


class EmbeddingPipe(nn.Module):
    def __init__(self, config: LlamaConfig):
        super().__init__()
        self.config = config
        if config.tensor_parallel_degree > 1:
            self.embed_tokens = VocabParallelEmbedding(config.vocab_size, config.hidden_size)
        else:
            self.embed_tokens = nn.Embedding(config.vocab_size, config.hidden_size)

    @property
    def embedding_weight(self):
        return self.embed_tokens.weight

    def forward(self, input_ids):
        h = self.embed_tokens(input_ids)
        if getattr(self.config, "sequence_parallel", False) and \
                self.config.tensor_parallel_degree > 1:
            # shard activations on the sequence dim for the whole stack
            # (reference modeling_pp first-stage ScatterOp; the shape-
            # negotiated p2p carries the [B, s/mp, H] layout across stages)
            from ...parallel.sequence_parallel import ScatterOp

            h = ScatterOp(h)
        return h


class RMSNormPipe(LlamaRMSNorm):
    pass


class LMHeadPipe(nn.Module):
    def __init__(self, config: LlamaConfig):
        super().__init__()
        self.config = config
        if config.tensor_parallel_degree > 1:
            self.lm_head = ColumnParallelLinear(
                config.hidden_size, config.vocab_size, bias=False,
                gather_output=not config.tensor_parallel_output,
            )
        else:
            self.lm_head = _Linear(config.hidden_size, config.vocab_size, bias=False)

    def forward(self, hidden):
        if getattr(self.config, "sequence_parallel", False) and \
                self.config.tensor_parallel_degree > 1:
            # re-assemble the full sequence before the vocab projection
            # (reference llama/modeling.py:1896 GatherOp)
            from ...parallel.sequence_parallel import GatherOp

            hidden = GatherOp(hidden)
        return self.lm_head(hidden)


class LlamaForCausalLMPipe(PipelineModule):
    """Build with `LlamaForCausalLMPipe(config)`; run through PipelineEngine.

    Weight names intentionally match LlamaForCausalLM once remapped by
    `pp_param_name_map` (PipelinePretrainedModel behavior in the reference
    model_utils.py: maps pipe-layer param names back to base names).

    Subclasses override `decoder_layer_cls` / `base_prefix` to give any
    llama-shaped family a pipe variant (Qwen2/Mistral below).
    """

    config_class = LlamaConfig
    decoder_layer_cls = LlamaDecoderLayer
    base_prefix = "llama"

    @classmethod
    def _get_tensor_parallel_mappings(cls, config, is_split=True):
        from .modeling import LlamaForCausalLM

        return LlamaForCausalLM._get_tensor_parallel_mappings(config, is_split)

    def __init__(self, config: LlamaConfig, num_virtual_stages: int = 1):
        criterion = LlamaPretrainingCriterion(config)
        descs = [LayerDesc(EmbeddingPipe, config, name="embedding")]
        for i in range(config.num_hidden_layers):
            descs.append(LayerDesc(self.decoder_layer_cls, config, i,
                                   name=f"layer_{i}"))
        descs.append(LayerDesc(RMSNormPipe, config, name="final_norm"))
        descs.append(LayerDesc(LMHeadPipe, config, name="lm_head"))

        def loss_fn(logits, micro_batch):
            return criterion(logits, micro_batch["labels"])

        super().__init__(
            descs,
            loss_fn=loss_fn,
            seg_method=f"layer:{self.decoder_layer_cls.__name__}",
            topology=get_topology(),
            num_virtual_stages=num_virtual_stages,
        )
        self.config = config
        self._local_names = [d.name for d in self._layer_descs_local]

    def pp_param_name_map(self):
        """pipe-local param name -> LlamaForCausalLM name (for checkpoints)."""
        mapping = {}
        for idx, (layer, desc) in enumerate(zip(self.local_layers, self._layer_descs_local)):
            for pname, _ in layer.named_parameters():
                local = f"local_layers.{idx}.{pname}"
                if desc.name == "embedding":
                    base = pname.replace(
                        "embed_tokens", f"{self.base_prefix}.embed_tokens")
                elif desc.name.startswith("layer_"):
                    li = desc.name.split("_")[1]
                    base = f"{self.base_prefix}.layers.{li}.{pname}"
                elif desc.name == "final_norm":
                    base = f"{self.base_prefix}.norm.{pname}"
                else:  # lm_head
                    base = pname if pname.startswith("lm_head") else f"lm_head.{pname}"
                mapping[local] = base
        return mapping

    def state_dict_with_base_names(self):
        mapping = self.pp_param_name_map()
        sd = self.state_dict()
        return {mapping.get(k, k): v for k, v in sd.items()}

    def load_base_state_dict(self, base_sd, strict: bool = False):
        mapping = {v: k for k, v in self.pp_param_name_map().items()}
        local_sd = {}
        for base_name, tensor in base_sd.items():
            if base_name in mapping:
                local_sd[mapping[base_name]] = tensor
        missing, unexpected = self.load_state_dict(local_sd, strict=False)
        real_missing = [m for m in missing if not m.startswith("local_layers")] if strict else []
        return missing, unexpected


def _make_pipe_variant(name, config_cls, layer_cls, prefix, doc):
    return type(name, (LlamaForCausalLMPipe,), {
        "config_class": config_cls,
        "decoder_layer_cls": layer_cls,
        "base_prefix": prefix,
        "__doc__": doc,
    })


def _qwen2_pipe():
    from ..qwen2.configuration import Qwen2Config
    from ..qwen2.modeling import Qwen2DecoderLayer

    return _make_pipe_variant(
        "Qwen2ForCausalLMPipe", Qwen2Config, Qwen2DecoderLayer, "qwen2",
        "Qwen2 pipeline variant (biased-QKV llama-shape layers).")


def _mistral_pipe():
    from ..mistral.configuration import MistralConfig

    # mistral reuses the llama decoder layer
    return _make_pipe_variant(
        "MistralForCausalLMPipe", MistralConfig, LlamaDecoderLayer,
        "mistral", "Mistral pipeline variant.")


Qwen2ForCausalLMPipe = _qwen2_pipe()
MistralForCausalLMPipe = _mistral_pipe()
