"""paddlenlp_amd: an MI355X-native LLM training + inference toolkit.

Brand-new framework with the capabilities and public API surface of
PaddleNLP (reference: PaddlePaddle/PaddleNLP @ 2024-10-24), built on
PyTorch-ROCm with hand-written CDNA4 (gfx950) HIP kernels for the hot
path and RCCL collectives over xGMI for 4D parallelism.

Public surface mirrors the reference layer map:
  - paddlenlp_amd.transformers: model zoo + Auto* registries + PretrainedModel
  - paddlenlp_amd.trainer:      Trainer / TrainingArguments / callbacks
  - paddlenlp_amd.data:         collators, Megatron-style causal datasets
  - paddlenlp_amd.parallel:     topology + DP/ZeRO/TP/SP/PP/CP engines (RCCL)
  - paddlenlp_amd.ops:          CDNA4 HIP fused ops (flash attention, RMSNorm,
                                RoPE, SwiGLU, fused AdamW, fused cross-entropy)
  - paddlenlp_amd.generation:   decoding loops (greedy/sample/beam)
  - paddlenlp_amd.peft:         LoRA / prefix tuning
"""

__version__ = "0.1.0"
