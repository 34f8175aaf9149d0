"""paddlenlp_amd: an MI355X-native LLM training + inference toolkit.

Brand-new framework with the capabilities and public API surface of
PaddleNLP (reference: PaddlePaddle/PaddleNLP @ 2024-10-24), built on
PyTorch-ROCm with hand-written CDNA4 (gfx950) HIP kernels for the hot
path and RCCL collectives over xGMI for 4D parallelism.

Public surface mirrors the reference layer map:
  - paddlenlp_amd.transformers: model zoo (32 families) + Auto* registries
  - paddlenlp_amd.trainer:      Trainer / TrainingArguments / callbacks /
                                unified checkpoint / compression
  - paddlenlp_amd.data:         collators, Megatron-style causal datasets,
                                Stack/Pad/Tuple/Dict batchify, Vocab
  - paddlenlp_amd.parallel:     topology + DP/ZeRO-1/2/3/TP/SP/PP(+VPP)/
                                SEP/CP(zigzag)/EP engines (RCCL over xGMI)
  - paddlenlp_amd.ops:          CDNA4 HIP fused ops (flash attention +
                                FlashMask, RMSNorm, RoPE, SwiGLU, fused
                                AdamW, fused CE, paged decode, int8 KV)
  - paddlenlp_amd.generation:   greedy/sample/beam/group-beam, speculative
                                decoding, stopping criteria
  - paddlenlp_amd.peft:         LoRA / prefix / VeRA / QLoRA
  - paddlenlp_amd.experimental: fused inference engine (paged KV, TP,
                                hipGraph), AutoNLP
  - paddlenlp_amd.quantization / taskflow / prompt / seq2vec / embeddings /
    dataaug / metrics / losses / server / cli
"""

__version__ = "0.1.0"
