"""model_type -> class-name registry shared by the Auto* classes.

Reference: paddlenlp/transformers/auto/{configuration,modeling,tokenizer}.py
name->class registries resolved via importlib.
"""
import importlib
import json
import os

# model_type -> {kind: class name}; module defaults to the model_type key.
MODEL_REGISTRY = {
    "llama": {"module": "llama", "config": "LlamaConfig",
              "causal_lm": "LlamaForCausalLM", "base": "LlamaModel"},
    "gpt2": {"module": "gpt", "config": "GPTConfig",
             "causal_lm": "GPTForCausalLM", "base": "GPTModel"},
    "gpt": {"module": "gpt", "config": "GPTConfig",
            "causal_lm": "GPTForCausalLM", "base": "GPTModel"},
    "qwen2": {"module": "qwen2", "config": "Qwen2Config",
              "causal_lm": "Qwen2ForCausalLM", "base": "Qwen2Model"},
    "mixtral": {"module": "mixtral", "config": "MixtralConfig",
                "causal_lm": "MixtralForCausalLM", "base": "MixtralModel"},
    "qwen2_moe": {"module": "qwen2_moe", "config": "Qwen2MoeConfig",
                  "causal_lm": "Qwen2MoeForCausalLM", "base": "Qwen2MoeModel"},
    "deepseek_v2": {"module": "deepseek_v2", "config": "DeepseekV2Config",
                    "causal_lm": "DeepseekV2ForCausalLM", "base": "DeepseekV2Model"},
    "mistral": {"module": "mistral", "config": "MistralConfig",
                "causal_lm": "MistralForCausalLM", "base": "MistralModel"},
    "gemma": {"module": "gemma", "config": "GemmaConfig",
              "causal_lm": "GemmaForCausalLM", "base": "GemmaModel"},
    "opt": {"module": "opt", "config": "OPTConfig",
            "causal_lm": "OPTForCausalLM", "base": "OPTModel"},
    "bloom": {"module": "bloom", "config": "BloomConfig",
              "causal_lm": "BloomForCausalLM", "base": "BloomModel"},
    "falcon": {"module": "falcon", "config": "FalconConfig",
               "causal_lm": "FalconForCausalLM", "base": "FalconModel"},
    "chatglm_v2": {"module": "chatglm_v2", "config": "ChatGLMv2Config",
                   "causal_lm": "ChatGLMv2ForCausalLM", "base": "ChatGLMv2Model"},
    "mamba": {"module": "mamba", "config": "MambaConfig",
              "causal_lm": "MambaForCausalLM", "base": "MambaModel"},
    "gptj": {"module": "gptj", "config": "GPTJConfig",
             "causal_lm": "GPTJForCausalLM", "base": "GPTJModel"},
    "jamba": {"module": "jamba", "config": "JambaConfig",
              "causal_lm": "JambaForCausalLM", "base": "JambaModel"},
    "t5": {"module": "t5", "config": "T5Config", "base": "T5Model",
           "seq2seq_lm": "T5ForConditionalGeneration"},
    "bart": {"module": "bart", "config": "BartConfig", "base": "BartModel",
             "seq2seq_lm": "BartForConditionalGeneration"},
    "mt5": {"module": "mt5", "config": "MT5Config", "base": "MT5Model",
            "seq2seq_lm": "MT5ForConditionalGeneration"},
    "pegasus": {"module": "pegasus", "config": "PegasusConfig",
                "base": "PegasusModel",
                "seq2seq_lm": "PegasusForConditionalGeneration"},
    "mbart": {"module": "mbart", "config": "MBartConfig",
              "base": "MBartModel",
              "seq2seq_lm": "MBartForConditionalGeneration"},
    "qwen": {"module": "qwen", "config": "QWenConfig",
             "causal_lm": "QWenForCausalLM", "base": "QWenModel"},
    "codegen": {"module": "codegen", "config": "CodeGenConfig",
                "causal_lm": "CodeGenForCausalLM", "base": "CodeGenModel"},
    "tinybert": {"module": "tinybert", "config": "TinyBertConfig",
                 "base": "TinyBertModel",
                 "sequence_classification": "TinyBertForSequenceClassification",
                 "question_answering": "TinyBertForQuestionAnswering"},
    "ppminilm": {"module": "ppminilm", "config": "PPMiniLMConfig",
                 "base": "PPMiniLMModel",
                 "sequence_classification": "PPMiniLMForSequenceClassification",
                 "question_answering": "PPMiniLMForQuestionAnswering"},
    "skep": {"module": "skep", "config": "SkepConfig", "base": "SkepModel",
             "sequence_classification": "SkepForSequenceClassification",
             "token_classification": "SkepForTokenClassification"},
    "yuan": {"module": "yuan", "config": "YuanConfig",
             "causal_lm": "YuanForCausalLM", "base": "YuanModel"},
    "blenderbot": {"module": "blenderbot", "config": "BlenderbotConfig",
                   "base": "BlenderbotModel",
                   "seq2seq_lm": "BlenderbotForConditionalGeneration"},
    "blenderbot_small": {
        "module": "blenderbot_small", "config": "BlenderbotSmallConfig",
        "base": "BlenderbotSmallModel",
        "seq2seq_lm": "BlenderbotSmallForConditionalGeneration"},
    "bert": {"module": "bert", "config": "BertConfig", "base": "BertModel",
             "sequence_classification": "BertForSequenceClassification",
             "token_classification": "BertForTokenClassification",
             "question_answering": "BertForQuestionAnswering",
             "masked_lm": "BertForMaskedLM"},
    "ernie_m": {"module": "ernie_m", "config": "ErnieMConfig",
                "base": "ErnieMModel",
                "sequence_classification": "ErnieMForSequenceClassification",
                "token_classification": "ErnieMForTokenClassification"},
    "ernie": {"module": "ernie", "config": "ErnieConfig", "base": "ErnieModel",
              "sequence_classification": "ErnieForSequenceClassification",
              "token_classification": "ErnieForTokenClassification",
              "question_answering": "ErnieForQuestionAnswering",
              "masked_lm": "ErnieForMaskedLM"},
    "roberta": {"module": "roberta", "config": "RobertaConfig", "base": "RobertaModel",
                "sequence_classification": "RobertaForSequenceClassification",
                "token_classification": "RobertaForTokenClassification",
                "question_answering": "RobertaForQuestionAnswering",
                "masked_lm": "RobertaForMaskedLM"},
    "albert": {"module": "albert", "config": "AlbertConfig", "base": "AlbertModel",
               "sequence_classification": "AlbertForSequenceClassification",
               "masked_lm": "AlbertForMaskedLM"},
    "deberta": {"module": "deberta", "config": "DebertaConfig",
                "base": "DebertaModel",
                "sequence_classification": "DebertaForSequenceClassification",
                "masked_lm": "DebertaForMaskedLM"},
    "distilbert": {"module": "distilbert", "config": "DistilBertConfig",
                   "base": "DistilBertModel",
                   "sequence_classification": "DistilBertForSequenceClassification",
                   "masked_lm": "DistilBertForMaskedLM"},
    "roformer": {"module": "roformer", "config": "RoFormerConfig",
                 "base": "RoFormerModel",
                 "sequence_classification": "RoFormerForSequenceClassification",
                 "masked_lm": "RoFormerForMaskedLM"},
    "electra": {"module": "electra", "config": "ElectraConfig", "base": "ElectraModel",
                "sequence_classification": "ElectraForSequenceClassification",
                "token_classification": "ElectraForTokenClassification"},
    "nezha": {"module": "nezha", "config": "NeZhaConfig", "base": "NeZhaModel",
              "sequence_classification": "NeZhaForSequenceClassification",
              "token_classification": "NeZhaForTokenClassification",
              "question_answering": "NeZhaForQuestionAnswering"},
    "mpnet": {"module": "mpnet", "config": "MPNetConfig", "base": "MPNetModel",
              "sequence_classification": "MPNetForSequenceClassification",
              "masked_lm": "MPNetForMaskedLM"},
    "fnet": {"module": "fnet", "config": "FNetConfig", "base": "FNetModel",
             "sequence_classification": "FNetForSequenceClassification",
             "masked_lm": "FNetForMaskedLM"},
    "ernie_gram": {"module": "ernie_gram", "config": "ErnieGramConfig",
                   "base": "ErnieGramModel",
                   "sequence_classification": "ErnieGramForSequenceClassification",
                   "token_classification": "ErnieGramForTokenClassification",
                   "question_answering": "ErnieGramForQuestionAnswering"},
    "rembert": {"module": "rembert", "config": "RemBertConfig",
                "base": "RemBertModel",
                "sequence_classification": "RemBertForSequenceClassification",
                "masked_lm": "RemBertForMaskedLM"},
    "xlm": {"module": "xlm", "config": "XLMConfig", "base": "XLMModel",
            "sequence_classification": "XLMForSequenceClassification",
            "masked_lm": "XLMWithLMHeadModel"},
    "unified_transformer": {
        "module": "unified_transformer", "config": "UnifiedTransformerConfig",
        "base": "UnifiedTransformerModel",
        "causal_lm": "UnifiedTransformerLMHeadModel"},
    "unimo": {"module": "unimo", "config": "UNIMOConfig",
              "base": "UNIMOModel", "causal_lm": "UNIMOLMHeadModel"},
    "chatglm": {"module": "chatglm", "config": "ChatGLMConfig",
                "causal_lm": "ChatGLMForCausalLM", "base": "ChatGLMModel"},
    "megatronbert": {
        "module": "megatronbert", "config": "MegatronBertConfig",
        "base": "MegatronBertModel",
        "sequence_classification": "MegatronBertForSequenceClassification",
        "question_answering": "MegatronBertForQuestionAnswering"},
    "layoutlm": {"module": "layoutlm", "config": "LayoutLMConfig",
                 "base": "LayoutLMModel",
                 "sequence_classification": "LayoutLMForSequenceClassification",
                 "token_classification": "LayoutLMForTokenClassification"},
    # XLNet's LM head is the permutation-LM objective (bidirectional
    # content stream), not a standard causal LM: no causal_lm registration
    "xlnet": {"module": "xlnet", "config": "XLNetConfig", "base": "XLNetModel",
              "sequence_classification": "XLNetForSequenceClassification"},
    "reformer": {"module": "reformer", "config": "ReformerConfig",
                 "base": "ReformerModel",
                 "causal_lm": "ReformerModelWithLMHead"},
    "bigbird": {"module": "bigbird", "config": "BigBirdConfig",
                "base": "BigBirdModel",
                "sequence_classification": "BigBirdForSequenceClassification",
                "masked_lm": "BigBirdForMaskedLM"},
    "nystromformer": {"module": "nystromformer",
                      "config": "NystromformerConfig",
                      "base": "NystromformerModel",
                      "sequence_classification": "NystromformerForSequenceClassification"},
    "convbert": {"module": "convbert", "config": "ConvBertConfig",
                 "base": "ConvBertModel",
                 "sequence_classification": "ConvBertForSequenceClassification",
                 "masked_lm": "ConvBertForMaskedLM"},
    "ctrl": {"module": "ctrl", "config": "CTRLConfig", "base": "CTRLModel",
             "causal_lm": "CTRLLMHeadModel",
             "sequence_classification": "CTRLForSequenceClassification"},
    "glm": {"module": "glm", "config": "GLMConfig", "base": "GLMModel",
            "causal_lm": "GLMForConditionalGeneration"},
    "mobilebert": {"module": "mobilebert", "config": "MobileBertConfig",
                   "base": "MobileBertModel",
                   "sequence_classification": "MobileBertForSequenceClassification"},
    "squeezebert": {"module": "squeezebert", "config": "SqueezeBertConfig",
                    "base": "SqueezeBertModel",
                    "sequence_classification": "SqueezeBertForSequenceClassification"},
    "gau_alpha": {"module": "gau_alpha", "config": "GAUAlphaConfig",
                  "base": "GAUAlphaModel",
                  "sequence_classification": "GAUAlphaForSequenceClassification"},
    "deberta-v2": {"module": "deberta_v2", "config": "DebertaV2Config",
                   "base": "DebertaV2Model",
                   "sequence_classification": "DebertaV2ForSequenceClassification",
                   "masked_lm": "DebertaV2ForMaskedLM"},
    "chinesebert": {"module": "chinesebert", "config": "ChineseBertConfig",
                    "base": "ChineseBertModel",
                    "sequence_classification": "ChineseBertForSequenceClassification"},
    "funnel": {"module": "funnel", "config": "FunnelConfig",
               "base": "FunnelModel",
               "sequence_classification": "FunnelForSequenceClassification",
               "token_classification": "FunnelForTokenClassification"},
    "prophetnet": {"module": "prophetnet", "config": "ProphetNetConfig",
                   "base": "ProphetNetModel",
                   "seq2seq_lm": "ProphetNetForConditionalGeneration"},
    "luke": {"module": "luke", "config": "LukeConfig", "base": "LukeModel",
             "entity_classification": "LukeForEntityClassification"},
    "roformerv2": {"module": "roformerv2", "config": "RoFormerv2Config",
                   "base": "RoFormerv2Model",
                   "sequence_classification": "RoFormerv2ForSequenceClassification"},
    "ernie_ctm": {"module": "ernie_ctm", "config": "ErnieCtmConfig",
                  "base": "ErnieCtmModel"},
    "ernie_doc": {"module": "ernie_doc", "config": "ErnieDocConfig",
                  "base": "ErnieDocModel",
                  "sequence_classification": "ErnieDocForSequenceClassification"},
    "transformer": {"module": "transformer", "config": "TransformerConfig",
                    "base": "TransformerModel"},
    "chineseclip": {"module": "chineseclip", "config": "ChineseCLIPConfig",
                    "base": "ChineseCLIPModel"},
    "bit": {"module": "bit", "config": "BitConfig", "base": "BitModel",
            "image_classification": "BitForImageClassification"},
    "ernie_code": {"module": "ernie_code", "config": "ErnieCodeConfig",
                   "base": "ErnieCodeModel",
                   "seq2seq_lm": "ErnieCodeForConditionalGeneration"},
    "speecht5": {"module": "speecht5", "config": "SpeechT5Config",
                 "base": "SpeechT5Model"},
    "clap": {"module": "clap", "config": "ClapConfig", "base": "ClapModel"},
    "blip-2": {"module": "blip_2", "config": "Blip2Config",
               "base": "Blip2Model"},
    "dpt": {"module": "dpt", "config": "DPTConfig", "base": "DPTModel",
            "depth_estimation": "DPTForDepthEstimation"},
    "minigpt4": {"module": "minigpt4", "config": "MiniGPT4Config",
                 "base": "MiniGPT4ForConditionalGeneration"},
    "visualglm": {"module": "visualglm", "config": "VisualGLMConfig",
                  "base": "VisualGLMForConditionalGeneration"},
    "ernie_vil": {"module": "ernie_vil", "config": "ErnieViLConfig",
                  "base": "ErnieViLModel"},
    "dallebart": {"module": "dallebart", "config": "DalleBartConfig",
                  "base": "DalleBartModel",
                  "seq2seq_lm": "DalleBartForConditionalGeneration"},
    "artist": {"module": "artist", "config": "ArtistConfig",
               "base": "ArtistModel",
               "causal_lm": "ArtistForConditionalGeneration"},
    "ernie_gen": {"module": "ernie_gen", "config": "ErnieGenConfig",
                  "base": "ErnieGenModel"},
    "clipseg": {"module": "clipseg", "config": "CLIPSegConfig",
                "base": "CLIPSegForImageSegmentation"},
    "layoutlmv2": {"module": "layoutlmv2", "config": "LayoutLMv2Config",
                   "base": "LayoutLMv2Model",
                   "token_classification": "LayoutLMv2ForTokenClassification"},
    "layoutxlm": {"module": "layoutxlm", "config": "LayoutXLMConfig",
                  "base": "LayoutXLMModel",
                  "token_classification": "LayoutXLMForTokenClassification"},
}


def resolve_model_type(path: str) -> str:
    config_file = os.path.join(path, "config.json") if os.path.isdir(path) else path
    with open(config_file) as f:
        cfg = json.load(f)
    model_type = cfg.get("model_type")
    if model_type is None:
        archs = cfg.get("architectures") or []
        for arch in archs:
            for mt, entry in MODEL_REGISTRY.items():
                if arch in entry.values():
                    return mt
        raise ValueError(f"Cannot infer model_type from {config_file}")
    return model_type


_TASK_HEAD_KINDS = {"sequence_classification", "token_classification",
                    "question_answering", "multiple_choice"}


def get_class(model_type: str, kind: str):
    """kind in {config, base, causal_lm, sequence_classification,
    token_classification, question_answering, multiple_choice, masked_lm}.

    Task-head kinds missing from a family's registry entry are synthesized
    over its base model (transformers.task_heads) — reference parity:
    nearly every family exposes these heads."""
    if model_type not in MODEL_REGISTRY:
        raise ValueError(
            f"Unknown model_type '{model_type}'. Registered: {sorted(MODEL_REGISTRY)}"
        )
    entry = MODEL_REGISTRY[model_type]
    if kind not in entry:
        if kind in _TASK_HEAD_KINDS and "base" in entry:
            from ..task_heads import synthesize_head

            module = importlib.import_module(
                f"paddlenlp_amd.transformers.{entry['module']}")
            base_cls = getattr(module, entry["base"])
            return synthesize_head(base_cls, kind)
        raise ValueError(f"model_type '{model_type}' has no {kind} head")
    module = importlib.import_module(
        f"paddlenlp_amd.transformers.{entry['module']}")
    return getattr(module, entry[kind])
