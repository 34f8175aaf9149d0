"""BLIP config (reference: paddlenlp/transformers/blip/configuration.py)."""
from ..configuration_utils import PretrainedConfig
from ..clip.configuration import CLIPVisionConfig

__all__ = ["BlipTextConfig", "BlipVisionConfig", "BlipConfig"]


class BlipTextConfig(PretrainedConfig):
    model_type = "blip_text_model"

    def __init__(self, vocab_size=30524, hidden_size=768,
                 num_hidden_layers=12, num_attention_heads=12,
                 intermediate_size=3072, hidden_act="gelu",
                 hidden_dropout_prob=0.0, attention_probs_dropout_prob=0.0,
                 max_position_embeddings=512, layer_norm_eps=1e-12,
                 initializer_range=0.02, pad_token_id=0, bos_token_id=30522,
                 eos_token_id=102, **kwargs):
        super().__init__(**kwargs)
        self.vocab_size = vocab_size
        self.hidden_size = hidden_size
        self.num_hidden_layers = num_hidden_layers
        self.num_attention_heads = num_attention_heads
        self.intermediate_size = intermediate_size
        self.hidden_act = hidden_act
        self.hidden_dropout_prob = hidden_dropout_prob
        self.attention_probs_dropout_prob = attention_probs_dropout_prob
        self.max_position_embeddings = max_position_embeddings
        self.layer_norm_eps = layer_norm_eps
        self.initializer_range = initializer_range
        self.pad_token_id = pad_token_id
        self.bos_token_id = bos_token_id
        self.eos_token_id = eos_token_id
        self.type_vocab_size = 0


class BlipVisionConfig(CLIPVisionConfig):
    model_type = "blip_vision_model"


class BlipConfig(PretrainedConfig):
    model_type = "blip"

    def __init__(self, text_config=None, vision_config=None,
                 projection_dim=256, logit_scale_init_value=2.6592, **kwargs):
        super().__init__(**kwargs)
        self.text_config = BlipTextConfig(**(text_config or {}))
        self.vision_config = BlipVisionConfig(**(vision_config or {}))
        self.projection_dim = projection_dim
        self.logit_scale_init_value = logit_scale_init_value
        self.initializer_range = 0.02

    def to_dict(self):
        d = dict(self.__dict__)
        d["text_config"] = dict(self.text_config.__dict__)
        d["vision_config"] = dict(self.vision_config.__dict__)
        d["model_type"] = self.model_type
        return d
