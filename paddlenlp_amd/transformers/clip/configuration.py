"""CLIP config (reference: paddlenlp/transformers/clip/configuration.py)."""
from ..configuration_utils import PretrainedConfig

__all__ = ["CLIPTextConfig", "CLIPVisionConfig", "CLIPConfig"]


class CLIPTextConfig(PretrainedConfig):
    model_type = "clip_text_model"

    def __init__(self, vocab_size=49408, hidden_size=512,
                 num_hidden_layers=12, num_attention_heads=8,
                 intermediate_size=2048, hidden_act="gelu",
                 max_position_embeddings=77, layer_norm_eps=1e-5,
                 initializer_range=0.02, pad_token_id=1, bos_token_id=49406,
                 eos_token_id=49407, **kwargs):
        super().__init__(**kwargs)
        self.vocab_size = vocab_size
        self.hidden_size = hidden_size
        self.num_hidden_layers = num_hidden_layers
        self.num_attention_heads = num_attention_heads
        self.intermediate_size = intermediate_size
        self.hidden_act = hidden_act
        self.max_position_embeddings = max_position_embeddings
        self.layer_norm_eps = layer_norm_eps
        self.initializer_range = initializer_range
        self.pad_token_id = pad_token_id
        self.bos_token_id = bos_token_id
        self.eos_token_id = eos_token_id


class CLIPVisionConfig(PretrainedConfig):
    model_type = "clip_vision_model"

    def __init__(self, hidden_size=768, num_hidden_layers=12,
                 num_attention_heads=12, intermediate_size=3072,
                 hidden_act="gelu", image_size=224, patch_size=32,
                 num_channels=3, layer_norm_eps=1e-5,
                 initializer_range=0.02, **kwargs):
        super().__init__(**kwargs)
        self.hidden_size = hidden_size
        self.num_hidden_layers = num_hidden_layers
        self.num_attention_heads = num_attention_heads
        self.intermediate_size = intermediate_size
        self.hidden_act = hidden_act
        self.image_size = image_size
        self.patch_size = patch_size
        self.num_channels = num_channels
        self.layer_norm_eps = layer_norm_eps
        self.initializer_range = initializer_range


class CLIPConfig(PretrainedConfig):
    model_type = "clip"

    def __init__(self, text_config=None, vision_config=None,
                 projection_dim=512, logit_scale_init_value=2.6592, **kwargs):
        super().__init__(**kwargs)
        self.text_config = CLIPTextConfig(**(text_config or {}))
        self.vision_config = CLIPVisionConfig(**(vision_config or {}))
        self.projection_dim = projection_dim
        self.logit_scale_init_value = logit_scale_init_value
        self.initializer_range = 0.02

    def to_dict(self):
        d = dict(self.__dict__)
        d["text_config"] = dict(self.text_config.__dict__)
        d["vision_config"] = dict(self.vision_config.__dict__)
        d["model_type"] = self.model_type
        return d
