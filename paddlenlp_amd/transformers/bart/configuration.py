"""BART config (reference: paddlenlp/transformers/bart/configuration.py)."""
from ..configuration_utils import PretrainedConfig

__all__ = ["BartConfig"]


class BartConfig(PretrainedConfig):
    model_type = "bart"

    attribute_map = {
        "hidden_size": "d_model",
        "num_attention_heads": "encoder_attention_heads",
        "num_hidden_layers": "encoder_layers",
    }

    def __init__(
        self,
        vocab_size=50265,
        d_model=768,
        encoder_layers=6,
        decoder_layers=6,
        encoder_attention_heads=12,
        decoder_attention_heads=12,
        encoder_ffn_dim=3072,
        decoder_ffn_dim=3072,
        activation_function="gelu",
        dropout=0.1,
        attention_dropout=0.0,
        max_position_embeddings=1024,
        init_std=0.02,
        scale_embedding=False,
        pad_token_id=1,
        bos_token_id=0,
        eos_token_id=2,
        decoder_start_token_id=2,
        forced_eos_token_id=2,
        tie_word_embeddings=True,
        **kwargs,
    ):
        kwargs.setdefault("tie_word_embeddings", tie_word_embeddings)
        super().__init__(**kwargs)
        self.vocab_size = vocab_size
        self.d_model = d_model
        self.encoder_layers = encoder_layers
        self.decoder_layers = decoder_layers
        self.encoder_attention_heads = encoder_attention_heads
        self.decoder_attention_heads = decoder_attention_heads
        self.encoder_ffn_dim = encoder_ffn_dim
        self.decoder_ffn_dim = decoder_ffn_dim
        self.activation_function = activation_function
        self.dropout = dropout
        self.attention_dropout = attention_dropout
        self.max_position_embeddings = max_position_embeddings
        self.init_std = init_std
        self.scale_embedding = scale_embedding
        self.pad_token_id = pad_token_id
        self.bos_token_id = bos_token_id
        self.eos_token_id = eos_token_id
        self.decoder_start_token_id = decoder_start_token_id
        self.forced_eos_token_id = forced_eos_token_id

    @property
    def head_dim(self):
        return self.d_model // self.encoder_attention_heads
