"""ALBERT config (reference: paddlenlp/transformers/albert/configuration.py)."""
from ..configuration_utils import PretrainedConfig

__all__ = ["AlbertConfig"]


class AlbertConfig(PretrainedConfig):
    model_type = "albert"

    attribute_map = {
        "num_classes": "num_labels",
    }

    def __init__(self, vocab_size=30000, embedding_size=128, hidden_size=768,
                 num_hidden_layers=12, num_hidden_groups=1,
                 num_attention_heads=12, intermediate_size=3072,
                 hidden_act="gelu", hidden_dropout_prob=0.0,
                 attention_probs_dropout_prob=0.0,
                 max_position_embeddings=512, type_vocab_size=2,
                 initializer_range=0.02, layer_norm_eps=1e-12,
                 classifier_dropout=None,
                 pad_token_id=0, bos_token_id=2, eos_token_id=3,
                 num_labels=2, **kwargs):
        super().__init__(**kwargs)
        self.vocab_size = vocab_size
        self.embedding_size = embedding_size
        self.hidden_size = hidden_size
        self.num_hidden_layers = num_hidden_layers
        self.num_hidden_groups = num_hidden_groups
        self.num_attention_heads = num_attention_heads
        self.intermediate_size = intermediate_size
        self.hidden_act = hidden_act
        self.hidden_dropout_prob = hidden_dropout_prob
        self.attention_probs_dropout_prob = attention_probs_dropout_prob
        self.max_position_embeddings = max_position_embeddings
        self.type_vocab_size = type_vocab_size
        self.initializer_range = initializer_range
        self.layer_norm_eps = layer_norm_eps
        self.classifier_dropout = classifier_dropout
        self.pad_token_id = pad_token_id
        self.bos_token_id = bos_token_id
        self.eos_token_id = eos_token_id
        self.num_labels = num_labels

    @property
    def head_dim(self):
        return self.hidden_size // self.num_attention_heads
