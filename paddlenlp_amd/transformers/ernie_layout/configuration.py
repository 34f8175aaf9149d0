"""ERNIE-Layout config (reference: paddlenlp/transformers/ernie_layout/)."""
from ..ernie.configuration import ErnieConfig

__all__ = ["ErnieLayoutConfig"]


class ErnieLayoutConfig(ErnieConfig):
    model_type = "ernie_layout"

    def __init__(self, max_2d_position_embeddings=1024, coordinate_size=None,
                 **kwargs):
        super().__init__(**kwargs)
        self.max_2d_position_embeddings = max_2d_position_embeddings
        self.coordinate_size = coordinate_size or self.hidden_size
