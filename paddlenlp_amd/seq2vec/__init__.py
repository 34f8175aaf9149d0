from .encoder import (  # noqa: F401
    BoWEncoder,
    CNNEncoder,
    GRUEncoder,
    LSTMEncoder,
    RNNEncoder,
    TCNEncoder,
)
