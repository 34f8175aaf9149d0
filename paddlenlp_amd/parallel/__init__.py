from .topology import Topology, get_topology, init_parallel_env, set_topology  # noqa: F401
from .data_parallel import (  # noqa: F401
    DataParallel,
    broadcast_parameters,
    fused_allreduce_gradients,
)
from .tensor_parallel import (  # noqa: F401
    ColumnParallelLinear,
    ParallelCrossEntropy,
    RowParallelLinear,
    VocabParallelEmbedding,
    parallel_matmul,
)
