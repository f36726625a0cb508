from . import comm
from .strategy import DistEmbeddingStrategy, TableConfig
from .dist_embedding import DistributedEmbedding
from .grad import (BroadcastParametersOnFirstStep, DistributedOptimizer,
                   allreduce_gradients, broadcast_parameters, broadcast_variables)
