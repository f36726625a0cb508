"""distributed_embeddings_amd — MI355X-native distributed embedding framework.

A from-scratch PyTorch-ROCm framework with the capabilities of
NVIDIA-Merlin/distributed-embeddings (reference at /root/reference), built for
AMD Instinct MI355X (gfx950 / CDNA4): hand-written HIP kernels for the
embedding hot ops, RCCL collectives over the 8-GPU xGMI full mesh via
``torch.distributed``, and memory layouts sized for 288 GB HBM3E per GPU.

Public API (parity: reference ``distributed_embeddings/__init__.py:17-27``):
"""

from .ops.embedding_lookup import Ragged, embedding_lookup, row_to_split
from .layers.embedding import ConcatOneHotEmbedding, Embedding, scaled_uniform_init
from .layers.integer_lookup import IntegerLookup
from .parallel.strategy import DistEmbeddingStrategy, TableConfig
from .parallel.dist_embedding import DistributedEmbedding
from .parallel.optim import SparseEmbeddingOptimizer
from .parallel.grad import (
    BroadcastParametersOnFirstStep,
    DistributedOptimizer,
    allreduce_gradients,
    broadcast_parameters,
    broadcast_variables,
)
from .parallel import comm
from .parallel.checkpoint import (load_embedding_checkpoint,
                                  save_embedding_checkpoint)

__version__ = "0.1.0"

__all__ = [
    "Ragged",
    "embedding_lookup",
    "row_to_split",
    "Embedding",
    "IntegerLookup",
    "ConcatOneHotEmbedding",
    "scaled_uniform_init",
    "TableConfig",
    "DistEmbeddingStrategy",
    "DistributedEmbedding",
    "DistributedOptimizer",
    "SparseEmbeddingOptimizer",
    "BroadcastParametersOnFirstStep",
    "allreduce_gradients",
    "broadcast_parameters",
    "save_embedding_checkpoint",
    "load_embedding_checkpoint",
    "broadcast_variables",
    "comm",
]
