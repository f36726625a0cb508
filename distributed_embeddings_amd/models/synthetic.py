"""Synthetic benchmark models.

Capability parity with the reference
(``/root/reference/examples/benchmarks/synthetic_models/synthetic_models.py:116-176``):
expand the ``EmbeddingConfig`` list into per-table ``Embedding(combiner='sum')``
layers + an ``input_table_map`` (shared tables get one input per hotness),
wrap in ``DistributedEmbedding(memory_balanced)``, emulate the interaction
with an average-pool when ``interact_stride`` is set, finish with an MLP.
"""

from typing import List, Optional

import torch
from torch import nn

from ..layers.embedding import Embedding
from ..parallel import comm
from ..parallel.dist_embedding import DistributedEmbedding
from .config import ModelConfig


def expand_tables(model_config: ModelConfig):
    """EmbeddingConfig list -> (table configs, input_table_map, hotness list).

    ``shared=True``: one table per ``num_tables``, one input per nnz entry
    (all mapped to the same table).  ``shared=False``: each (table, nnz)
    combination is its OWN table — ``num_tables * len(nnz)`` tables in total
    (parity: reference ``config_v3.py:21-24`` /
    ``synthetic_models.py:116-176``).
    """
    tables, input_table_map, hotness = [], [], []
    for cfg in model_config.embedding_configs:
        for _ in range(cfg.num_tables):
            if cfg.shared:
                t = len(tables)
                tables.append((cfg.num_rows, cfg.width))
                for nnz in cfg.nnz:
                    input_table_map.append(t)
                    hotness.append(nnz)
            else:
                for nnz in cfg.nnz:
                    input_table_map.append(len(tables))
                    tables.append((cfg.num_rows, cfg.width))
                    hotness.append(nnz)
    return tables, input_table_map, hotness


class SyntheticModel(nn.Module):
    def __init__(self, model_config: ModelConfig,
                 column_slice_threshold: Optional[int] = None,
                 dp_input: bool = True,
                 strategy: str = "memory_balanced",
                 data_parallel_threshold: Optional[int] = None):
        super().__init__()
        self.config = model_config
        from ..parallel.strategy import TableConfig
        tables, input_table_map, self.hotness = expand_tables(model_config)
        self.input_table_map = input_table_map
        layers = [TableConfig(rows, width, "sum") for rows, width in tables]
        self.distributed = comm.world_size() > 1
        self.embeddings = DistributedEmbedding(
            layers, strategy=strategy, dp_input=dp_input,
            input_table_map=input_table_map,
            column_slice_threshold=column_slice_threshold,
            data_parallel_threshold=data_parallel_threshold)

        self.interact_stride = model_config.interact_stride
        if self.interact_stride:
            self.interact = nn.AvgPool1d(self.interact_stride,
                                         stride=self.interact_stride,
                                         ceil_mode=True)
        else:
            self.interact = None

        total_width = sum(tables[t][1] for t in input_table_map)
        if self.interact_stride:
            import math
            total_width = math.ceil(total_width / self.interact_stride)
        mlp_in = total_width + model_config.num_numerical_features
        mods: List[nn.Module] = []
        d = mlp_in
        for s in model_config.mlp_sizes:
            mods += [nn.Linear(d, s), nn.ReLU(inplace=True)]
            d = s
        mods.append(nn.Linear(d, 1))
        self.mlp = nn.Sequential(*mods)

    def forward(self, numerical: torch.Tensor, cat_features) -> torch.Tensor:
        # under autocast the MLP runs bf16: ask for bf16 embedding outputs
        want = torch.bfloat16 if torch.is_autocast_enabled() else numerical.dtype
        embs = self.embeddings(list(cat_features), output_dtype=want)
        x = torch.cat([e.to(want) for e in embs], dim=1)
        if self.interact is not None:
            x = self.interact(x.unsqueeze(1)).squeeze(1)
        x = torch.cat([x, numerical], dim=1)
        return self.mlp(x)
