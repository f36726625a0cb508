"""DLRM model (MI355X-native flagship).

Capability parity with the reference example
(``/root/reference/examples/dlrm/main.py:74-147`` + ``utils.py:92-113``):
bottom MLP over dense features, per-category embedding tables behind
``DistributedEmbedding`` when world>1, pairwise dot-product feature
interaction (lower triangle) re-concatenated with the bottom MLP output, and
a top MLP producing one logit.

MI355X notes: the MLPs run as plain ``nn.Linear`` (hipBLASLt GEMMs, TunableOp
selections shipped in profiles/) under bf16 autocast; the pairwise-dot
interaction runs as one fused MFMA kernel each direction
(``csrc/dot_interact.hip``, dispatched by ``ops/dot_interact.py``; plain
bmm fallback when the shape or dtype gate fails).
"""

import math
import os
from typing import List, Optional, Sequence

import torch
from torch import nn

from ..layers.embedding import Embedding, scaled_uniform_init
from ..ops.dot_interact import dot_interact as fused_dot_interact
from ..ops.dot_interact import dot_interact_packed
from ..parallel import comm
from ..parallel.dist_embedding import DistributedEmbedding


def dot_interact(emb_outs: List[torch.Tensor], bottom_mlp_out: torch.Tensor,
                 pad_to: int = 0) -> torch.Tensor:
    """Pairwise-dot interaction over [bottom_out] + embeddings.

    Parity: reference ``utils.py:92-113`` — lower-triangular portion of the
    [F+1, F+1] Gram matrix, then re-concat the bottom MLP output.
    ``pad_to``: zero-pad the output width to this size so the following GEMM
    gets an MFMA-friendly K (479 -> 512 for Criteo).
    """
    feats = torch.stack([bottom_mlp_out] + emb_outs, dim=1)  # [B, F, D]
    gram = torch.bmm(feats, feats.transpose(1, 2))           # [B, F, F]
    f = gram.shape[1]
    ii, jj = torch.tril_indices(f, f, offset=-1, device=gram.device)
    interactions = gram[:, ii, jj]                           # [B, F(F-1)/2]
    parts = [interactions, bottom_mlp_out]
    width = interactions.shape[1] + bottom_mlp_out.shape[1]
    if pad_to > width:
        parts.append(bottom_mlp_out.new_zeros(bottom_mlp_out.shape[0], pad_to - width))
    return torch.cat(parts, dim=1)


def _mlp(sizes: Sequence[int], in_dim: int, final_linear: bool) -> nn.Sequential:
    layers: List[nn.Module] = []
    d = in_dim
    for i, s in enumerate(sizes):
        lin = nn.Linear(d, s)
        nn.init.xavier_normal_(lin.weight)
        nn.init.normal_(lin.bias, std=math.sqrt(1.0 / s))
        layers.append(lin)
        if not (final_linear and i == len(sizes) - 1):
            layers.append(nn.ReLU(inplace=True))
        d = s
    return nn.Sequential(*layers)


class DLRM(nn.Module):
    """DLRM with hybrid data+model parallel embeddings.

    Args:
      table_sizes: vocab size per categorical feature.
      embedding_dim: embedding width (default 128).
      bottom_mlp_dims / top_mlp_dims: hidden sizes (reference defaults).
      num_numerical: dense feature count (13 for Criteo).
      strategy: DistributedEmbedding placement strategy.
      dp_input: data-parallel input mode (see DistributedEmbedding).
    """

    def __init__(
        self,
        table_sizes: Sequence[int],
        embedding_dim: int = 128,
        bottom_mlp_dims: Sequence[int] = (512, 256, 128),
        top_mlp_dims: Sequence[int] = (1024, 1024, 512, 256, 1),
        num_numerical: int = 13,
        strategy: str = "memory_balanced",
        dp_input: bool = True,
        column_slice_threshold: Optional[int] = None,
        data_parallel_threshold: Optional[int] = None,
        table_dtype: torch.dtype = torch.float32,
    ):
        super().__init__()
        self.table_sizes = list(table_sizes)
        self.embedding_dim = embedding_dim
        self.distributed = comm.world_size() > 1
        self.dp_input = dp_input

        bottom_mlp_dims = list(bottom_mlp_dims)
        if bottom_mlp_dims[-1] != embedding_dim:
            # the interaction stacks bottom output with embeddings: dims match
            bottom_mlp_dims[-1] = embedding_dim
        self.bottom_mlp = _mlp(bottom_mlp_dims, num_numerical, final_linear=False)
        num_feats = len(table_sizes) + 1
        interact_dim = num_feats * (num_feats - 1) // 2 + bottom_mlp_dims[-1]
        # zero-pad to a multiple of 64 for MFMA-friendly GEMM K (479 -> 512)
        self.interact_pad = ((interact_dim + 63) // 64) * 64
        self.top_mlp = _mlp(top_mlp_dims, self.interact_pad, final_linear=True)

        from ..parallel.strategy import TableConfig
        tables = [
            TableConfig(s, embedding_dim, None, initializer=scaled_uniform_init)
            for s in table_sizes
        ]
        # DistributedEmbedding at every world size: at world==1 it still fuses
        # all same-width tables into one variable -> one lookup kernel + one
        # backward pipeline per step.
        self.embeddings = DistributedEmbedding(
            tables, strategy=strategy, dp_input=dp_input,
            column_slice_threshold=column_slice_threshold,
            data_parallel_threshold=data_parallel_threshold,
            table_dtype=table_dtype)

        # packed interaction fast path: the fused-group lookup output (or the
        # a2a recv buffer) feeds dot_interact as one [F-1, B, D] view; the
        # plan's worker-order permutation lives in a device buffer so output
        # columns stay in input order at every world size.
        if (self.embeddings.packed_forward_available()
                and os.environ.get("DE_PACKED", "1") != "0"):
            self.register_buffer(
                "_dot_perm",
                torch.tensor(self.embeddings.packed_order(), dtype=torch.int32),
                persistent=False)
        else:
            self._dot_perm = None
        self._emb_stream = None  # lazy side stream (world==1 overlap)

    def local_cat_feature_ids(self) -> List[int]:
        if self.distributed and not self.dp_input:
            return self.embeddings.local_input_ids()
        return list(range(len(self.table_sizes)))

    def forward(self, numerical: torch.Tensor,
                cat_features: Sequence[torch.Tensor]) -> torch.Tensor:
        cats = list(cat_features)
        handle = None
        if self.distributed and self.dp_input and \
                os.environ.get("DE_OVERLAP_A2A", "1") != "0":
            # post the dp->mp ID all-to-all before the bottom MLP: the xGMI
            # id exchange overlaps the MLP GEMMs (ids carry no grad)
            handle = self.embeddings.redistribute_async(cats)
        use_packed = self._dot_perm is not None and all(
            isinstance(x, torch.Tensor) and x.dim() == 1 for x in cats)
        want = torch.bfloat16 if (numerical.is_cuda and
                                  torch.is_autocast_enabled()) \
            else numerical.dtype
        if use_packed and self.dp_input and numerical.is_cuda and \
                os.environ.get("DE_STREAM_OVERLAP", "1") != "0":
            # The (bandwidth-bound) lookup chain — and at world>1 the mp->dp
            # output all-to-all — runs on a side stream concurrently with
            # the (MFMA-bound) bottom MLP; autograd keeps the same streams
            # in backward, so the reverse all-to-all + table update overlap
            # the MLP grads too.  hipGraph capture spans the fork (world==1).
            if self._emb_stream is None:
                self._emb_stream = torch.cuda.Stream()
            main = torch.cuda.current_stream()
            self._emb_stream.wait_stream(main)
            with torch.cuda.stream(self._emb_stream):
                packed, smaj = self.embeddings.forward_packed(
                    cats, output_dtype=want, async_handle=handle)
            bottom = self.bottom_mlp(numerical)
            main.wait_stream(self._emb_stream)
            if not torch.cuda.is_current_stream_capturing():
                packed.record_stream(main)
            x = dot_interact_packed(packed, bottom, self._dot_perm,
                                    pad_to=self.interact_pad,
                                    sample_major=smaj)
            return self.top_mlp(x)
        bottom = self.bottom_mlp(numerical)
        if use_packed:
            packed, smaj = self.embeddings.forward_packed(
                cats, output_dtype=bottom.dtype, async_handle=handle)
            x = dot_interact_packed(packed, bottom, self._dot_perm,
                                    pad_to=self.interact_pad,
                                    sample_major=smaj)
        else:
            emb = self.embeddings(cats, output_dtype=bottom.dtype,
                                  async_handle=handle)
            x = fused_dot_interact(emb, bottom, pad_to=self.interact_pad)
        return self.top_mlp(x)
