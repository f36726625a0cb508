"""Fused DLRM pairwise-dot interaction op (MFMA bf16, gfx950).

Replaces the stack + bmm + tril-mask + concat chain of the reference
(``examples/dlrm/utils.py:92-113``) with one MFMA kernel each direction
(``csrc/dot_interact.hip``).  Falls back to plain torch ops off-GPU or for
unsupported shapes (F > 32 or D % 32 != 0).
"""

from typing import List

import torch

from . import _backend


class _DotInteract(torch.autograd.Function):
    @staticmethod
    def forward(ctx, feats, out_w):
        ctx.save_for_backward(feats)
        return _backend.ops().dot_interact_fwd(feats, out_w)

    @staticmethod
    def backward(ctx, gout):
        (feats,) = ctx.saved_tensors
        gfeats = _backend.ops().dot_interact_bwd(gout.contiguous(), feats)
        return gfeats, None


def _torch_dot_interact(feats: torch.Tensor, pad_to: int) -> torch.Tensor:
    bottom = feats[:, 0, :]
    gram = torch.bmm(feats, feats.transpose(1, 2))
    f = gram.shape[1]
    ii, jj = torch.tril_indices(f, f, offset=-1, device=gram.device)
    interactions = gram[:, ii, jj]
    parts = [interactions, bottom]
    width = interactions.shape[1] + bottom.shape[1]
    if pad_to > width:
        parts.append(bottom.new_zeros(bottom.shape[0], pad_to - width))
    return torch.cat(parts, dim=1)


class _DotInteractPacked(torch.autograd.Function):
    @staticmethod
    def forward(ctx, bottom, packed, perm, out_w, sample_major):
        ctx.save_for_backward(bottom, packed, perm)
        ctx.sample_major = sample_major
        return _backend.ops().dot_interact_fwd_packed(bottom, packed, perm,
                                                      out_w, sample_major)

    @staticmethod
    def backward(ctx, gout):
        bottom, packed, perm = ctx.saved_tensors
        gbottom, gpacked = _backend.ops().dot_interact_bwd_packed(
            gout.contiguous(), bottom, packed, perm, ctx.sample_major)
        return gbottom, gpacked, None, None, None


def dot_interact_packed(packed: torch.Tensor, bottom_mlp_out: torch.Tensor,
                        perm: torch.Tensor, pad_to: int = 0,
                        sample_major: bool = False) -> torch.Tensor:
    """Interaction over packed embeddings (zero-copy input).

    ``packed``: [P, B, D] feature-major (the mp->dp all-to-all recv buffer,
    world>1) or [B, P, D] sample-major (the world==1 fused lookup output) —
    see ``DistributedEmbedding.forward_packed``.  ``perm[f-1]`` maps model
    feature f to its packed row, so output columns are in model input order
    at every world size.
    """
    if sample_major:
        b, p, d = packed.shape
    else:
        p, b, d = packed.shape
    f = p + 1
    width = f * (f - 1) // 2 + d
    out_w = max(pad_to, width)
    if (packed.is_cuda and packed.dtype == torch.bfloat16
            and bottom_mlp_out.dtype == torch.bfloat16 and f <= 32
            and d % 32 == 0):
        return _DotInteractPacked.apply(bottom_mlp_out.contiguous(),
                                        packed.contiguous(), perm, out_w,
                                        sample_major)
    # fallback (CPU / unsupported shapes): materialize [B, F, D] and reuse
    # the plain path
    sel = packed.index_select(1, perm.long()) if sample_major else \
        packed.index_select(0, perm.long()).transpose(0, 1)
    feats = torch.cat([bottom_mlp_out.unsqueeze(1), sel], dim=1)
    return _torch_dot_interact(feats.contiguous(), out_w)


def dot_interact(emb_outs: List[torch.Tensor], bottom_mlp_out: torch.Tensor,
                 pad_to: int = 0) -> torch.Tensor:
    """[tril(feats @ feats^T) | bottom | 0-pad], feats = [bottom] + emb_outs."""
    feats = torch.stack([bottom_mlp_out] + emb_outs, dim=1)
    f, d = feats.shape[1], feats.shape[2]
    width = f * (f - 1) // 2 + d
    out_w = max(pad_to, width)
    if (feats.is_cuda and feats.dtype == torch.bfloat16 and f <= 32
            and d % 32 == 0):
        return _DotInteract.apply(feats.contiguous(), out_w)
    return _torch_dot_interact(feats, out_w)
