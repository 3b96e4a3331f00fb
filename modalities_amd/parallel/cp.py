"""Context parallelism (CP): sequence-sharded attention for long contexts.

The reference only plumbs a cp mesh dimension with NO compute implementation
(reference: src/modalities/running_env/fsdp/device_mesh.py:23,92-145; no ring
attention / Ulysses anywhere — SURVEY.md §5 long-context). This module
implements CP natively for the MI355X node topology: K/V are all-gathered
along the sequence over the CP group (on an 8-GPU fully-connected xGMI node
one all-gather uses all 7 links in parallel; KV bytes per layer at seq 8192 /
GQA-20 heads are ~84 MB — transfer hides behind the block's attention
compute), and each rank computes its Q chunk against the full K/V with an
OFFSET-causal mask (rank r's queries sit at global positions
[r*T_local, (r+1)*T_local)) — on device this runs the K1 HIP flash kernels,
which support Tq != Tkv and the query offset natively.

Memory: block activations, attention state and logits stay 1/cp of the full
sequence; only the transient K/V gather and the embedding output are full-T.
"""

from typing import Optional

import torch
import torch.distributed as dist

from modalities_amd.ops.attention import flash_attention
from modalities_amd.parallel.tp import _GatherSeq


class _SliceSeqPartial(torch.autograd.Function):
    """Replicated -> seq-sharded with PARTIAL backward: grad is zero outside
    this rank's chunk (no communication). This keeps every parameter's CP
    gradient partial — summed across the CP group they equal the full-seq
    gradient, uniformly for embeddings and block weights — so the CP grad
    sync is one all-reduce(SUM) over the cp group (`cp_grad_allreduce_`).

    (TP's _SliceSeq all-gathers instead, because TP replicates the
    downstream compute; CP does not.)"""

    @staticmethod
    def forward(ctx, x, group, dim):
        world = dist.get_world_size(group)
        rank = dist.get_rank(group)
        ctx.dim, ctx.world, ctx.rank = dim, world, rank
        ctx.full_size = x.shape[dim]
        return x.chunk(world, dim=dim)[rank].contiguous()

    @staticmethod
    def backward(ctx, grad):
        shape = list(grad.shape)
        shape[ctx.dim] = ctx.full_size
        full = grad.new_zeros(shape)
        chunk = ctx.full_size // ctx.world
        idx = [slice(None)] * grad.dim()
        idx[ctx.dim] = slice(ctx.rank * chunk, (ctx.rank + 1) * chunk)
        full[tuple(idx)] = grad
        return full, None, None


def cp_attention(q_local: torch.Tensor, k_local: torch.Tensor,
                 v_local: torch.Tensor, group, cp_rank: int, cp_size: int
                 ) -> torch.Tensor:
    """q/k/v: [B, T_local, H, D] (this rank's seq chunk). Returns the local
    attention output [B, T_local, Hq, D]. Backward reduce-scatters dK/dV."""
    if cp_size == 1:
        return flash_attention(q_local, k_local, v_local, causal=True)
    k_full = _GatherSeq.apply(k_local, group, 1)
    v_full = _GatherSeq.apply(v_local, group, 1)
    q_offset = cp_rank * q_local.shape[1]
    return flash_attention(q_local, k_full, v_full, causal=True,
                           q_offset=q_offset)


class _AllToAllSeqHead(torch.autograd.Function):
    """Ulysses exchange: [B, T/cp, H, D] (seq-sharded) <-> [B, T, H/cp, D]
    (head-sharded). One all-to-all over the CP group each direction; on a
    fully-connected 8-GPU xGMI node the 7 pairwise transfers run in
    parallel. Backward is the inverse exchange. gloo (CPU tests) has no
    all_to_all: emulated with all_gather + local slice."""

    @staticmethod
    def _exchange(x, group, to_heads: bool):
        world = dist.get_world_size(group)
        rank = dist.get_rank(group)
        B, T_in, H_in, D = x.shape
        if to_heads:
            # split heads into `world` groups; send seq-chunk i of head-group j
            send = [x[:, :, j * (H_in // world):(j + 1) * (H_in // world)]
                    .contiguous() for j in range(world)]
        else:
            # split seq back into chunks
            send = [c.contiguous() for c in x.chunk(world, dim=1)]
        recv = [torch.empty_like(send[0]) for _ in range(world)]
        backend = dist.get_backend(group) if group is not None else dist.get_backend()
        if backend == "gloo":
            for j in range(world):
                gathered = [torch.empty_like(send[j]) for _ in range(world)]
                dist.all_gather(gathered, send[j], group=group)
                if j == rank:
                    recv = gathered
        else:
            dist.all_to_all(recv, send, group=group)
        return torch.cat(recv, dim=1 if to_heads else 2)

    @staticmethod
    def forward(ctx, x, group, to_heads):
        ctx.group, ctx.to_heads = group, to_heads
        return _AllToAllSeqHead._exchange(x, group, to_heads)

    @staticmethod
    def backward(ctx, grad):
        return (_AllToAllSeqHead._exchange(grad.contiguous(), ctx.group,
                                           not ctx.to_heads), None, None)


def cp_attention_ulysses(q_local, k_local, v_local, group, cp_rank: int,
                         cp_size: int) -> torch.Tensor:
    """Ulysses-style CP: all-to-all scatters HEADS and gathers the full
    sequence, each rank runs plain causal attention on H/cp heads, then the
    inverse all-to-all restores seq sharding. Requires Hq % cp == 0 and
    Hkv % cp == 0 (GQA kv heads must still split)."""
    if cp_size == 1:
        return flash_attention(q_local, k_local, v_local, causal=True)
    Hq, Hkv = q_local.shape[2], k_local.shape[2]
    if Hq % cp_size or Hkv % cp_size:
        raise ValueError(f"Ulysses CP needs Hq ({Hq}) and Hkv ({Hkv}) "
                         f"divisible by cp={cp_size}")
    qh = _AllToAllSeqHead.apply(q_local, group, True)   # [B, T, Hq/cp, D]
    kh = _AllToAllSeqHead.apply(k_local, group, True)
    vh = _AllToAllSeqHead.apply(v_local, group, True)
    oh = flash_attention(qh, kh, vh, causal=True)
    return _AllToAllSeqHead.apply(oh, group, False)     # [B, T/cp, Hq, D]


class _CPAttentionForward:
    """Replacement forward for CausalSelfAttention under CP: x is this
    rank's seq chunk; rope tables are sliced at the rank's global offset."""

    def __init__(self, attn, group, cp_rank: int, cp_size: int,
                 variant: str = "allgather"):
        self.attn = attn
        self.group = group
        self.cp_rank = cp_rank
        self.cp_size = cp_size
        self.variant = variant

    def __call__(self, x: torch.Tensor, rope_cos, rope_sin) -> torch.Tensor:
        attn = self.attn
        B, Tl, C = x.shape
        off = self.cp_rank * Tl
        kv_dim = attn.head_dim * attn.n_head_kv
        if getattr(attn, "fused_qkv", False):
            qkv = attn.qkv_attn(x)
            q, k, v = qkv.split([C, kv_dim, kv_dim], dim=-1)
            q = q.view(B, Tl, attn.n_head_q, attn.head_dim)
            k = k.view(B, Tl, attn.n_head_kv, attn.head_dim)
            v = v.view(B, Tl, attn.n_head_kv, attn.head_dim)
        else:
            q = attn.q_attn(x).view(B, Tl, attn.n_head_q, attn.head_dim)
            k = attn.k_attn(x).view(B, Tl, attn.n_head_kv, attn.head_dim)
            v = attn.v_attn(x).view(B, Tl, attn.n_head_kv, attn.head_dim)
        if attn.q_norm is not None:
            q = attn.q_norm(q)
            k = attn.k_norm(k)
        if rope_cos is not None:
            from modalities_amd.ops import rope_apply
            q = rope_apply(q, rope_cos[off:off + Tl], rope_sin[off:off + Tl])
            k = rope_apply(k, rope_cos[off:off + Tl], rope_sin[off:off + Tl])
        if self.variant == "ulysses":
            y = cp_attention_ulysses(q, k, v, self.group, self.cp_rank,
                                     self.cp_size)
        else:
            y = cp_attention(q, k, v, self.group, self.cp_rank, self.cp_size)
        y = y.reshape(B, Tl, attn.n_head_q * attn.head_dim)
        return attn.resid_dropout(attn.c_proj(y))


def get_gpt2_context_parallel_model(model, device_mesh=None, group=None,
                                    cp_rank: Optional[int] = None,
                                    cp_size: Optional[int] = None,
                                    variant: str = "allgather"):
    """Patch a GPT2LLM for CP: the model still takes FULL input_ids; the
    residual stream is seq-sliced after the embedding dropout; every
    attention runs offset-causal against all-gathered K/V; logits come back
    seq-sharded [B, T/cp, V] — slice the targets identically
    (`slice_targets_for_cp`)."""
    if device_mesh is not None:
        from modalities_amd.parallel.mesh import ParallelismDegrees
        dim = device_mesh.dims[ParallelismDegrees.CP]
        group, cp_rank, cp_size = dim.group, dim.rank, dim.size
    if cp_size in (None, 1):
        return model
    if variant not in ("allgather", "ulysses"):
        raise ValueError(f"Unknown CP variant {variant!r}")
    for block in model.blocks:
        block.attn.forward = _CPAttentionForward(block.attn, group, cp_rank,
                                                 cp_size, variant=variant)
    model.drop.register_forward_hook(
        lambda mod, args, out: _SliceSeqPartial.apply(out, group, 1))
    model._cp_info = (group, cp_rank, cp_size)
    return model


def slice_targets_for_cp(targets: torch.Tensor, cp_rank: int, cp_size: int
                         ) -> torch.Tensor:
    return targets.chunk(cp_size, dim=1)[cp_rank]


@torch.no_grad()
def cp_grad_allreduce_(model, group) -> None:
    """Sum partial CP gradients across the cp group (call after backward,
    before the optimizer / DP reduce)."""
    for p in model.parameters():
        if p.grad is not None:
            dist.all_reduce(p.grad, group=group)
