"""Model-agnostic pipeline splitting (capability parity with the
reference's FQN-tree splitter + weight-balanced stages generator,
reference: src/modalities/models/parallelism/pipeline_parallelism.py:131-277
and stages_generator.py:15-120).

The reference splits a traced module graph by FQN lists. The MI355X-native
equivalent is a SEGMENT protocol: a model exposes an ordered list of
(fqn, nn.Module) segments whose composition over a single boundary tensor
is its forward. Any model implementing ``pipeline_segments()`` — or any
``nn.Sequential`` — can then be split into weight-balanced stages by REAL
parameter counts; each rank registers only its kept segments (the rest are
droppable), and the stage composes them.

GPT2LLM's segments are provided here (embedding / per-block / head);
``split_model_into_stages`` in pp.py remains the GPT2-specialized fast
path and produces equivalent stages."""

from typing import Callable, Optional

import torch
import torch.nn as nn


# ---------------------------------------------------------------------------
# segment wrappers for GPT2-shaped models
# ---------------------------------------------------------------------------

class EmbedSegment(nn.Module):
    def __init__(self, wte, wpe, drop):
        super().__init__()
        self.wte = wte
        self.wpe = wpe
        self.drop = drop

    def forward(self, input_ids: torch.Tensor) -> torch.Tensor:
        x = self.wte(input_ids)
        if self.wpe is not None:
            pos = torch.arange(input_ids.shape[1], dtype=torch.long,
                               device=input_ids.device)
            x = x + self.wpe(pos)
        return self.drop(x)


class BlockSegment(nn.Module):
    """One transformer block with its own (tiny) RoPE cache so the segment
    is self-contained (no reference back to the full model)."""

    def __init__(self, block, rope_theta: Optional[float], head_dim: int):
        super().__init__()
        self.block = block
        self.rope_theta = rope_theta
        self.head_dim = head_dim
        self._rope_cache = None

    def _rope(self, T: int, device):
        if self.rope_theta is None:
            return None, None
        if self._rope_cache is None or self._rope_cache[0].shape[0] < T \
                or self._rope_cache[0].device != device:
            from modalities_amd.ops import precompute_rope_cos_sin
            cos, sin = precompute_rope_cos_sin(T, self.head_dim,
                                               self.rope_theta, device=device)
            self._rope_cache = (cos, sin)
        cos, sin = self._rope_cache
        return cos[:T], sin[:T]

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        cos, sin = self._rope(x.shape[1], x.device)
        return self.block(x, cos, sin)


class HeadSegment(nn.Module):
    def __init__(self, norm, lm_head):
        super().__init__()
        self.norm = norm
        self.lm_head = lm_head

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        return self.lm_head(self.norm(x))


def gpt2_pipeline_segments(model) -> list[tuple[str, nn.Module]]:
    """Ordered single-tensor-boundary segments of a GPT2LLM."""
    from modalities_amd.models.gpt2 import QueryKeyValueTransformType
    cfg = model.config
    theta = cfg.rope_base \
        if cfg.qkv_transform == QueryKeyValueTransformType.ROTARY else None
    segs: list[tuple[str, nn.Module]] = [
        ("embed", EmbedSegment(model.wte, model.wpe, model.drop))]
    head_dim = cfg.n_embd // cfg.n_head_q
    for i, block in enumerate(model.blocks):
        segs.append((f"blocks.{i}", BlockSegment(block, theta, head_dim)))
    segs.append(("head", HeadSegment(model.lm_head_norm, model.lm_head)))
    return segs


# ---------------------------------------------------------------------------
# generic splitting
# ---------------------------------------------------------------------------

def collect_pipeline_segments(model) -> list[tuple[str, nn.Module]]:
    if hasattr(model, "pipeline_segments"):
        return model.pipeline_segments()
    from modalities_amd.models.gpt2 import GPT2LLM
    if isinstance(model, GPT2LLM):
        return gpt2_pipeline_segments(model)
    if isinstance(model, nn.Sequential):
        return [(name, child) for name, child in model.named_children()]
    raise TypeError(
        f"{type(model).__name__} exposes no pipeline segmentation: implement "
        "pipeline_segments() -> [(fqn, module)] with single-tensor "
        "boundaries, or pass an nn.Sequential")


def balanced_segment_partition(weights: list[int], pp_size: int) -> list[list[int]]:
    """Contiguous partition of segment indices minimizing imbalance of the
    summed weights (greedy threshold sweep; weights = REAL param counts,
    reference stages_generator.py balances block counts + equivalence
    weights)."""
    n = len(weights)
    if n < pp_size:
        raise ValueError(f"{n} segments cannot fill {pp_size} stages")
    total = sum(weights)
    target = total / pp_size
    parts: list[list[int]] = []
    i = 0
    acc = 0.0
    for s in range(pp_size):
        remaining_stages = pp_size - s - 1
        part = []
        run = 0
        # take segments while below target, always leaving enough for the
        # remaining stages
        while i < n - remaining_stages and (not part or run + weights[i] / 2 <= target):
            part.append(i)
            run += weights[i]
            i += 1
            if run >= target and remaining_stages > 0:
                break
        parts.append(part)
        acc += run
        if remaining_stages:
            target = (total - acc) / remaining_stages
    # any residue goes to the last stage
    while i < n:
        parts[-1].append(i)
        i += 1
    return parts


class SegmentedPipelineStage(nn.Module):
    """A pipeline stage composed of contiguous model segments. Registers
    only the kept segments (param iteration / engine sharding sees exactly
    this stage's weights); forward folds the segment calls over the single
    boundary tensor."""

    def __init__(self, model, stage_idx: int, num_stages: int,
                 segments: list[tuple[str, nn.Module]],
                 keep: list[int]):
        super().__init__()
        self.stage_idx = stage_idx
        self.num_stages = num_stages
        self.is_first = stage_idx == 0
        self.is_last = stage_idx == num_stages - 1
        self.config = getattr(model, "config", None)
        self.fqns = [segments[i][0] for i in keep]
        self.segs = nn.ModuleList([segments[i][1] for i in keep])

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        for seg in self.segs:
            x = seg(x)
        return x


def split_model_into_stages_by_fqn(model, pp_size: int,
                                   stage_fqns: Optional[list[list[str]]] = None
                                   ) -> list[SegmentedPipelineStage]:
    """Split ANY segmentable model into pp_size stages.

    stage_fqns: explicit per-stage segment-FQN lists (the reference's
    user-facing contract); None = weight-balanced automatic assignment by
    real parameter counts."""
    segments = collect_pipeline_segments(model)
    by_fqn = {fqn: i for i, (fqn, _) in enumerate(segments)}
    if stage_fqns is not None:
        if len(stage_fqns) != pp_size:
            raise ValueError(f"stage_fqns has {len(stage_fqns)} entries for "
                             f"pp={pp_size}")
        parts = []
        for fqns in stage_fqns:
            try:
                parts.append(sorted(by_fqn[f] for f in fqns))
            except KeyError as e:
                raise KeyError(f"Unknown segment FQN {e.args[0]!r}; known: "
                               f"{sorted(by_fqn)}") from None
    else:
        weights = [max(1, sum(p.numel() for p in m.parameters()))
                   for _, m in segments]
        parts = balanced_segment_partition(weights, pp_size)
    return [SegmentedPipelineStage(model, s, pp_size, segments, parts[s])
            for s in range(pp_size)]
