"""Pipeline parallelism (PP) for MI355X: stage splitting + p2p activation
transport + schedule state machines.

Capability parity with the reference's PipelineFactory over
torch.distributed.pipelining (reference:
src/modalities/models/parallelism/pipeline_parallelism.py:91-338,
stages_generator.py:15-120 — FQN-tree splitting, weight-balanced stage
assignment, GPipe/1F1B schedules) — but the schedules are implemented
directly as explicit send/recv state machines over the PP process group
(RCCL p2p over xGMI on device; gloo in CPU tests), not via
torch.distributed.pipelining.

Stage layout for a GPT2-style model:
  stage 0:        wte (+wpe, dropout) + blocks[0:k1]
  stage i:        blocks[ki:ki+1]
  last stage:     blocks[...:L] + lm_head_norm + lm_head
Block counts are assigned by weight balancing with configurable equivalence
weights for the embedding/head (reference stages_generator.py:15-120).
"""

from typing import Callable, Optional

import torch
import torch.distributed as dist
import torch.nn as nn


# ---------------------------------------------------------------------------
# stage assignment
# ---------------------------------------------------------------------------

def balanced_stage_assignment(num_blocks: int, pp_size: int,
                              input_weight: float = 1.0,
                              output_weight: float = 1.0) -> list[int]:
    """Return blocks-per-stage so that (blocks + embedding/head equivalence
    weights) are as equal as possible. input_weight/output_weight express the
    embedding/head cost in units of one transformer block (reference
    stages_generator.py:15-120)."""
    if num_blocks < pp_size:
        raise ValueError(f"{num_blocks} blocks cannot fill {pp_size} stages")
    total = num_blocks + input_weight + output_weight
    target = total / pp_size
    counts = []
    remaining = num_blocks
    for s in range(pp_size):
        want = target
        if s == 0:
            want -= input_weight
        if s == pp_size - 1:
            want -= output_weight
        take = max(1, round(want))
        take = min(take, remaining - (pp_size - 1 - s))  # leave >=1 per stage
        counts.append(take)
        remaining -= take
    counts[-1] += remaining
    return counts


# ---------------------------------------------------------------------------
# stage modules
# ---------------------------------------------------------------------------

class GPT2PipelineStage(nn.Module):
    """One pipeline stage of a GPT2LLM. Holds references to the original
    model's submodules (the rest are dropped so each rank only materializes
    its stage)."""

    def __init__(self, full_model, stage_idx: int, blocks_per_stage: list[int]):
        super().__init__()
        self.stage_idx = stage_idx
        self.num_stages = len(blocks_per_stage)
        self.is_first = stage_idx == 0
        self.is_last = stage_idx == self.num_stages - 1
        self.config = full_model.config
        self.sample_key = full_model.sample_key
        self.prediction_key = full_model.prediction_key

        start = sum(blocks_per_stage[:stage_idx])
        end = start + blocks_per_stage[stage_idx]
        self.blocks = nn.ModuleList(list(full_model.blocks)[start:end])
        if self.is_first:
            self.wte = full_model.wte
            self.wpe = full_model.wpe
            self.drop = full_model.drop
        if self.is_last:
            self.lm_head_norm = full_model.lm_head_norm
            self.lm_head = full_model.lm_head
        self._rope = full_model._rope.__func__.__get__(self)  # reuse cache fn
        self._rope_cache = None

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        """First stage: x = input_ids [B,T] -> hidden. Middle: hidden ->
        hidden. Last: hidden -> logits."""
        if self.is_first:
            input_ids = x
            B, T = input_ids.shape
            x = self.wte(input_ids)
            if self.wpe is not None:
                pos = torch.arange(T, dtype=torch.long, device=input_ids.device)
                x = x + self.wpe(pos)
            x = self.drop(x)
        T = x.shape[1]
        rope_cos, rope_sin = self._rope(T, x.device)
        for block in self.blocks:
            x = block(x, rope_cos, rope_sin)
        if self.is_last:
            x = self.lm_head_norm(x)
            x = self.lm_head(x)
        return x


def split_model_into_stages(model, pp_size: int, input_weight: float = 1.0,
                            output_weight: float = 1.0) -> list[GPT2PipelineStage]:
    counts = balanced_stage_assignment(len(model.blocks), pp_size,
                                       input_weight, output_weight)
    return [GPT2PipelineStage(model, s, counts) for s in range(pp_size)]


# ---------------------------------------------------------------------------
# schedules
# ---------------------------------------------------------------------------

class PipelineSchedule:
    """Base: microbatch split + p2p plumbing. Subclasses implement step().

    loss_fn is called as loss_fn((logits, targets)) on the LAST stage
    (reference: the dual call signature of CLMCrossEntropyLoss)."""

    def __init__(self, stage: nn.Module, stage_idx: int, num_stages: int,
                 n_microbatches: int, group=None, device=None,
                 activation_shape_fn: Optional[Callable] = None,
                 sharded_engine=None):
        self.stage = stage
        self.stage_idx = stage_idx
        self.num_stages = num_stages
        self.is_first = stage_idx == 0
        self.is_last = stage_idx == num_stages - 1
        self.n_microbatches = n_microbatches
        self.group = group
        self.device = device or torch.device("cpu")
        self.activation_shape_fn = activation_shape_fn
        self.sharded_engine = sharded_engine  # XGMIShardedModel wrap, if any
        self._pending: list = []
        # global ranks of prev/next stage within the PP group
        ranks = dist.get_process_group_ranks(group) if group is not None else \
            list(range(dist.get_world_size())) if dist.is_initialized() else [0]
        self._prev_rank = ranks[stage_idx - 1] if stage_idx > 0 else None
        self._next_rank = ranks[stage_idx + 1] if stage_idx < num_stages - 1 else None

    # -- p2p helpers ------------------------------------------------------
    # Sends are non-blocking (isend): a fully blocking send deadlocks 1F1B's
    # steady state (stage i blocked sending activation i+1 while stage i+1
    # is blocked sending a gradient back). Pending works are drained at the
    # end of step().
    def _send(self, tensor: torch.Tensor, dst: int):
        t = tensor.detach().contiguous()
        work = dist.isend(t, dst=dst, group=self.group)
        self._pending.append((work, t))

    def _drain_sends(self):
        for work, _ in self._pending:
            work.wait()
        self._pending = []

    def _recv(self, shape, dtype) -> torch.Tensor:
        buf = torch.empty(shape, dtype=dtype, device=self.device)
        dist.recv(buf, src=self._prev_rank if dtype != torch.long else
                  self._prev_rank, group=self.group)
        return buf

    def _act_shape(self, mb_size: int, seq_len: int):
        if self.activation_shape_fn is not None:
            return self.activation_shape_fn(mb_size, seq_len)
        return (mb_size, seq_len, self.stage.config.n_embd)

    def _act_dtype(self):
        p = next(self.stage.parameters())
        return p.dtype

    # -- per-microbatch fwd/bwd ------------------------------------------
    def _forward_mb(self, mb_input, mb_target, losses_out):
        """Returns (stage_input, stage_output, loss or None)."""
        if self.is_first:
            x = mb_input
        else:
            x = self._recv(self._act_shape(mb_input.shape[0], mb_input.shape[1]),
                           self._act_dtype())
            x.requires_grad_(True)
        out = self.stage(x) if self.sharded_engine is None \
            else self.sharded_engine(x)
        loss = None
        if self.is_last:
            if self.loss_fn is not None:
                loss = self.loss_fn((out, mb_target))
                losses_out.append(loss.detach())
        else:
            self._send(out, self._next_rank)
        return x, out, loss

    def _backward_mb(self, x, out, loss):
        if self.is_last:
            (loss / self.n_microbatches).backward()
        else:
            dgrad = torch.empty_like(out)
            dist.recv(dgrad, src=self._next_rank, group=self.group)
            torch.autograd.backward(out, grad_tensors=dgrad)
        if self.sharded_engine is not None:
            self.sharded_engine.backward_epilogue()
        if not self.is_first:
            self._send(x.grad, self._prev_rank)

    def step(self, inputs: torch.Tensor, targets: Optional[torch.Tensor],
             loss_fn: Optional[Callable], losses_out: Optional[list] = None):
        raise NotImplementedError

    @torch.no_grad()
    def eval_step(self, inputs: torch.Tensor, targets: Optional[torch.Tensor],
                  loss_fn: Optional[Callable],
                  losses_out: Optional[list] = None) -> list:
        """Forward-only schedule pass for evaluation (reference
        evaluator.py:88-180 dispatches the PP schedule in eval): every
        microbatch flows through the stages, the last stage computes the
        loss, no backward and no gradient traffic."""
        self.loss_fn = loss_fn
        losses_out = losses_out if losses_out is not None else []
        mb_inputs = inputs.chunk(self.n_microbatches, dim=0)
        mb_targets = targets.chunk(self.n_microbatches, dim=0) \
            if targets is not None else [None] * self.n_microbatches
        for mb_x, mb_y in zip(mb_inputs, mb_targets):
            self._forward_mb(mb_x, mb_y, losses_out)
        self._drain_sends()
        return losses_out

    def broadcast_mean_loss(self, losses: list) -> torch.Tensor:
        """Mean micro-batch loss, broadcast from the last stage so every PP
        rank logs the true value (the trainer's loss reduction is then
        PP-agnostic)."""
        ranks = dist.get_process_group_ranks(self.group) if self.group is not None \
            else list(range(dist.get_world_size()))
        src = ranks[self.num_stages - 1]
        if self.is_last and losses:
            val = torch.stack([l.detach().float().cpu() for l in losses]).mean()
        else:
            val = torch.zeros((), dtype=torch.float32)
        buf = val.to(self.device) if self.device.type == "cuda" else val
        dist.broadcast(buf, src=src, group=self.group)
        return buf.cpu()


class ScheduleGPipe(PipelineSchedule):
    """All forwards, then all backwards (reference maps
    get_schedule_class('GPipe'))."""

    def step(self, inputs, targets, loss_fn, losses_out=None):
        self.loss_fn = loss_fn
        losses_out = losses_out if losses_out is not None else []
        mb_inputs = inputs.chunk(self.n_microbatches, dim=0)
        mb_targets = targets.chunk(self.n_microbatches, dim=0) \
            if targets is not None else [None] * self.n_microbatches
        saved = []
        for mb_x, mb_y in zip(mb_inputs, mb_targets):
            saved.append(self._forward_mb(mb_x, mb_y, losses_out))
        for x, out, loss in saved:
            self._backward_mb(x, out, loss)
        self._drain_sends()
        return losses_out


class Schedule1F1B(PipelineSchedule):
    """One-forward-one-backward steady state: warmup of
    (num_stages - stage_idx - 1) forwards, then alternate, then cooldown
    (reference maps get_schedule_class('1F1B'))."""

    def step(self, inputs, targets, loss_fn, losses_out=None):
        self.loss_fn = loss_fn
        losses_out = losses_out if losses_out is not None else []
        mb_inputs = list(inputs.chunk(self.n_microbatches, dim=0))
        mb_targets = list(targets.chunk(self.n_microbatches, dim=0)) \
            if targets is not None else [None] * self.n_microbatches

        n_warmup = min(self.num_stages - self.stage_idx - 1, self.n_microbatches)
        saved = []
        fwd_i = 0
        for _ in range(n_warmup):
            saved.append(self._forward_mb(mb_inputs[fwd_i], mb_targets[fwd_i],
                                          losses_out))
            fwd_i += 1
        n_steady = self.n_microbatches - n_warmup
        for _ in range(n_steady):
            saved.append(self._forward_mb(mb_inputs[fwd_i], mb_targets[fwd_i],
                                          losses_out))
            fwd_i += 1
            x, out, loss = saved.pop(0)
            self._backward_mb(x, out, loss)
        for x, out, loss in saved:
            self._backward_mb(x, out, loss)
        self._drain_sends()
        return losses_out


class ScheduleInterleaved1F1B:
    """Interleaved 1F1B over VIRTUAL stages (reference exposes torch
    pipelining's Interleaved1F1B, pipeline_parallelism.py:14-20): the model
    is split into pp_size * num_chunks stages; rank r holds chunks
    c = 0..v-1 as global stages c*pp + r ("loop" placement). Warmup depth
    (pp - r - 1)*2 + (v-1)*pp forwards, then 1F1B steady state with the
    Megatron chunk rotation — smaller bubble than 1F1B at equal
    microbatches ((pp-1)/(m*v) vs (pp-1)/m).

    All sends are isend (ordering per channel matches both endpoints'
    schedules); activations flow r -> r+1 within a chunk and wrap
    pp-1 -> 0 between chunks; gradients mirror."""

    def __init__(self, stages: list, pp_rank: int, pp_size: int,
                 n_microbatches: int, group=None, device=None,
                 activation_shape_fn: Optional[Callable] = None,
                 sharded_engines: Optional[list] = None):
        if n_microbatches % pp_size:
            raise ValueError(
                f"interleaved 1F1B needs n_microbatches ({n_microbatches}) "
                f"divisible by pp_size ({pp_size})")
        self.stages = stages                      # this rank's chunks
        self.v = len(stages)
        self.pp_rank = pp_rank
        self.pp_size = pp_size
        self.num_stages = pp_size                 # for broadcast_mean_loss
        self.n_microbatches = n_microbatches
        self.group = group
        self.device = device or torch.device("cpu")
        self.activation_shape_fn = activation_shape_fn
        self.sharded_engines = sharded_engines
        self.is_first = pp_rank == 0              # owns global stage 0
        self.is_last = pp_rank == pp_size - 1     # owns the final stage
        self._pending: list = []
        ranks = dist.get_process_group_ranks(group) if group is not None else             list(range(dist.get_world_size())) if dist.is_initialized() else [0]
        self._ranks = ranks
        self._prev_rank = ranks[pp_rank - 1] if pp_rank > 0 else ranks[-1]
        self._next_rank = ranks[pp_rank + 1] if pp_rank < pp_size - 1 else ranks[0]

    # channel helpers --------------------------------------------------------
    def _send(self, tensor, dst):
        t = tensor.detach().contiguous()
        work = dist.isend(t, dst=dst, group=self.group)
        self._pending.append((work, t))

    def _drain_sends(self):
        for work, _ in self._pending:
            work.wait()
        self._pending = []

    def _recv_from(self, src, shape, dtype):
        buf = torch.empty(shape, dtype=dtype, device=self.device)
        dist.recv(buf, src=src, group=self.group)
        return buf

    def _act_shape(self, mb_size, seq_len):
        if self.activation_shape_fn is not None:
            return self.activation_shape_fn(mb_size, seq_len)
        return (mb_size, seq_len, self.stages[0].config.n_embd)

    def _act_dtype(self):
        return next(self.stages[0].parameters()).dtype

    # stage-boundary roles ---------------------------------------------------
    def _chunk_is_first(self, c):   # global stage c*pp + r == 0
        return c == 0 and self.pp_rank == 0

    def _chunk_is_last(self, c):
        return c == self.v - 1 and self.pp_rank == self.pp_size - 1

    def _fwd_chunk_at(self, k):
        return (k // self.pp_size) % self.v

    def _bwd_chunk_at(self, k):
        return self.v - 1 - ((k // self.pp_size) % self.v)

    # one fwd / bwd op -------------------------------------------------------
    def _forward_op(self, c, mb_idx, mb_inputs, mb_targets, losses_out,
                    saved):
        if self._chunk_is_first(c):
            x = mb_inputs[mb_idx]
        else:
            mb = mb_inputs[mb_idx]
            x = self._recv_from(self._prev_rank,
                                self._act_shape(mb.shape[0], mb.shape[1]),
                                self._act_dtype())
            x.requires_grad_(True)
        eng = self.sharded_engines[c] if self.sharded_engines else None
        out = eng(x) if eng is not None else self.stages[c](x)
        loss = None
        if self._chunk_is_last(c):
            if self.loss_fn is not None:
                loss = self.loss_fn((out, mb_targets[mb_idx]))
                losses_out.append(loss.detach())
        else:
            self._send(out, self._next_rank)
        saved[c].append((x, out, loss))

    def _backward_op(self, c, saved):
        x, out, loss = saved[c].pop(0)
        if self._chunk_is_last(c):
            (loss / self.n_microbatches).backward()
        else:
            dgrad = torch.empty_like(out)
            dist.recv(dgrad, src=self._next_rank, group=self.group)
            torch.autograd.backward(out, grad_tensors=dgrad)
        if self.sharded_engines:
            self.sharded_engines[c].backward_epilogue()
        if not self._chunk_is_first(c):
            self._send(x.grad, self._prev_rank)

    # the schedule -----------------------------------------------------------
    def step(self, inputs, targets, loss_fn, losses_out=None):
        self.loss_fn = loss_fn
        losses_out = losses_out if losses_out is not None else []
        mb_inputs = list(inputs.chunk(self.n_microbatches, dim=0))
        mb_targets = list(targets.chunk(self.n_microbatches, dim=0)) \
            if targets is not None else [None] * self.n_microbatches

        total = self.n_microbatches * self.v
        n_warmup = min(total, (self.pp_size - self.pp_rank - 1) * 2
                       + (self.v - 1) * self.pp_size)
        saved = [[] for _ in range(self.v)]
        fwd_count = [0] * self.v   # per-chunk microbatch cursor
        kf = kb = 0

        def do_fwd():
            nonlocal kf
            c = self._fwd_chunk_at(kf)
            self._forward_op(c, fwd_count[c], mb_inputs, mb_targets,
                             losses_out, saved)
            fwd_count[c] += 1
            kf += 1

        def do_bwd():
            nonlocal kb
            self._backward_op(self._bwd_chunk_at(kb), saved)
            kb += 1

        for _ in range(n_warmup):
            do_fwd()
        for _ in range(total - n_warmup):   # steady 1F1B
            do_fwd()
            do_bwd()
        while kb < total:                   # cooldown
            do_bwd()
        self._drain_sends()
        return losses_out

    @torch.no_grad()
    def eval_step(self, inputs, targets, loss_fn, losses_out=None):
        self.loss_fn = loss_fn
        losses_out = losses_out if losses_out is not None else []
        mb_inputs = list(inputs.chunk(self.n_microbatches, dim=0))
        mb_targets = list(targets.chunk(self.n_microbatches, dim=0)) \
            if targets is not None else [None] * self.n_microbatches
        saved = [[] for _ in range(self.v)]
        fwd_count = [0] * self.v
        for k in range(self.n_microbatches * self.v):
            c = self._fwd_chunk_at(k)
            self._forward_op(c, fwd_count[c], mb_inputs, mb_targets,
                             losses_out, saved)
            fwd_count[c] += 1
        self._drain_sends()
        return losses_out

    def broadcast_mean_loss(self, losses: list) -> torch.Tensor:
        src = self._ranks[self.pp_size - 1]
        if self.is_last and losses:
            val = torch.stack([l.detach().float().cpu() for l in losses]).mean()
        else:
            val = torch.zeros((), dtype=torch.float32)
        buf = val.to(self.device) if self.device.type == "cuda" else val
        dist.broadcast(buf, src=src, group=self.group)
        return buf.cpu()


def interleaved_stage_ids(pp_rank: int, pp_size: int, num_chunks: int) -> list[int]:
    """Global stage ids held by one rank under loop placement."""
    return [c * pp_size + pp_rank for c in range(num_chunks)]


SCHEDULES = {"gpipe": ScheduleGPipe, "1f1b": Schedule1F1B,
             "interleaved": ScheduleInterleaved1F1B}


def get_pipeline_schedule(variant: str, **kwargs) -> PipelineSchedule:
    try:
        return SCHEDULES[variant.lower()](**kwargs)
    except KeyError:
        raise ValueError(f"Unknown PP schedule {variant!r}; "
                         f"have {sorted(SCHEDULES)}") from None


# ---------------------------------------------------------------------------
# config-driven construction (reference PipelineFactory +
# ComponentSelectorFromPipeline, pipeline_parallelism.py:75-129)
# ---------------------------------------------------------------------------

def get_staged_pipeline_schedule(model, device_mesh, variant: str = "1f1b",
                                 n_microbatches: int = 1,
                                 input_weight: float = 1.0,
                                 output_weight: float = 1.0,
                                 use_fqn_split: bool = False,
                                 num_chunks: int = 1, device=None):
    """Registry factory: split `model` over the mesh's PP dimension and
    return the schedule (the stage is reachable as `.stage` /
    `.stages` for the optimizer/engine wiring; see
    get_stage_from_schedule)."""
    from modalities_amd.parallel.mesh import ParallelismDegrees
    pp = device_mesh.dims[ParallelismDegrees.PP]
    if pp.size == 1:
        raise ValueError("staged pipeline requested but pp degree is 1")
    if variant.lower() == "interleaved":
        if num_chunks < 2:
            raise ValueError("interleaved schedule needs num_chunks >= 2")
        stages_all = split_model_into_stages(model, pp.size * num_chunks,
                                             input_weight, output_weight)
        mine = [stages_all[i]
                for i in interleaved_stage_ids(pp.rank, pp.size, num_chunks)]
        return ScheduleInterleaved1F1B(mine, pp_rank=pp.rank, pp_size=pp.size,
                                       n_microbatches=n_microbatches,
                                       group=pp.group, device=device)
    if use_fqn_split:
        from modalities_amd.parallel.pp_split import \
            split_model_into_stages_by_fqn
        stages = split_model_into_stages_by_fqn(model, pp.size)
    else:
        stages = split_model_into_stages(model, pp.size, input_weight,
                                         output_weight)
    return get_pipeline_schedule(
        variant, stage=stages[pp.rank], stage_idx=pp.rank,
        num_stages=pp.size, n_microbatches=n_microbatches, group=pp.group,
        device=device)


def get_stage_from_schedule(pp_schedule):
    """Component selector: the local stage module (for optimizer / engine
    wiring in YAML; reference ComponentSelectorFromPipeline)."""
    if hasattr(pp_schedule, "stages"):
        import torch.nn as nn

        class _Chunks(nn.Module):
            def __init__(self, stages):
                super().__init__()
                self.chunks = nn.ModuleList(stages)

        return _Chunks(pp_schedule.stages)
    return pp_schedule.stage
