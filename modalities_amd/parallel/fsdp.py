"""XGMI-sharded data parallelism: an explicit ZeRO-3-style engine for MI355X.

This replaces the reference's dependency on torch FSDP2/DTensor internals
(reference: src/modalities/models/model_factory.py:168-246 `fully_shard`,
MixedPrecisionPolicy bf16 param/reduce) with a from-scratch engine designed
for the MI355X node topology:

- Parameters of each block group live as ONE flat fp32 master shard per rank
  (1/world of the group), plus a bf16 working copy materialized by a single
  bucketed RCCL all-gather per group ("big shards, few large collectives" —
  xGMI is 7 point-to-point links x ~153 GB/s, so per-message efficiency
  matters more than on a switched fabric).
- Gradients reduce-scatter in bf16 into persistent fp32 shard accumulators
  (grad accumulation sums in fp32).
- 288 GB HBM3E per GPU means models <= ~10B can keep ALL bf16 params resident
  for the whole step (`reshard_after_forward=False` default): backward needs
  no re-gathers at all; each step costs exactly one all-gather and one
  reduce-scatter per group, overlapped with compute on dedicated HIP streams.
- With `reshard_after_forward=True` (70B-class models) full params are freed
  after each group's forward and re-gathered just-in-time in backward
  (prefetched in reverse execution order on the comm stream).
- The optimizer steps on the flat fp32 master shards via the fused AdamW HIP
  kernel (K9) with a per-element weight-decay mask, so weight-decay grouping
  (reference: optimizers/optimizer_factory.py:77-99) works on flat shards.
- Gradient clipping is sharded-native: local sq-sum over shard accumulators +
  one scalar all-reduce (K10).

CPU/gloo fallback (tests): gloo has no reduce_scatter, so it is emulated with
all_reduce + local slice; streams collapse to synchronous calls.
"""

import re
from contextlib import contextmanager
from typing import Optional

import torch
import torch.distributed as dist
import torch.nn as nn


def _pad_to(n: int, multiple: int) -> int:
    return (n + multiple - 1) // multiple * multiple


class _CommStreams:
    def __init__(self, device: torch.device):
        self.on_gpu = device.type == "cuda"
        self.gather = torch.cuda.Stream(device=device) if self.on_gpu else None
        self.reduce = torch.cuda.Stream(device=device) if self.on_gpu else None

    @contextmanager
    def on(self, stream):
        if self.on_gpu and stream is not None:
            with torch.cuda.stream(stream):
                yield
        else:
            yield


class FlatParamUnit:
    """One shard unit: a set of modules whose params are flattened into one
    fp32 master shard per rank + one bf16 working buffer."""

    def __init__(self, name: str, modules: list[nn.Module], device: torch.device,
                 group, rank: int, world: int, param_dtype: torch.dtype,
                 no_decay_patterns: Optional[list[str]] = None):
        self.name = name
        self.modules = modules
        self.device = device
        self.group = group
        self.rank = rank
        self.world = world
        self.param_dtype = param_dtype

        # Unique params in deterministic order (tied/shared params once).
        seen: dict[int, int] = {}
        self.param_infos: list[tuple[nn.Module, str, nn.Parameter]] = []
        for mod in modules:
            for pname, p in mod.named_parameters(recurse=True):
                if id(p) in seen:
                    continue
                seen[id(p)] = len(self.param_infos)
                self.param_infos.append((mod, pname, p))

        self.shapes = [p.shape for _, _, p in self.param_infos]
        self.numels = [p.numel() for _, _, p in self.param_infos]
        self.offsets = []
        off = 0
        for n in self.numels:
            self.offsets.append(off)
            off += n
        self.total_numel = _pad_to(max(off, 1), max(world, 1) * 64)
        self.shard_numel = self.total_numel // world

        # Weight-decay mask per flat element (1 = decay, 0 = no decay).
        nd_res = [re.compile(pat) for pat in (no_decay_patterns or [])]
        flat_mask = torch.ones(self.total_numel, dtype=torch.float32)
        for (mod, pname, p), offset, numel in zip(self.param_infos, self.offsets,
                                                  self.numels):
            full_name = f"{type(mod).__name__}.{pname}"
            no_decay = p.ndim < 2 or any(r.search(full_name) for r in nd_res)
            if no_decay:
                flat_mask[offset:offset + numel] = 0.0
        s = self.rank * self.shard_numel
        self.wd_mask_shard = flat_mask[s:s + self.shard_numel].to(device)

        self.master_shard = torch.zeros(self.shard_numel, dtype=torch.float32, device=device)
        # grad_shard carries no valid data until the first reduce of a step
        # OVERWRITES it (grad_fresh flag) — avoids a full fp32 fill per step.
        self.grad_shard = torch.empty(self.shard_numel, dtype=torch.float32, device=device)
        self.grad_fresh = False  # True once this step's first reduce landed
        self.bf16_shard = torch.zeros(self.shard_numel, dtype=param_dtype, device=device)
        self.full_buf: Optional[torch.Tensor] = None
        self.grad_full: Optional[torch.Tensor] = None
        self.is_gathered = False
        self.grads_allocated = False
        self.grads_seen = 0
        self._gather_event = None
        self._reduce_event = None

    # -- init ------------------------------------------------------------

    @torch.no_grad()
    def init_from_materialized(self):
        flat = torch.zeros(self.total_numel, dtype=torch.float32, device=self.device)
        for (_, _, p), off, n in zip(self.param_infos, self.offsets, self.numels):
            flat[off:off + n] = p.detach().float().reshape(-1).to(self.device)
        s = self.rank * self.shard_numel
        self.master_shard.copy_(flat[s:s + self.shard_numel])
        self.bf16_shard.copy_(self.master_shard.to(self.param_dtype))
        del flat
        self._point_params_to(None)

    @torch.no_grad()
    def _point_params_to(self, full_buf: Optional[torch.Tensor]):
        for (_, _, p), off, n, shape in zip(self.param_infos, self.offsets,
                                            self.numels, self.shapes):
            if full_buf is None:
                p.data = torch.empty(0, dtype=self.param_dtype, device=self.device)
            else:
                p.data = full_buf[off:off + n].view(shape)

    # -- gather / free ----------------------------------------------------

    def gather(self, streams: _CommStreams):
        if self.is_gathered:
            return
        if self.world == 1:
            # shard IS the full tensor: alias, no copy, no event
            self.full_buf = self.bf16_shard
            self.is_gathered = True
            return
        with streams.on(streams.gather):
            if streams.on_gpu:
                streams.gather.wait_stream(torch.cuda.current_stream(self.device))
            self.full_buf = torch.empty(self.total_numel, dtype=self.param_dtype,
                                        device=self.device)
            dist.all_gather_into_tensor(self.full_buf, self.bf16_shard, group=self.group)
            if streams.on_gpu:
                self._gather_event = torch.cuda.Event()
                self._gather_event.record(streams.gather)
        self.is_gathered = True

    def wait_gather(self, streams: _CommStreams):
        if streams.on_gpu and self._gather_event is not None:
            torch.cuda.current_stream(self.device).wait_event(self._gather_event)
            self._gather_event = None
            # full_buf was allocated on the gather stream but is consumed
            # (and eventually freed) on the compute stream: tell the caching
            # allocator so it cannot hand the pages to another stream while
            # compute still reads them (classic cross-stream FSDP hazard;
            # only reachable at world > 1 on GPU).
            if self.full_buf is not None and self.full_buf.is_cuda:
                self.full_buf.record_stream(
                    torch.cuda.current_stream(self.device))
        self._point_params_to(self.full_buf)

    @torch.no_grad()
    def free_full(self):
        self._point_params_to(None)
        self.full_buf = None
        self.is_gathered = False

    # -- backward ----------------------------------------------------------

    @torch.no_grad()
    def alloc_grad_views(self):
        if self.grad_full is None:
            self.grad_full = torch.zeros(self.total_numel, dtype=self.param_dtype,
                                         device=self.device)
        for (_, _, p), off, n, shape in zip(self.param_infos, self.offsets,
                                            self.numels, self.shapes):
            p.grad = self.grad_full[off:off + n].view(shape)
        self.grads_allocated = True

    def reduce_scatter_grads(self, streams: _CommStreams):
        """Reduce-scatter bf16 grads (mean over DP) into the fp32 shard."""
        gf = self.grad_full
        if gf is None:
            return
        with streams.on(streams.reduce):
            if streams.on_gpu:
                streams.reduce.wait_stream(torch.cuda.current_stream(self.device))
            if self.world > 1:
                backend = dist.get_backend(self.group) if self.group is not None \
                    else dist.get_backend()
                if backend != "gloo":
                    out = torch.empty(self.shard_numel, dtype=self.param_dtype,
                                      device=self.device)
                    dist.reduce_scatter_tensor(out, gf, group=self.group)
                else:
                    dist.all_reduce(gf, group=self.group)
                    s = self.rank * self.shard_numel
                    out = gf[s:s + self.shard_numel]
            else:
                out = gf
            scale = 1.0 / self.world
            if self.grad_fresh:
                self.grad_shard.add_(out, alpha=scale)
            else:  # first reduce of the step overwrites (grad_shard is stale)
                self.grad_shard.copy_(out)
                if scale != 1.0:
                    self.grad_shard.mul_(scale)
                self.grad_fresh = True
            if streams.on_gpu:
                self._reduce_event = torch.cuda.Event()
                self._reduce_event.record(streams.reduce)
                # grad_full was allocated on the compute stream (grad views)
                # but the collective reads it on the reduce stream; dropping
                # the reference below frees it against the ALLOCATION stream
                # — record the reduce stream so the allocator delays reuse
                # until the collective completed (world > 1 GPU only).
                gf.record_stream(streams.reduce)
        for _, _, p in self.param_infos:
            p.grad = None
        self.grad_full = None
        self.grads_allocated = False

    def wait_reduce(self):
        if self._reduce_event is not None:
            torch.cuda.current_stream(self.device).wait_event(self._reduce_event)
            self._reduce_event = None

    @torch.no_grad()
    def publish_master(self):
        self.bf16_shard.copy_(self.master_shard)  # single cast kernel
        if self.world > 1 and self.full_buf is not None:
            self.free_full()  # stale after the update; regathered next fwd


class _UnshardBackwardAnchor(torch.autograd.Function):
    """Identity on a unit-module OUTPUT: backward fires before that module's
    op backwards -> ensure params are gathered + grad views allocated."""

    @staticmethod
    def forward(ctx, engine, unit_idx, t):
        ctx.engine = engine
        ctx.unit_idx = unit_idx
        return t.view_as(t)

    @staticmethod
    def backward(ctx, grad):
        ctx.engine._pre_backward(ctx.unit_idx)
        return None, None, grad


class XGMIShardedModel(nn.Module):
    """Wrap a module for sharded data parallelism over an RCCL group.

    `unit_modules` defines the shard units as lists of modules, e.g.
    [[block0, block1], ..., [wte, lm_head, final_norm]]. Use
    `from_transformer` to derive them from a GPT2-style model."""

    def __init__(self, module: nn.Module, device: torch.device,
                 process_group=None, rank: Optional[int] = None,
                 world_size: Optional[int] = None,
                 unit_modules: Optional[list[list[nn.Module]]] = None,
                 param_dtype: torch.dtype = torch.bfloat16,
                 reshard_after_forward: bool = False,
                 no_decay_patterns: Optional[list[str]] = None,
                 replicate_group=None):
        super().__init__()
        self.module = module
        self.device = device
        self.group = process_group
        if world_size is None:
            world_size = dist.get_world_size(process_group) if dist.is_initialized() else 1
        if rank is None:
            rank = dist.get_rank(process_group) if dist.is_initialized() else 0
        self.world = world_size
        self.rank = rank
        self.reshard_after_forward = reshard_after_forward
        self.streams = _CommStreams(device)
        self._replicate_group = replicate_group

        if unit_modules is None:
            unit_modules = [[module]]
        self.units: list[FlatParamUnit] = []
        for i, mods in enumerate(unit_modules):
            self.units.append(FlatParamUnit(f"unit{i}", mods, device, process_group,
                                            rank, world_size, param_dtype,
                                            no_decay_patterns))
        for buf in module.buffers():
            buf.data = buf.data.to(device)

        self._units_needing_reduce: set = set()
        self._fwd_order: list[int] = []
        self._known_order: list[int] = []
        self._install_hooks(unit_modules)
        self._install_grad_hooks()
        for u in self.units:
            u.init_from_materialized()

    # ------------------------------------------------------------------
    @staticmethod
    def from_transformer(model, device, process_group=None, blocks_per_unit: int = 1,
                         reshard_after_forward: bool = False,
                         param_dtype: torch.dtype = torch.bfloat16,
                         no_decay_patterns: Optional[list[str]] = None,
                         rank=None, world_size=None,
                         replicate_group=None) -> "XGMIShardedModel":
        """Shard a GPT2LLM-shaped model: block groups + a root unit with
        embeddings/head/norms (reference analog: layers_per_fsdp_unit
        grouping, model_factory.py:199-246)."""
        blocks = list(model.blocks)
        unit_modules = [blocks[i:i + blocks_per_unit]
                        for i in range(0, len(blocks), blocks_per_unit)]
        root_mods = [m for name, m in model.named_children() if name != "blocks"]
        unit_modules.append(root_mods)
        return XGMIShardedModel(model, device, process_group=process_group,
                                unit_modules=unit_modules, param_dtype=param_dtype,
                                reshard_after_forward=reshard_after_forward,
                                no_decay_patterns=no_decay_patterns,
                                rank=rank, world_size=world_size,
                                replicate_group=replicate_group)

    # -- hooks -----------------------------------------------------------

    def _install_hooks(self, unit_modules: list[list[nn.Module]]):
        for idx, mods in enumerate(unit_modules):
            for m in mods:
                if not any(True for _ in m.parameters(recurse=True)):
                    continue  # e.g. Dropout: nothing to shard/anchor
                m.register_forward_pre_hook(self._make_fwd_pre(idx))
                m.register_forward_hook(self._make_fwd_post(idx))

    def _install_grad_hooks(self):
        """Per-param post-accumulate-grad hooks drive the reduce-scatter: a
        unit reduces exactly when every one of its params has its grad for
        this backward (immune to graph-order surprises, e.g. the embedding
        backward running after the lm-head anchor)."""
        for idx, unit in enumerate(self.units):
            expected = len(unit.param_infos)
            for _, _, p in unit.param_infos:
                p.register_post_accumulate_grad_hook(
                    self._make_grad_hook(idx, expected))

    def _make_grad_hook(self, idx, expected):
        def hook(param):
            unit = self.units[idx]
            unit.grads_seen += 1
            if unit.grads_seen >= expected:
                unit.grads_seen = 0
                self._finish_unit_backward(idx)
        return hook

    def _wrap(self, fn, idx, obj):
        if isinstance(obj, torch.Tensor) and obj.is_floating_point() and obj.requires_grad:
            return fn.apply(self, idx, obj)
        if isinstance(obj, tuple):
            return tuple(self._wrap(fn, idx, o) for o in obj)
        return obj

    def _make_fwd_pre(self, idx):
        def hook(module, args):
            unit = self.units[idx]
            if not unit.is_gathered:
                unit.gather(self.streams)
            unit.wait_gather(self.streams)
            if idx not in self._fwd_order:
                self._fwd_order.append(idx)
                self._prefetch_after(idx)
            if torch.is_grad_enabled():
                self._units_needing_reduce.add(idx)
            return args
        return hook

    def _make_fwd_post(self, idx):
        def hook(module, args, output):
            if torch.is_grad_enabled():
                output = self._wrap(_UnshardBackwardAnchor, idx, output)
            if self.reshard_after_forward and torch.is_grad_enabled():
                unit = self.units[idx]
                if module is unit.modules[-1]:
                    unit.free_full()
            return output
        return hook

    def _prefetch_after(self, idx):
        try:
            pos = self._known_order.index(idx)
            nxt = self._known_order[pos + 1]
        except (ValueError, IndexError):
            return
        self.units[nxt].gather(self.streams)

    # -- backward engine callbacks ----------------------------------------

    def _pre_backward(self, idx):
        unit = self.units[idx]
        if not unit.is_gathered:
            unit.gather(self.streams)
            # prefetch the unit that backwards next (reverse forward order)
            try:
                pos = self._known_order.index(idx)
                if pos > 0 and self.reshard_after_forward:
                    self.units[self._known_order[pos - 1]].gather(self.streams)
            except ValueError:
                pass
        unit.wait_gather(self.streams)
        if not unit.grads_allocated:
            unit.alloc_grad_views()

    def _finish_unit_backward(self, idx):
        # Reduce whenever this unit has accumulated grads — NOT only when
        # it is still in _units_needing_reduce. With several forwards
        # queued before their backwards (pipeline schedules), the set is
        # discharged by the FIRST micro-batch's backward; gating the
        # reduce on membership silently dropped every later micro-batch's
        # gradient (caught by the PP warmstart-equivalence test).
        # reduce_scatter_grads no-ops when no grads are allocated.
        self._units_needing_reduce.discard(idx)
        unit = self.units[idx]
        unit.reduce_scatter_grads(self.streams)
        if self.reshard_after_forward:
            unit.free_full()

    # -- public API --------------------------------------------------------

    def forward(self, *args, **kwargs):
        if self._fwd_order:
            self._known_order = list(self._fwd_order)
        self._fwd_order = []
        for u in self.units:
            u.grads_seen = 0
        if not self.reshard_after_forward:
            for u in self.units:
                if not u.is_gathered:
                    u.gather(self.streams)
        return self.module(*args, **kwargs)

    def backward_epilogue(self):
        """Call after loss.backward(): flush units whose input anchors could
        not fire (e.g. units entered only via int inputs) and sync streams."""
        for idx in list(self._units_needing_reduce):
            self._finish_unit_backward(idx)
        for u in self.units:
            u.wait_reduce()
            # per-backward hook counters must not leak into the next
            # backward (a unit with hook-less params would otherwise carry
            # residue that shifts reduce timing after a warmstart)
            u.grads_seen = 0
        if self._replicate_group is not None:
            ws = dist.get_world_size(self._replicate_group)
            for u in self.units:
                dist.all_reduce(u.grad_shard, group=self._replicate_group)
                u.grad_shard.div_(ws)

    @torch.no_grad()
    def zero_grad_shards(self):
        # no fill: the next step's first reduce overwrites (grad_fresh)
        for u in self.units:
            u.grad_fresh = False

    @torch.no_grad()
    def publish_master(self):
        for u in self.units:
            u.publish_master()

    @torch.no_grad()
    def prefetch_unit_gather(self, idx: int):
        """Kick off unit idx's next-step all-gather on the comm stream (call
        right after the optimizer updated that unit's bf16 shard: the first
        unit's gather then hides behind the remaining units' optimizer
        work instead of stalling the next forward). No-op when resharding
        after forward (the gather there must stay just-in-time)."""
        if self.reshard_after_forward or self.world == 1:
            return
        u = self.units[idx]
        if not u.is_gathered:
            u.gather(self.streams)

    @torch.no_grad()
    def free_stale_fulls(self):
        """Post-optimizer bookkeeping when the optimizer kernel already
        wrote the bf16 shards itself (fused publish): only drop the stale
        gathered buffers (world 1 aliases the shard — nothing to do)."""
        if self.world > 1:
            for u in self.units:
                if u.full_buf is not None:
                    u.free_full()

    @torch.no_grad()
    def clip_grad_norm_(self, max_norm: Optional[float],
                        norm_type: float = 2.0,
                        pp_group=None) -> torch.Tensor:
        """Global-norm clip over the DP-sharded grads; with pipeline
        parallelism pass the PP group so the norm combines over stages
        (reference fsdp_gradient_clipper.py:166-169) — each rank then clips
        by the MODEL-global norm, and the published value is global."""
        from modalities_amd.ops.adamw import multi_tensor_l2norm, multi_tensor_scale_
        fresh = [u.grad_shard for u in self.units if u.grad_fresh]
        local = multi_tensor_l2norm(fresh) ** 2
        if local.device != torch.device(self.device):
            local = local.to(self.device)
        if self.world > 1:
            dist.all_reduce(local, group=self.group)
        if pp_group is not None:
            dist.all_reduce(local, group=pp_group)
        total = local.sqrt()
        if max_norm is not None and max_norm > 0 and fresh:
            clip = (max_norm / (total + 1e-6)).clamp(max=1.0)
            if getattr(self, "_optimizer_consumes_grad_scale", False) \
                    and total.is_cuda:
                # ShardedAdamW folds the clip coefficient into its fused
                # kernel's grad read — deferring skips a full read+write
                # pass over every grad shard here.
                self._pending_grad_scale = clip
            else:
                multi_tensor_scale_(fresh, clip)
        return total

    # -- state (sharded checkpoints) --------------------------------------

    def shard_state_dict(self) -> dict[str, torch.Tensor]:
        return {f"{u.name}.master_shard": u.master_shard for u in self.units}

    def load_shard_state_dict(self, sd: dict):
        for u in self.units:
            u.master_shard.copy_(sd[f"{u.name}.master_shard"])
            u.publish_master()

    def shard_meta(self) -> dict:
        """Layout metadata for world-size-resharding checkpoint loads."""
        return {
            u.name: {
                "total_numel": u.total_numel,
                "params": [
                    {"module": type(mod).__name__, "name": pname,
                     "offset": off, "numel": n, "shape": list(shape)}
                    for (mod, pname, _), off, n, shape in
                    zip(u.param_infos, u.offsets, u.numels, u.shapes)
                ],
            }
            for u in self.units
        }

    @torch.no_grad()
    def gather_full_state_dict(self) -> dict[str, torch.Tensor]:
        """Gather full fp32 params keyed by original module param names (for
        export / single-process inference). All ranks must call; result is
        identical on all ranks."""
        name_of = {}
        for name, p in self.module.named_parameters():
            name_of[id(p)] = name
        out = {}
        for u in self.units:
            full = torch.empty(u.total_numel, dtype=torch.float32, device=self.device)
            if self.world > 1:
                dist.all_gather_into_tensor(full, u.master_shard, group=self.group)
            else:
                full.copy_(u.master_shard)
            for (mod, pname, p), off, n, shape in zip(u.param_infos, u.offsets,
                                                      u.numels, u.shapes):
                key = name_of.get(id(p), f"{u.name}:{pname}")
                out[key] = full[off:off + n].view(shape).clone()
        return out
