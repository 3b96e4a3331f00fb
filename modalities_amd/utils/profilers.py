"""Steppable profiler framework (capability parity with reference
src/modalities/utils/profilers/profilers.py:12-220): a context-manager
interface the trainer wraps around its batch loop with `.step()` per
micro-batch; implementations capture torch.profiler kernel traces
(torch.profiler on ROCm emits HIP kernel events) and CUDA/HIP memory
snapshots. rocprofv3 counter capture runs OUTSIDE the process (see
tools/rocpd_stats.py + profiles/) — the in-process hooks here cover the
reference's surface."""

import json
import pickle
from pathlib import Path
from typing import Optional

import torch


class SteppableProfilerIF:
    def __enter__(self):
        return self

    def __exit__(self, *a):
        return False

    def step(self) -> None:
        raise NotImplementedError

    def __len__(self) -> int:
        """Total scheduled steps (0 = unbounded)."""
        return 0


class SteppableNoProfiler(SteppableProfilerIF):
    def step(self) -> None:
        pass


class SteppableKernelProfiler(SteppableProfilerIF):
    """torch.profiler with a wait/warmup/active schedule; exports a Chrome
    trace JSON + a key_averages table per active window (reference:
    profilers.py:131-220)."""

    def __init__(self, output_dir: Path, wait: int = 1, warmup: int = 1,
                 active: int = 3, repeat: int = 1, with_stack: bool = False,
                 record_shapes: bool = False, profile_memory: bool = False):
        self.output_dir = Path(output_dir)
        self.output_dir.mkdir(parents=True, exist_ok=True)
        self._len = (wait + warmup + active) * max(repeat, 1)
        activities = [torch.profiler.ProfilerActivity.CPU]
        if torch.cuda.is_available():
            activities.append(torch.profiler.ProfilerActivity.CUDA)
        self._prof = torch.profiler.profile(
            activities=activities,
            schedule=torch.profiler.schedule(wait=wait, warmup=warmup,
                                             active=active, repeat=repeat),
            on_trace_ready=self._on_trace_ready,
            with_stack=with_stack, record_shapes=record_shapes,
            profile_memory=profile_memory)

    def _on_trace_ready(self, prof):
        n = prof.step_num
        prof.export_chrome_trace(str(self.output_dir / f"trace_step{n}.json"))
        table = prof.key_averages().table(
            sort_by="cuda_time_total" if torch.cuda.is_available()
            else "cpu_time_total", row_limit=40)
        (self.output_dir / f"key_averages_step{n}.txt").write_text(table)

    def __enter__(self):
        self._prof.__enter__()
        return self

    def __exit__(self, *a):
        return self._prof.__exit__(*a)

    def step(self) -> None:
        self._prof.step()

    def __len__(self) -> int:
        return self._len


class SteppableMemoryProfiler(SteppableProfilerIF):
    """Records the CUDA/HIP caching-allocator history and pickles a snapshot
    after `num_steps` (reference: profilers.py:86-129)."""

    def __init__(self, output_dir: Path, num_steps: int = 5,
                 max_entries: int = 100_000):
        self.output_dir = Path(output_dir)
        self.output_dir.mkdir(parents=True, exist_ok=True)
        self.num_steps = num_steps
        self.max_entries = max_entries
        self._step = 0

    def __enter__(self):
        if torch.cuda.is_available():
            torch.cuda.memory._record_memory_history(max_entries=self.max_entries)
        return self

    def __exit__(self, *a):
        if torch.cuda.is_available():
            torch.cuda.memory._record_memory_history(enabled=None)
        return False

    def step(self) -> None:
        self._step += 1
        if self._step == self.num_steps and torch.cuda.is_available():
            snap = torch.cuda.memory._snapshot()
            with open(self.output_dir / "memory_snapshot.pickle", "wb") as f:
                pickle.dump(snap, f)

    def __len__(self) -> int:
        return self.num_steps


class SteppableCombinedProfiler(SteppableProfilerIF):
    def __init__(self, profilers: list):
        self.profilers = profilers

    def __enter__(self):
        for p in self.profilers:
            p.__enter__()
        return self

    def __exit__(self, *a):
        for p in reversed(self.profilers):
            p.__exit__(*a)
        return False

    def step(self) -> None:
        for p in self.profilers:
            p.step()

    def __len__(self) -> int:
        return max((len(p) for p in self.profilers), default=0)


def get_profiler(variant: str = "no", output_dir: Optional[Path] = None,
                 global_rank: int = 0, tracked_ranks: Optional[list[int]] = None,
                 **kwargs) -> SteppableProfilerIF:
    """Factory with rank filtering: non-tracked ranks get a NoProfiler
    (reference: profiler_factory.py:60-64)."""
    if tracked_ranks is not None and global_rank not in tracked_ranks:
        return SteppableNoProfiler()
    if variant == "no":
        return SteppableNoProfiler()
    if variant == "kernel":
        return SteppableKernelProfiler(Path(output_dir), **kwargs)
    if variant == "memory":
        return SteppableMemoryProfiler(Path(output_dir), **kwargs)
    if variant == "combined":
        return SteppableCombinedProfiler([
            SteppableKernelProfiler(Path(output_dir), **kwargs),
            SteppableMemoryProfiler(Path(output_dir))])
    raise ValueError(f"Unknown profiler variant {variant!r}")


class RandomDatasetBatchGenerator:
    """Synthetic batch source for the standalone profiling harness
    (reference: utils/profilers/batch_generator.py:28-63)."""

    def __init__(self, vocab_size: int, sequence_length: int, batch_size: int,
                 sample_key: str = "input_ids", target_key: str = "target_ids",
                 seed: int = 0):
        self.vocab_size = vocab_size
        self.sequence_length = sequence_length
        self.batch_size = batch_size
        self.sample_key = sample_key
        self.target_key = target_key
        self._g = torch.Generator().manual_seed(seed)

    def get_batch(self):
        from modalities_amd.batch import DatasetBatch
        ids = torch.randint(0, self.vocab_size,
                            (self.batch_size, self.sequence_length + 1),
                            generator=self._g)
        return DatasetBatch(samples={self.sample_key: ids[:, :-1]},
                            targets={self.target_key: ids[:, 1:]})


class SteppableForwardPass:
    """fwd+bwd+optim steppable component for `profile` runs (reference:
    utils/profilers/steppable_components.py:12-51)."""

    def __init__(self, model, optimizer, loss_fn, batch_generator, device=None):
        self.model = model
        self.optimizer = optimizer
        self.loss_fn = loss_fn
        self.batch_generator = batch_generator
        self.device = device or torch.device("cpu")

    def run_step(self):
        from modalities_amd.models.model import model_predict_batch
        batch = self.batch_generator.get_batch().to(self.device)
        result = model_predict_batch(self.model, batch)
        loss = self.loss_fn(result)
        loss.backward()
        if hasattr(self.model, "backward_epilogue"):
            self.model.backward_epilogue()
        self.optimizer.step()
        self.optimizer.zero_grad()
        return loss.detach()
