"""Benchmark sweep tooling (capability parity with reference
src/modalities/utils/benchmarking/sweep_utils.py:22-120 and
benchmarking_utils.py:14-193): cartesian expansion of a sweep config into
per-world-size experiment folders with config hashes, and
remaining-run computation from evaluation_results.jsonl + error logs."""

import hashlib
import itertools
import json
from pathlib import Path

import yaml


def _expand(sweep: dict) -> list[dict]:
    """Keys whose value is {"sweep": [...]} expand cartesian."""
    sweep_keys, sweep_values = [], []

    def walk(node, path):
        if isinstance(node, dict):
            if set(node.keys()) == {"sweep"} and isinstance(node["sweep"], list):
                sweep_keys.append(path)
                sweep_values.append(node["sweep"])
                return
            for k, v in node.items():
                walk(v, path + [k])

    walk(sweep, [])
    combos = list(itertools.product(*sweep_values)) if sweep_values else [()]
    out = []
    for combo in combos:
        cfg = json.loads(json.dumps(sweep))  # deep copy
        for path, value in zip(sweep_keys, combo):
            node = cfg
            for p in path[:-1]:
                node = node[p]
            node[path[-1]] = value
        out.append(cfg)
    return out


def config_hash(cfg: dict) -> str:
    return hashlib.sha256(json.dumps(cfg, sort_keys=True, default=str)
                          .encode()).hexdigest()[:10]


def prepare_sweep_configs(sweep_config_path: Path, output_dir: Path) -> int:
    """Write each expanded config under
    output_dir/world_size_{N}/{hash}/config.yaml."""
    with open(sweep_config_path) as f:
        sweep = yaml.safe_load(f)
    configs = _expand(sweep)
    output_dir = Path(output_dir)
    for cfg in configs:
        world = (cfg.get("settings", {}).get("cuda_env", {})
                 .get("world_size", 1))
        h = config_hash(cfg)
        folder = output_dir / f"world_size_{world}" / h
        folder.mkdir(parents=True, exist_ok=True)
        with open(folder / "config.yaml", "w") as f:
            yaml.safe_dump(cfg, f, sort_keys=False)
    return len(configs)


def _run_is_complete(run_dir: Path) -> bool:
    results = run_dir / "evaluation_results.jsonl"
    if not results.exists():
        return False
    cfg_file = run_dir / "config.yaml"
    target_steps = None
    if cfg_file.exists():
        with open(cfg_file) as f:
            cfg = yaml.safe_load(f)
        target_steps = (cfg.get("settings", {}).get("training_target", {})
                        .get("num_target_steps"))
    with open(results) as f:
        lines = [json.loads(ln) for ln in f if ln.strip()]
    if not lines:
        return False
    if target_steps is None:
        return True
    last = max((ln.get("num_train_steps_done", 0) for ln in lines), default=0)
    return last >= target_steps


def _run_failed_fatally(run_dir: Path, skip_exception_types=("OutOfMemoryError",
                                                             "torch.OutOfMemoryError")):
    for err in run_dir.glob("logs/error_rank_*.json"):
        with open(err) as f:
            rec = json.load(f)
        if rec.get("exception_type") in skip_exception_types:
            return True
    return False


def list_remaining_runs(sweep_dir: Path) -> list[Path]:
    """Configs that still need a (re)run: not complete and not known-fatal
    (reference benchmarking_utils.py:58-193)."""
    remaining = []
    for cfg_path in sorted(Path(sweep_dir).glob("world_size_*/*/config.yaml")):
        run_dir = cfg_path.parent
        if _run_is_complete(run_dir):
            continue
        if _run_failed_fatally(run_dir):
            continue
        remaining.append(cfg_path)
    return remaining
