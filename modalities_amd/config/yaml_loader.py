"""YAML config loading with ${...} interpolation and env resolvers.

Capability parity with the reference's OmegaConf loader + custom resolvers
(reference: src/modalities/config/config.py:528-582): supports

- ``${path.to.key}``       - interpolation within the config tree
- ``${cuda_env:RANK}``     - process env (RANK / LOCAL_RANK / WORLD_SIZE)
- ``${modalities_env:X}``  - run context (experiment_id, config_file_path)
- ``${node_env:num_cpus}`` - node facts
- custom injectable resolvers (e.g. warmstart_env) via `additional_resolvers`.

Implemented directly on pyyaml (OmegaConf is not a dependency of this
framework); scalar results preserve their type when the whole string is one
interpolation.
"""

import os
import re
from pathlib import Path
from typing import Any, Callable, Optional

import yaml

_INTERP = re.compile(r"\$\{([^{}]+)\}")


def _node_env(key: str):
    if key == "num_cpus":
        return os.cpu_count()
    raise KeyError(f"Unknown node_env key: {key}")


def load_app_config_dict(
    config_file_path: Path,
    experiment_id: Optional[str] = None,
    additional_resolver_funs: Optional[dict[str, Callable]] = None,
) -> dict:
    with open(config_file_path) as f:
        cfg = yaml.safe_load(f)
    resolvers: dict[str, Callable[[str], Any]] = {
        "cuda_env": lambda k: os.environ[k],
        "node_env": _node_env,
        "modalities_env": lambda k: {
            "experiment_id": experiment_id,
            "config_file_path": str(config_file_path),
        }[k],
    }
    if additional_resolver_funs:
        resolvers.update(additional_resolver_funs)
    return resolve_interpolations(cfg, resolvers)


def resolve_interpolations(cfg: Any, resolvers: Optional[dict[str, Callable]] = None) -> Any:
    resolvers = resolvers or {}

    def lookup_path(path: str):
        node = cfg
        for part in path.split("."):
            if isinstance(node, list):
                node = node[int(part)]
            elif isinstance(node, dict):
                if part not in node:
                    raise KeyError(f"Interpolation path {path!r} not found (missing {part!r})")
                node = node[part]
            else:
                raise KeyError(f"Interpolation path {path!r} descends into scalar at {part!r}")
        return node

    resolving: set = set()

    def resolve_value(v, where: str):
        if isinstance(v, str):
            return resolve_str(v, where)
        if isinstance(v, dict):
            return {k: resolve_value(x, f"{where}.{k}") for k, x in v.items()}
        if isinstance(v, list):
            return [resolve_value(x, f"{where}.{i}") for i, x in enumerate(v)]
        return v

    def resolve_expr(expr: str, where: str):
        if ":" in expr and not expr.split(":", 1)[0].count("."):
            name, arg = expr.split(":", 1)
            if name in resolvers:
                return resolvers[name](arg)
            raise KeyError(f"Unknown resolver {name!r} in ${{{expr}}} at {where}")
        if expr in resolving:
            raise ValueError(f"Interpolation cycle at {expr!r}")
        resolving.add(expr)
        try:
            target = lookup_path(expr)
            return resolve_value(target, expr)
        finally:
            resolving.discard(expr)

    def resolve_str(s: str, where: str):
        m = _INTERP.fullmatch(s.strip())
        if m:  # whole-string interpolation: preserve type
            return resolve_expr(m.group(1), where)
        return _INTERP.sub(lambda mm: str(resolve_expr(mm.group(1), where)), s)

    return resolve_value(cfg, "")
