"""Recursive config->object builder (DAG of components).

Capability parity with the reference ComponentFactory (reference:
src/modalities/config/component_factory.py:23-213): a YAML config is a DAG of
``{component_key, variant_key, config}`` nodes and
``{instance_key, pass_type}`` references; this resolves it depth-first into
live objects, caching top-level instances so ``BY_REFERENCE`` edges share one
instance.
"""

from typing import Any, Type, TypeVar

from pydantic import BaseModel

from modalities_amd.registry.registry import Registry

T = TypeVar("T", bound=BaseModel)


class ComponentFactory:
    def __init__(self, registry: Registry):
        self.registry = registry

    def build_components(self, config_dict: dict, components_model_type: Type[T]) -> T:
        """Build the top-level components named by the fields of
        `components_model_type` and validate them into that model."""
        fields = set(components_model_type.model_fields.keys())
        required = {name for name, f in components_model_type.model_fields.items()
                    if f.is_required()}
        missing = required - set(config_dict.keys())
        if missing:
            raise KeyError(f"Top-level components missing from config: "
                           f"{sorted(missing)} (have: {sorted(config_dict.keys())})")
        to_build = fields & set(config_dict.keys())
        component_dict = self._build_config(config_dict,
                                            top_level_components_to_build=to_build)
        return components_model_type(**component_dict)

    def build_component_by_key(self, config_dict: dict, key: str) -> Any:
        return self._build_config(config_dict, {key})[key]

    # ---- internals ------------------------------------------------------

    def _build_config(self, config_dict: dict, top_level_components_to_build: set) -> dict:
        top_level_components: dict[str, Any] = {}
        out = {}
        for name in top_level_components_to_build:
            if name not in config_dict:
                raise KeyError(f"Top-level component {name!r} missing from config "
                               f"(have: {sorted(config_dict.keys())})")
            out[name], top_level_components = self._build_component(
                config_dict[name], config_dict, top_level_components, traversal_path=[name]
            )
        return out

    @staticmethod
    def _is_component_config(d: Any) -> bool:
        return isinstance(d, dict) and "component_key" in d

    @staticmethod
    def _is_reference_config(d: Any) -> bool:
        return isinstance(d, dict) and set(d.keys()) == {"instance_key", "pass_type"}

    def _build_component(self, current, full_config, top_level_components, traversal_path):
        if self._is_component_config(current):
            # Cache at top level by name (only for top-level entries).
            top_name = traversal_path[0] if len(traversal_path) == 1 else None
            if top_name is not None and top_name in top_level_components:
                return top_level_components[top_name], top_level_components

            component_key = current["component_key"]
            variant_key = current.get("variant_key", "default")
            cfg = current.get("config", {}) or {}
            # Resolve children first (references and nested components).
            resolved_cfg, top_level_components = self._map_nested(
                cfg, full_config, top_level_components, traversal_path + ["config"]
            )
            entity = self.registry.get_entity(component_key, variant_key)
            instance = self._instantiate(entity, resolved_cfg,
                                         f"{'.'.join(traversal_path)} ({component_key}/{variant_key})")
            if top_name is not None:
                top_level_components[top_name] = instance
            return instance, top_level_components

        if self._is_reference_config(current):
            ref = current["instance_key"]
            pass_type = current["pass_type"]
            if pass_type not in ("BY_REFERENCE", "BY_OWN_REFERENCE"):
                raise ValueError(f"Unknown pass_type {pass_type!r} at {'.'.join(traversal_path)}")
            if ref not in top_level_components:
                if ref not in full_config:
                    raise KeyError(f"Reference {ref!r} at {'.'.join(traversal_path)} does not "
                                   f"name a top-level config entry")
                instance, top_level_components = self._build_component(
                    full_config[ref], full_config, top_level_components, traversal_path=[ref]
                )
                top_level_components[ref] = instance
            return top_level_components[ref], top_level_components

        if isinstance(current, dict):
            return self._map_nested(current, full_config, top_level_components, traversal_path)

        if isinstance(current, list):
            out = []
            for i, item in enumerate(current):
                built, top_level_components = self._build_component(
                    item, full_config, top_level_components, traversal_path + [str(i)]
                )
                out.append(built)
            return out, top_level_components

        return current, top_level_components

    def _map_nested(self, d: dict, full_config, top_level_components, traversal_path):
        out = {}
        for k, v in d.items():
            out[k], top_level_components = self._build_component(
                v, full_config, top_level_components, traversal_path + [k]
            )
        return out, top_level_components

    @staticmethod
    def _instantiate(entity, resolved_cfg: dict, where: str):
        ctype = entity.component_type
        cfg_type = entity.component_config_type
        try:
            if cfg_type is None:
                return ctype(**resolved_cfg)
            validated = cfg_type.model_validate(resolved_cfg, strict=False)
            kwargs = {}
            for field_name in cfg_type.model_fields:
                kwargs[field_name] = getattr(validated, field_name)
            return ctype(**kwargs)
        except Exception as e:
            raise type(e)(f"Error building component at {where}: {e}") from e
