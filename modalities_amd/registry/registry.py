"""Component registry: (component_key, variant_key) -> (type, config type).

Equivalent capability to the reference registry (reference:
src/modalities/registry/registry.py and registry/components.py:187-531) with
the same two-level key namespace so reference-shaped YAML configs carry over.
"""

from dataclasses import dataclass
from typing import Optional, Type


@dataclass(frozen=True)
class ComponentEntity:
    component_key: str
    variant_key: str
    component_type: Type
    component_config_type: Optional[Type] = None  # pydantic model or None


class Registry:
    def __init__(self, components: Optional[list] = None):
        self._registry: dict[tuple[str, str], ComponentEntity] = {}
        for entity in components or []:
            self.add_entity(entity)

    def add_entity(self, entity: ComponentEntity) -> None:
        self._registry[(entity.component_key, entity.variant_key)] = entity

    def register(self, component_key: str, variant_key: str, component_type: Type,
                 component_config_type: Optional[Type] = None) -> None:
        self.add_entity(ComponentEntity(component_key, variant_key, component_type,
                                        component_config_type))

    def get_entity(self, component_key: str, variant_key: str) -> ComponentEntity:
        try:
            return self._registry[(component_key, variant_key)]
        except KeyError:
            known = sorted(k for k in self._registry if k[0] == component_key)
            raise ValueError(
                f"Component ({component_key!r}, {variant_key!r}) not registered. "
                f"Known variants for {component_key!r}: {[k[1] for k in known]}"
            ) from None

    def get_component(self, component_key: str, variant_key: str) -> Type:
        return self.get_entity(component_key, variant_key).component_type

    def get_config(self, component_key: str, variant_key: str) -> Optional[Type]:
        return self.get_entity(component_key, variant_key).component_config_type

    def keys(self):
        return sorted(self._registry.keys())
