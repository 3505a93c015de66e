"""Plugin factory registry (parity: pkg/epp/framework/interface/plugin/registry.go).

Factories are keyed by type name; `instantiate` builds a named instance from
an EndpointPickerConfig PluginSpec. Legacy type aliases are registered the
same way the reference registers `pd-profile-handler` etc.
"""
from typing import Callable, Dict, List, Optional

from ..utils.logging import get_logger

log = get_logger("plugins.registry")


class PluginRegistry:
    def __init__(self):
        self._factories: Dict[str, Callable] = {}
        self._default_producers: Dict[str, str] = {}  # produced key -> type name

    def register(self, type_name: str, factory: Callable,
                 aliases: Optional[List[str]] = None,
                 default_producer_for: Optional[str] = None) -> None:
        self._factories[type_name] = factory
        for a in aliases or []:
            self._factories[a] = factory
        if default_producer_for:
            self._default_producers[default_producer_for] = type_name

    def instantiate(self, type_name: str, name: str = "", **params):
        if type_name not in self._factories:
            raise KeyError(f"unknown plugin type {type_name!r}; known: "
                           f"{sorted(self._factories)}")
        plugin = self._factories[type_name](name=name or type_name, **params)
        plugin.type_name = getattr(plugin, "type_name", type_name) or type_name
        return plugin

    def known_types(self) -> List[str]:
        return sorted(self._factories)

    def default_producer_for(self, produced_key: str) -> Optional[str]:
        return self._default_producers.get(produced_key)


global_registry = PluginRegistry()


def register_plugin(type_name: str, aliases: Optional[List[str]] = None,
                    default_producer_for: Optional[str] = None):
    """Class decorator: register the class as a plugin factory."""
    def deco(cls):
        cls.type_name = type_name
        global_registry.register(type_name, cls, aliases=aliases,
                                 default_producer_for=default_producer_for)
        return cls
    return deco
