"""Two-phase EndpointPickerConfig loader
(parity: pkg/epp/config/loader/configloader.go:79-303 LoadRawConfig +
InstantiateAndConfigure, and runner.go:517/:572 parseConfigurationPhase
One/Two: gate extraction, plugin instantiation, auto-creation of default
DataProducers for consumed-but-unproduced keys, producer-DAG topo sort).
"""
from dataclasses import dataclass, field
from typing import Any, Dict, List, Optional

import yaml

from ..api.endpointpicker_config import EndpointPickerConfig
from ..datalayer.attributes import (IN_FLIGHT_LOAD, LATENCY_PREDICTION_INFO,
                                    PREFIX_CACHE_MATCH_INFO)
from ..plugins import register_all_plugins
from ..plugins.interface import (Admitter, DataProducer, Filter, Picker,
                                 ProfileHandler, Scorer)
from ..plugins.registry import global_registry
from ..scheduling.scheduler import SchedulerConfig, SchedulerProfile
from ..requestcontrol.director import RequestControlConfig
from ..utils.logging import get_logger

log = get_logger("config.loader")

# which attribute keys each consumer plugin type needs (the reference infers
# this from typed plugin declarations; runner.go:591-598 auto-creates the
# registered default producer for any consumed-but-unproduced key)
_CONSUMES: Dict[str, List[str]] = {
    "prefix-cache-scorer": [PREFIX_CACHE_MATCH_INFO],
    "prefix-cache-affinity-filter": [PREFIX_CACHE_MATCH_INFO],
    "prefix-based-pd-decider": [PREFIX_CACHE_MATCH_INFO],
    "token-load-scorer": [IN_FLIGHT_LOAD],
    "active-request-scorer": [IN_FLIGHT_LOAD],
    "latency-scorer": [LATENCY_PREDICTION_INFO],
    "slo-headroom-tier-filter": [LATENCY_PREDICTION_INFO],
    "latency-slo-admitter": [LATENCY_PREDICTION_INFO],
    "probabilistic-admitter": [LATENCY_PREDICTION_INFO],
}


@dataclass
class LoadedConfig:
    raw: EndpointPickerConfig
    plugins: Dict[str, Any] = field(default_factory=dict)
    scheduler_config: Optional[SchedulerConfig] = None
    request_control: Optional[RequestControlConfig] = None
    feature_gates: Dict[str, bool] = field(default_factory=dict)

    def gate(self, name: str, default: bool = False) -> bool:
        return self.feature_gates.get(name, default)


def load_config(yaml_text: str) -> LoadedConfig:
    data = yaml.safe_load(yaml_text) or {}
    return load_config_dict(data)


def load_config_dict(data: Dict[str, Any]) -> LoadedConfig:
    register_all_plugins()
    raw = EndpointPickerConfig.from_dict(data)
    loaded = LoadedConfig(raw=raw, feature_gates=dict(raw.feature_gates))

    # phase two: instantiate named plugin instances
    for spec in raw.plugins:
        loaded.plugins[spec.name] = global_registry.instantiate(
            spec.type, name=spec.name, **spec.parameters)

    _resolve_decider_refs(loaded)

    produced = {p.produces for p in loaded.plugins.values()
                if isinstance(p, DataProducer) and p.produces}
    # auto-create default producers for consumed-but-unproduced keys
    consumed: List[str] = []
    for p in loaded.plugins.values():
        consumed.extend(_CONSUMES.get(getattr(p, "type_name", ""), []))
    for key in consumed:
        if key in produced:
            continue
        default_type = global_registry.default_producer_for(key)
        if default_type and default_type not in loaded.plugins:
            log.v(2).info("auto-creating default producer",
                          key=key, type=default_type)
            loaded.plugins[default_type] = global_registry.instantiate(
                default_type)
            produced.add(key)
    # producers' own requires (e.g. approx-prefix needs token-producer)
    changed = True
    while changed:
        changed = False
        for p in list(loaded.plugins.values()):
            if not isinstance(p, DataProducer):
                continue
            for dep in getattr(p, "requires", []):
                if dep not in loaded.plugins:
                    loaded.plugins[dep] = global_registry.instantiate(dep)
                    changed = True

    # scheduling profiles
    profiles: Dict[str, SchedulerProfile] = {}
    handler: Optional[ProfileHandler] = None
    for pspec in raw.scheduling_profiles:
        prof = SchedulerProfile(name=pspec.name)
        for ref in pspec.plugins:
            plugin = loaded.plugins.get(ref.plugin_ref)
            if plugin is None:
                raise KeyError(f"profile {pspec.name!r} references unknown "
                               f"plugin {ref.plugin_ref!r}")
            if isinstance(plugin, Filter):
                prof.filters.append(plugin)
            elif isinstance(plugin, Scorer):
                prof.scorers.append((plugin, ref.weight))
            elif isinstance(plugin, Picker):
                prof.picker = plugin
                prof.max_endpoints = int(
                    plugin.params.get("maxNumOfEndpoints", 1))
            else:
                raise TypeError(f"plugin {ref.plugin_ref!r} "
                                f"({type(plugin).__name__}) not usable in a "
                                "scheduling profile")
        if prof.picker is None:
            prof.picker = global_registry.instantiate("max-score-picker")
        profiles[pspec.name] = prof

    for p in loaded.plugins.values():
        if isinstance(p, ProfileHandler):
            if handler is not None:
                raise ValueError("multiple profile handlers configured")
            handler = p
    if handler is None:
        handler = global_registry.instantiate("single-profile-handler")
    if profiles:
        loaded.scheduler_config = SchedulerConfig(profiles=profiles,
                                                  profile_handler=handler)

    # request-control plugin lists
    rcc = RequestControlConfig()
    for p in loaded.plugins.values():
        if isinstance(p, DataProducer):
            rcc.data_producers.append(p)
        if isinstance(p, Admitter):
            rcc.admitters.append(p)
        if hasattr(p, "pre_request") and not isinstance(p, ProfileHandler):
            rcc.pre_request.append(p)
        if hasattr(p, "response_complete"):
            rcc.response_complete.append(p)
        if hasattr(p, "response_received") and not isinstance(p, ProfileHandler):
            rcc.response_received.append(p)
        if hasattr(p, "response_streaming"):
            rcc.response_streaming.append(p)
    rcc.topo_sort_producers()
    loaded.request_control = rcc
    return loaded


def _resolve_decider_refs(loaded: LoadedConfig) -> None:
    """Resolve disagg-handler decider references: parameters named
    `pdDecider`/`encodeDecider` may name a plugin instance or a type."""
    for p in loaded.plugins.values():
        for attr in ("pd_decider", "encode_decider"):
            ref = getattr(p, attr, None)
            if isinstance(ref, str):
                target = loaded.plugins.get(ref)
                if target is None:
                    target = global_registry.instantiate(ref)
                    loaded.plugins[ref] = target
                setattr(p, attr, target)
