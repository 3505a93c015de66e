"""EndpointPickerConfig schema (parity: apix/config/v1alpha1/endpointpickerconfig_types.go).

Same YAML shape as the reference (docs/architecture.md:147-168):

    apiVersion: inference.networking.x-k8s.io/v1alpha1
    kind: EndpointPickerConfig
    featureGates: { flowControl: true }
    plugins:
      - name: prefix
        type: prefix-cache-scorer
        parameters: { blockSizeTokens: 16 }
    schedulingProfiles:
      - name: default
        plugins:
          - pluginRef: prefix
            weight: 2
"""
from dataclasses import dataclass, field
from typing import Any, Dict, List, Optional


@dataclass
class PluginSpec:
    type: str
    name: str = ""           # defaults to type when omitted
    parameters: Dict[str, Any] = field(default_factory=dict)

    def __post_init__(self):
        if not self.name:
            self.name = self.type


@dataclass
class ProfilePluginRef:
    plugin_ref: str
    weight: float = 1.0


@dataclass
class SchedulingProfileSpec:
    name: str
    plugins: List[ProfilePluginRef] = field(default_factory=list)


@dataclass
class EndpointPickerConfig:
    plugins: List[PluginSpec] = field(default_factory=list)
    scheduling_profiles: List[SchedulingProfileSpec] = field(default_factory=list)
    feature_gates: Dict[str, bool] = field(default_factory=dict)
    parser: Optional[str] = None          # request parser plugin name
    data_layer: Dict[str, Any] = field(default_factory=dict)
    flow_control: Dict[str, Any] = field(default_factory=dict)

    @staticmethod
    def from_dict(d: Dict[str, Any]) -> "EndpointPickerConfig":
        plugins = [PluginSpec(type=p["type"], name=p.get("name", ""),
                              parameters=p.get("parameters", {}) or {})
                   for p in d.get("plugins", []) or []]
        profiles = []
        for pr in d.get("schedulingProfiles", []) or []:
            refs = [ProfilePluginRef(plugin_ref=x["pluginRef"],
                                     weight=float(x.get("weight", 1.0)))
                    for x in pr.get("plugins", []) or []]
            profiles.append(SchedulingProfileSpec(name=pr["name"], plugins=refs))
        return EndpointPickerConfig(
            plugins=plugins,
            scheduling_profiles=profiles,
            feature_gates=dict(d.get("featureGates", {}) or {}),
            parser=d.get("parser"),
            data_layer=dict(d.get("dataLayer", {}) or {}),
            flow_control=dict(d.get("flowControl", {}) or {}),
        )
