"""InferenceGraph spec types.

Reference parity: pkg/apis/serving/v1alpha1/inference_graph.go:98-113 (node
types Sequence/Splitter/Ensemble/Switch), :305-340 (InferenceStep fields
Data/Weight/Condition/Dependency), :121 (router timeouts).
"""

from __future__ import annotations

from dataclasses import dataclass, field
from enum import Enum
from typing import Dict, List, Optional


class NodeType(str, Enum):
    Sequence = "Sequence"
    Splitter = "Splitter"
    Ensemble = "Ensemble"
    Switch = "Switch"


class Dependency(str, Enum):
    Soft = "Soft"
    Hard = "Hard"


@dataclass
class InferenceStep:
    name: Optional[str] = None
    node_name: Optional[str] = None        # recurse into another node
    service_name: Optional[str] = None
    service_url: Optional[str] = None
    data: Optional[str] = None             # "$request" | "$response"
    weight: Optional[int] = None           # Splitter
    condition: Optional[str] = None        # gjson-style condition
    dependency: Dependency = Dependency.Soft

    @property
    def step_name(self) -> str:
        return self.name or self.node_name or self.service_name or self.service_url or ""

    @classmethod
    def from_dict(cls, d: Dict) -> "InferenceStep":
        return cls(
            name=d.get("name"),
            node_name=d.get("nodeName"),
            service_name=d.get("serviceName"),
            service_url=d.get("serviceUrl"),
            data=d.get("data"),
            weight=d.get("weight"),
            condition=d.get("condition"),
            dependency=Dependency(d.get("dependency", "Soft")),
        )


@dataclass
class InferenceRouter:
    router_type: NodeType
    steps: List[InferenceStep] = field(default_factory=list)

    @classmethod
    def from_dict(cls, d: Dict) -> "InferenceRouter":
        return cls(
            router_type=NodeType(d["routerType"]),
            steps=[InferenceStep.from_dict(s) for s in d.get("steps", [])],
        )


@dataclass
class InferenceGraphSpec:
    nodes: Dict[str, InferenceRouter]
    timeouts: Optional[Dict[str, int]] = None

    @classmethod
    def from_dict(cls, d: Dict) -> "InferenceGraphSpec":
        return cls(
            nodes={k: InferenceRouter.from_dict(v) for k, v in d["nodes"].items()},
            timeouts=d.get("resourceRequirements", {}).get("timeouts")
            if "resourceRequirements" in d
            else d.get("timeouts"),
        )
