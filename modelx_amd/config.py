"""Model metadata config ``modelx.yaml``
(reference: cmd/modelx/model/config.go:3-18)."""
from __future__ import annotations

from dataclasses import dataclass, field
from typing import Any, Dict, List, Optional

import yaml

MODEL_CONFIG_FILENAME = "modelx.yaml"
README_FILENAME = "README.md"


@dataclass
class ModelConfig:
    description: str = ""
    framework: str = ""
    task: str = ""
    tags: List[str] = field(default_factory=list)
    resources: Dict[str, Any] = field(default_factory=dict)
    maintainers: List[str] = field(default_factory=list)
    annotations: Dict[str, str] = field(default_factory=dict)
    model_files: List[str] = field(default_factory=list)
    config: Any = None

    def to_dict(self) -> Dict[str, Any]:
        d: Dict[str, Any] = {
            "description": self.description,
            "framework": self.framework,
            "task": self.task,
            "tags": self.tags,
            "resources": self.resources,
            "maintainers": self.maintainers,
            "modelFiles": self.model_files,
            "config": self.config,
        }
        if self.annotations:
            d["annotations"] = self.annotations
        return d

    @classmethod
    def from_dict(cls, d: Optional[Dict[str, Any]]) -> "ModelConfig":
        d = d or {}
        return cls(
            description=d.get("description", "") or "",
            framework=d.get("framework", "") or "",
            task=d.get("task", "") or "",
            tags=list(d.get("tags") or []),
            resources=dict(d.get("resources") or {}),
            maintainers=list(d.get("maintainers") or d.get("mantainers") or []),
            annotations=dict(d.get("annotations") or {}),
            model_files=list(d.get("modelFiles") or []),
            config=d.get("config"),
        )

    def to_yaml(self) -> str:
        return yaml.safe_dump(self.to_dict(), sort_keys=False)

    @classmethod
    def from_yaml(cls, text: str) -> "ModelConfig":
        return cls.from_dict(yaml.safe_load(text))

    @classmethod
    def load(cls, path: str) -> "ModelConfig":
        with open(path, "r") as f:
            return cls.from_yaml(f.read())
