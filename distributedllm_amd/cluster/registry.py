"""Models registry: provisioned models and their slice placement.

Parity with the reference's ``models_registry/registry.json``
(/root/reference/distllm/cli_api/provision.py:103-121): model_id →
{metadata, model_dir, slices: [{path, a, b, address}], extra_layers_file}.
"""
from __future__ import annotations

import json
import os
from dataclasses import dataclass, field
from typing import Dict, List, Optional

REGISTRY_DIR = "models_registry"
REGISTRY_FILE = "registry.json"


@dataclass
class SliceEntry:
    path: str
    a: int            # first layer (inclusive)
    b: int            # last layer (inclusive)
    address: str = ""  # "host:port" of the node this slice belongs to

    def to_dict(self):
        return dict(path=self.path, a=self.a, b=self.b, address=self.address)


@dataclass
class ModelEntry:
    model_id: str
    metadata: dict
    model_dir: str
    slices: List[SliceEntry] = field(default_factory=list)
    extra_layers_file: str = ""

    def to_dict(self):
        return dict(model_id=self.model_id, metadata=self.metadata,
                    model_dir=self.model_dir,
                    slices=[s.to_dict() for s in self.slices],
                    extra_layers_file=self.extra_layers_file)

    @classmethod
    def from_dict(cls, d):
        return cls(model_id=d["model_id"], metadata=d.get("metadata", {}),
                   model_dir=d.get("model_dir", ""),
                   slices=[SliceEntry(**s) for s in d.get("slices", [])],
                   extra_layers_file=d.get("extra_layers_file", ""))


class Registry:
    def __init__(self, root: str = "."):
        self.path = os.path.join(root, REGISTRY_DIR, REGISTRY_FILE)
        self.models: Dict[str, ModelEntry] = {}
        self._load()

    def _load(self):
        try:
            with open(self.path) as f:
                data = json.load(f)
        except FileNotFoundError:
            return
        for mid, d in data.items():
            self.models[mid] = ModelEntry.from_dict(d)

    def save(self):
        os.makedirs(os.path.dirname(self.path), exist_ok=True)
        tmp = self.path + ".tmp"
        with open(tmp, "w") as f:
            json.dump({k: v.to_dict() for k, v in self.models.items()}, f,
                      indent=1)
        os.replace(tmp, self.path)

    def add(self, entry: ModelEntry):
        self.models[entry.model_id] = entry
        self.save()

    def get(self, model_id: str) -> Optional[ModelEntry]:
        return self.models.get(model_id)
