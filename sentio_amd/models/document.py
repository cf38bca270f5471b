"""The universal data unit (reference src/core/models/document.py:8-20)."""

from __future__ import annotations

import uuid
from dataclasses import dataclass, field
from typing import Any


@dataclass
class Document:
    text: str
    metadata: dict[str, Any] = field(default_factory=dict)
    id: str = field(default_factory=lambda: str(uuid.uuid4()))

    def to_dict(self) -> dict[str, Any]:
        return {"id": self.id, "text": self.text, "metadata": dict(self.metadata)}

    @classmethod
    def from_dict(cls, d: dict[str, Any]) -> "Document":
        return cls(
            text=d.get("text", ""),
            metadata=dict(d.get("metadata") or {}),
            id=str(d.get("id") or uuid.uuid4()),
        )
