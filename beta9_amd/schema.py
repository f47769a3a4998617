"""Typed IO schemas for endpoints (reference: sdk/src/beta9/schema.py —
Inputs/Outputs declared on stubs, validated at the boundary)."""

from __future__ import annotations

from typing import Any, Dict, Optional, Type

from pydantic import BaseModel, ValidationError


class Schema:
    """Wraps a pydantic model (or a plain field dict) as a stub IO schema."""

    def __init__(self, model: Optional[Type[BaseModel]] = None, fields: Optional[Dict[str, str]] = None):
        self.model = model
        self.fields = fields or {}

    @classmethod
    def from_pydantic(cls, model: Type[BaseModel]) -> "Schema":
        return cls(model=model)

    def to_spec(self) -> Dict[str, Any]:
        if self.model is not None:
            return {"kind": "pydantic", "json_schema": self.model.model_json_schema()}
        return {"kind": "fields", "fields": self.fields}

    @staticmethod
    def validate_payload(spec: Dict[str, Any], payload: Any) -> Any:
        """Best-effort validation against a serialized spec (runner side).
        Raises ValueError on mismatch."""
        if not spec or not isinstance(payload, dict):
            return payload
        if spec.get("kind") == "pydantic":
            js = spec.get("json_schema", {})
            required = js.get("required", [])
            props = js.get("properties", {})
            for field in required:
                if field not in payload:
                    raise ValueError(f"missing required field {field!r}")
            _TYPES = {"integer": int, "number": (int, float), "string": str, "boolean": bool, "array": list, "object": dict}
            for name, val in payload.items():
                prop = props.get(name)
                if prop and "type" in prop:
                    expect = _TYPES.get(prop["type"])
                    if expect and not isinstance(val, expect):
                        raise ValueError(f"field {name!r} expected {prop['type']}, got {type(val).__name__}")
            return payload
        if spec.get("kind") == "fields":
            for field in spec.get("fields", {}):
                if field not in payload:
                    raise ValueError(f"missing required field {field!r}")
        return payload
