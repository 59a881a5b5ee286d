"""Model base: snake_case python attributes ↔ camelCase wire JSON.

Mirrors the contract of the reference's OpenAPI-generated model classes
(attribute_map + to_dict + equality); from_dict is added because users of
the MI355X stack round-trip watch events back into models."""
from __future__ import annotations

import pprint
from typing import Any, ClassVar


class SdkModel:
    # subclass contract: {python_name: wire_name}
    attribute_map: ClassVar[dict] = {}
    # subclass contract: {python_name: type or (SdkModel subclass) or
    #   ["list", cls] or ["dict", cls]}
    openapi_types: ClassVar[dict] = {}

    def __init__(self, **kwargs):
        for name in self.attribute_map:
            setattr(self, name, kwargs.get(name))
        unknown = set(kwargs) - set(self.attribute_map)
        if unknown:
            raise TypeError(f"{type(self).__name__}: unknown arguments {sorted(unknown)}")

    # ---- serialization ----
    @staticmethod
    def _ser(v: Any):
        if isinstance(v, SdkModel):
            return v.to_dict()
        if isinstance(v, list):
            return [SdkModel._ser(x) for x in v]
        if isinstance(v, dict):
            return {k: SdkModel._ser(x) for k, x in v.items()}
        return v

    def to_dict(self) -> dict:
        out = {}
        for name, wire in self.attribute_map.items():
            v = getattr(self, name)
            if v is not None:
                out[wire] = self._ser(v)
        return out

    @classmethod
    def from_dict(cls, data: dict):
        if data is None:
            return None
        kwargs = {}
        for name, wire in cls.attribute_map.items():
            if wire not in data:
                continue
            v = data[wire]
            t = cls.openapi_types.get(name)
            if isinstance(t, type) and issubclass(t, SdkModel):
                v = t.from_dict(v)
            elif isinstance(t, list) and len(t) == 2:
                kind, sub = t
                if v is not None and isinstance(sub, type) and issubclass(sub, SdkModel):
                    if kind == "list":
                        v = [sub.from_dict(x) for x in v]
                    elif kind == "dict":
                        v = {k: sub.from_dict(x) for k, x in v.items()}
            kwargs[name] = v
        return cls(**kwargs)

    # ---- comparison / repr (generated-SDK parity) ----
    def __eq__(self, other):
        return type(other) is type(self) and self.to_dict() == other.to_dict()

    def __ne__(self, other):
        return not self == other

    def __repr__(self):
        return pprint.pformat(self.to_dict())

    def to_str(self):
        return repr(self)
