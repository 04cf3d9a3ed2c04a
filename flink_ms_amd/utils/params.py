"""Flag layer mirroring Flink's ``ParameterTool.fromArgs``.

The reference parses every job's CLI with ``ParameterTool.fromArgs(args)``
(e.g. reference flink-als/src/main/scala/de/tub/it4bi/ALSImpl.scala:18-19,
als-ms/.../qs/SGD.java:40-41): ``--key value`` pairs, ``--key`` alone being a
boolean true, with typed getters and defaults.  This module reproduces that
contract so every CLI job in `flink_ms_amd/cli/` accepts the same flag names
with the same semantics (full flag inventory: SURVEY.md §5).
"""

from __future__ import annotations

from typing import Dict, List, Optional


class ParamsError(ValueError):
    pass


_TRUE = {"true", "1", "yes", "y"}
_FALSE = {"false", "0", "no", "n"}


def _is_number(tok: str) -> bool:
    """Negative numbers (e.g. ``--bias -0.5``) are values, not flags."""
    try:
        float(tok)
        return True
    except ValueError:
        return False


class Params:
    """``--key value`` flag map with typed getters (ParameterTool semantics)."""

    def __init__(self, data: Optional[Dict[str, str]] = None):
        self._data: Dict[str, str] = dict(data or {})

    @classmethod
    def from_args(cls, args: List[str]) -> "Params":
        data: Dict[str, str] = {}
        i = 0
        while i < len(args):
            tok = args[i]
            if not tok.startswith("--") and not tok.startswith("-"):
                raise ParamsError(f"expected --key, got {tok!r}")
            key = tok.lstrip("-")
            if not key:
                raise ParamsError("empty flag name")
            nxt = args[i + 1] if i + 1 < len(args) else None
            if nxt is not None and (not nxt.startswith("-")
                                    or _is_number(nxt)):
                data[key] = nxt
                i += 2
            else:
                # bare flag == boolean true (ParameterTool "no value" behavior)
                data[key] = "__NO_VALUE_KEY"
                i += 1
        return cls(data)

    def has(self, key: str) -> bool:
        return key in self._data

    def get(self, key: str, default: Optional[str] = None) -> Optional[str]:
        v = self._data.get(key)
        if v is None or v == "__NO_VALUE_KEY":
            return default if v is None else default
        return v

    def get_required(self, key: str) -> str:
        if key not in self._data or self._data[key] == "__NO_VALUE_KEY":
            raise ParamsError(f"required flag --{key} missing")
        return self._data[key]

    def get_int(self, key: str, default: Optional[int] = None) -> Optional[int]:
        v = self.get(key)
        if v is None:
            return default
        try:
            return int(v)
        except ValueError as e:
            raise ParamsError(f"--{key}={v!r} is not an int") from e

    def get_required_int(self, key: str) -> int:
        return int(self.get_required(key))

    def get_float(self, key: str, default: Optional[float] = None) -> Optional[float]:
        v = self.get(key)
        if v is None:
            return default
        try:
            return float(v)
        except ValueError as e:
            raise ParamsError(f"--{key}={v!r} is not a float") from e

    def get_bool(self, key: str, default: bool = False) -> bool:
        if key not in self._data:
            return default
        v = self._data[key]
        if v == "__NO_VALUE_KEY":
            return True  # bare --flag
        lv = v.lower()
        if lv in _TRUE:
            return True
        if lv in _FALSE:
            return False
        raise ParamsError(f"--{key}={v!r} is not a bool")

    def to_dict(self) -> Dict[str, str]:
        return dict(self._data)

    def __repr__(self) -> str:
        return f"Params({self._data!r})"
