"""Config system for the DSIN-AMD framework.

Re-implements the text-config contract of the reference's external
``fjcommon.config_parser`` (used at reference ``src/main.py:184-185``):

* lines of the form ``key = <python expression>`` (tuples, numbers, strings,
  ``None``, arithmetic like ``2*0.02`` all allowed),
* enum constraints of the form ``constrain key :: A, B, C`` whose values are
  bare words interpreted as strings (reference ``src/run_configs/ae_run_configs:22,29,52,62``),
* ``#`` comments and blank lines ignored.

The parsed result is an attribute-style object whose ``str()`` dumps every
``key = value`` pair, which is what the reference writes into
``configs_<model>.txt`` sidecars (reference ``src/main.py:159-163``).
"""

from __future__ import annotations

import ast
import os
from typing import Any, Dict, List, Optional, Tuple

__all__ = ["Config", "parse", "parse_string"]


class ConfigError(ValueError):
    pass


class Config:
    """Attribute-style config with enum validation."""

    def __init__(self, values: Dict[str, Any], constraints: Dict[str, List[str]],
                 source_path: Optional[str] = None):
        self._values = dict(values)
        self._constraints = dict(constraints)
        self._source_path = source_path
        self._validate()

    def _validate(self) -> None:
        for key, allowed in self._constraints.items():
            if key in self._values and self._values[key] is not None:
                val = self._values[key]
                if val not in allowed:
                    raise ConfigError(
                        f"config key {key!r} = {val!r} violates constraint :: {', '.join(allowed)}")

    def __getattr__(self, name: str) -> Any:
        if name.startswith("_"):
            raise AttributeError(name)
        try:
            return self._values[name]
        except KeyError:
            raise AttributeError(f"config has no key {name!r}") from None

    def __setattr__(self, name: str, value: Any) -> None:
        if name.startswith("_"):
            object.__setattr__(self, name, value)
        else:
            self._values[name] = value
            self._validate()

    def __contains__(self, name: str) -> bool:
        return name in self._values

    def get(self, name: str, default: Any = None) -> Any:
        return self._values.get(name, default)

    def keys(self):
        return self._values.keys()

    def as_dict(self) -> Dict[str, Any]:
        return dict(self._values)

    def clone(self, **overrides: Any) -> "Config":
        vals = dict(self._values)
        vals.update(overrides)
        return Config(vals, self._constraints, self._source_path)

    def __str__(self) -> str:
        lines = []
        for key in sorted(self._values):
            lines.append(f"{key} = {self._values[key]!r}")
        return "\n".join(lines)

    def __repr__(self) -> str:
        return f"Config({self._source_path or 'inline'}, {len(self._values)} keys)"


_EVAL_GLOBALS = {"__builtins__": {}}


def _eval_expr(expr: str, path: str, lineno: int) -> Any:
    """Evaluate a config value expression.

    Accepts literals plus arithmetic on them (the reference's configs use
    e.g. ``2*0.02``); a bare identifier is an (unquoted) string — the format
    writes enum values like ``distortion_to_minimize = mae`` without quotes;
    any other name use is rejected.
    """
    if expr not in ("True", "False", "None") and expr.isidentifier():
        return expr
    try:
        node = ast.parse(expr, mode="eval")
    except SyntaxError as e:
        raise ConfigError(f"{path}:{lineno}: cannot parse value {expr!r}: {e}") from None
    for sub in ast.walk(node):
        if isinstance(sub, ast.Name) and sub.id not in ("True", "False", "None"):
            raise ConfigError(
                f"{path}:{lineno}: name {sub.id!r} not allowed in config value {expr!r}")
        if isinstance(sub, (ast.Call, ast.Attribute, ast.Subscript, ast.Lambda)):
            raise ConfigError(
                f"{path}:{lineno}: expression kind not allowed in config value {expr!r}")
    return eval(compile(node, path, "eval"), _EVAL_GLOBALS)  # noqa: S307 - sanitized above


def _strip_comment(line: str) -> str:
    # no string values containing '#' appear in the format; simple split is the contract
    in_sq = in_dq = False
    for i, ch in enumerate(line):
        if ch == "'" and not in_dq:
            in_sq = not in_sq
        elif ch == '"' and not in_sq:
            in_dq = not in_dq
        elif ch == "#" and not in_sq and not in_dq:
            return line[:i]
    return line


def parse_string(text: str, path: str = "<string>") -> Config:
    values: Dict[str, Any] = {}
    constraints: Dict[str, List[str]] = {}
    for lineno, raw in enumerate(text.splitlines(), start=1):
        line = _strip_comment(raw).strip()
        if not line:
            continue
        if line.startswith("constrain "):
            body = line[len("constrain "):]
            if "::" not in body:
                raise ConfigError(f"{path}:{lineno}: malformed constrain line {raw!r}")
            key, allowed_s = body.split("::", 1)
            allowed = [a.strip() for a in allowed_s.split(",") if a.strip()]
            constraints[key.strip()] = allowed
            continue
        if "=" not in line:
            raise ConfigError(f"{path}:{lineno}: expected 'key = value', got {raw!r}")
        key, expr = line.split("=", 1)
        values[key.strip()] = _eval_expr(expr.strip(), path, lineno)
    return Config(values, constraints, path)


def parse(path: str) -> Tuple[Config, str]:
    """Parse a config file. Returns (config, relative path) like the
    reference's ``config_parser.parse`` (``src/main.py:184-185``)."""
    with open(path, "r") as f:
        text = f.read()
    cfg = parse_string(text, path)
    return cfg, os.path.relpath(path)
