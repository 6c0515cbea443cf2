"""JSON-schema → regex for constrained decoding.

Role of the reference's outlines-core build_regex_from_schema
(vllm/v1/structured_output/backend_outlines.py): compile a JSON schema
into a regex accepted by guided_regex.RegexFSM. Design choices kept
deliberately FSM-friendly:

- All declared properties are emitted in declaration order and treated
  as required (emitting an optional property still VALIDATES against the
  schema, and fixed ordering keeps the DFA small).
- Whitespace is a single optional space after ``:`` / ``,`` and inside
  brackets (the model can always emit compact JSON).
- ``type: object/array`` without item/property schemas, and the generic
  ``json_object`` response format, expand to an any-JSON regex with
  bounded nesting depth (regexes cannot recurse).
"""

from __future__ import annotations

import json
import re
from typing import Any

_WS = "[ ]?"

STRING_INNER = r'(?:[^"\\\x00-\x1f]|\\["\\/bfnrt]|\\u[0-9a-fA-F]{4})'
STRING = f'"{STRING_INNER}*"'
INTEGER = r"(?:-?(?:0|[1-9][0-9]*))"
NUMBER = r"(?:-?(?:0|[1-9][0-9]*)(?:\.[0-9]+)?(?:[eE][+-]?[0-9]+)?)"
BOOLEAN = r"(?:true|false)"
NULL = r"null"


def _regex_escape(text: str) -> str:
    return re.sub(r"([\\^$.|?*+()\[\]{}])", r"\\\1", text)


def _literal(value: Any) -> str:
    return _regex_escape(json.dumps(value, ensure_ascii=True,
                                    separators=(",", ":")))


def any_json_regex(depth: int = 3) -> str:
    """Any JSON value, nesting bounded at `depth`."""
    scalar = f"(?:{STRING}|{NUMBER}|{BOOLEAN}|{NULL})"
    value = scalar
    for _ in range(depth):
        obj = (rf"\{{{_WS}(?:{STRING}{_WS}:{_WS}{value}"
               rf"(?:,{_WS}{STRING}{_WS}:{_WS}{value})*)?{_WS}\}}")
        arr = rf"\[{_WS}(?:{value}(?:,{_WS}{value})*)?{_WS}\]"
        value = f"(?:{scalar}|{obj}|{arr})"
    obj = (rf"\{{{_WS}(?:{STRING}{_WS}:{_WS}{value}"
           rf"(?:,{_WS}{STRING}{_WS}:{_WS}{value})*)?{_WS}\}}")
    return obj


def schema_to_regex(schema: Any, depth: int = 3) -> str:
    """Regex matching JSON documents valid under (a practical subset of)
    `schema`. Raises ValueError on unsupported constructs."""
    if isinstance(schema, str):
        schema = json.loads(schema)
    if schema is True or schema == {}:
        return any_json_regex(depth)
    return _compile(schema, schema, depth)


def _resolve_ref(ref: str, root: Any) -> Any:
    if not ref.startswith("#/"):
        raise ValueError(f"unsupported $ref {ref!r} (only local refs)")
    node = root
    for part in ref[2:].split("/"):
        part = part.replace("~1", "/").replace("~0", "~")
        node = node[part]
    return node


def _compile(schema: Any, root: Any, depth: int) -> str:
    if "$ref" in schema:
        return _compile(_resolve_ref(schema["$ref"], root), root, depth)
    if "const" in schema:
        return _literal(schema["const"])
    if "enum" in schema:
        return "(?:" + "|".join(_literal(v) for v in schema["enum"]) + ")"
    for key in ("anyOf", "oneOf"):
        if key in schema:
            return "(?:" + "|".join(
                _compile(s, root, depth) for s in schema[key]) + ")"
    if "allOf" in schema and len(schema["allOf"]) == 1:
        return _compile(schema["allOf"][0], root, depth)

    t = schema.get("type")
    if isinstance(t, list):
        return "(?:" + "|".join(
            _compile({**schema, "type": ti}, root, depth) for ti in t) + ")"

    if t == "string":
        if "pattern" in schema:
            pat = schema["pattern"].lstrip("^").rstrip("$")
            return f'"{pat}"'
        lo = schema.get("minLength")
        hi = schema.get("maxLength")
        if lo is not None or hi is not None:
            lo = lo or 0
            rep = f"{{{lo},{hi}}}" if hi is not None else f"{{{lo},}}"
            return f'"{STRING_INNER}{rep}"'
        return STRING
    if t == "integer":
        return INTEGER
    if t == "number":
        return NUMBER
    if t == "boolean":
        return BOOLEAN
    if t == "null":
        return NULL
    if t == "array":
        items = schema.get("items")
        if "prefixItems" in schema:
            parts = [_compile(s, root, depth)
                     for s in schema["prefixItems"]]
            body = f",{_WS}".join(parts)
            return rf"\[{_WS}{body}{_WS}\]"
        item = (_compile(items, root, depth) if isinstance(items, dict)
                else any_json_regex(max(depth - 1, 0)))
        lo = schema.get("minItems", 0)
        hi = schema.get("maxItems")
        if lo == 0:
            tail_hi = "" if hi is None else str(hi - 1)
            body = (f"(?:{item}(?:,{_WS}{item})"
                    f"{{0,{tail_hi}}})?" if hi is not None else
                    f"(?:{item}(?:,{_WS}{item})*)?")
        else:
            head = f",{_WS}".join([item] * lo)
            if hi is None:
                body = f"{head}(?:,{_WS}{item})*"
            else:
                body = f"{head}(?:,{_WS}{item}){{0,{hi - lo}}}"
        return rf"\[{_WS}{body}{_WS}\]"
    if t == "object" or "properties" in schema:
        props = schema.get("properties")
        if not props:
            return any_json_regex(max(depth - 1, 0))
        parts = []
        for name, sub in props.items():
            key = _regex_escape(json.dumps(name, ensure_ascii=True))
            parts.append(f"{key}{_WS}:{_WS}{_compile(sub, root, depth)}")
        body = f",{_WS}".join(parts)
        return rf"\{{{_WS}{body}{_WS}\}}"
    if t is None:
        return any_json_regex(depth)
    raise ValueError(f"unsupported schema type {t!r}")
