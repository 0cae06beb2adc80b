"""Minimal JSON-schema validator.

The reference validates PostgresMgr config and the pg_overrides tunables with
json-schema (``lib/postgresMgr.js:60-161`` CONFIG_SCHEMA / TUNABLES_SCHEMA).
No jsonschema package ships in this image, so this is a from-scratch validator
covering the subset those schemas use: ``type`` (object/string/number/integer/
boolean/array/null), ``required`` (draft-03 style boolean on the property, as
the reference uses, plus draft-04 list style), ``properties``, ``items``,
``enum``, and ``additionalProperties``.
"""

from __future__ import annotations

from typing import Any, List


class ValidationError(ValueError):
    def __init__(self, path: str, message: str):
        self.path = path or "(root)"
        super().__init__("%s: %s" % (self.path, message))


_TYPES = {
    "object": dict,
    "array": list,
    "string": str,
    "boolean": bool,
    "null": type(None),
}


def _type_ok(value: Any, typ: str) -> bool:
    if typ == "number":
        return isinstance(value, (int, float)) and not isinstance(value, bool)
    if typ == "integer":
        return isinstance(value, int) and not isinstance(value, bool)
    if typ == "any":
        return True
    pytype = _TYPES.get(typ)
    if pytype is None:
        raise ValueError("unsupported schema type %r" % (typ,))
    if pytype is dict or pytype is list or pytype is str:
        return isinstance(value, pytype)
    if pytype is bool:
        return isinstance(value, bool)
    return value is None


def validate(value: Any, schema: dict, path: str = "") -> None:
    """Raise ValidationError if ``value`` does not conform to ``schema``."""
    typ = schema.get("type")
    if typ is not None:
        types = typ if isinstance(typ, list) else [typ]
        if not any(_type_ok(value, t) for t in types):
            raise ValidationError(path, "expected type %s, got %s"
                                  % ("/".join(types), type(value).__name__))

    if "enum" in schema and value not in schema["enum"]:
        raise ValidationError(path, "value %r not in enum %r"
                              % (value, schema["enum"]))

    if isinstance(value, dict):
        props = schema.get("properties", {})
        # draft-04 style required list
        for req in schema.get("required", []) if isinstance(schema.get("required"), list) else []:
            if req not in value:
                raise ValidationError(path, "missing required property %r" % req)
        for name, subschema in props.items():
            sub_path = "%s.%s" % (path, name) if path else name
            if name not in value:
                # draft-03 style: {"required": true} on the property schema
                if subschema.get("required") is True:
                    raise ValidationError(path, "missing required property %r" % name)
                continue
            validate(value[name], subschema, sub_path)
        if schema.get("additionalProperties") is False:
            extra = set(value) - set(props)
            if extra:
                raise ValidationError(path, "unexpected properties %r" % sorted(extra))

    if isinstance(value, list) and "items" in schema:
        items = schema["items"]
        for i, item in enumerate(value):
            validate(item, items, "%s[%d]" % (path, i))


def check(value: Any, schema: dict) -> List[str]:
    """Like validate() but returns a list of error strings (empty = valid)."""
    try:
        validate(value, schema)
        return []
    except ValidationError as exc:
        return [str(exc)]
