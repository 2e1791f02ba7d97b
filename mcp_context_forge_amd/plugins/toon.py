"""TOON (Token-Oriented Object Notation) encoder/decoder.

Reference analog: plugins/toon_encoder/toon.py (JSON→TOON token compression,
30-60% savings per plugins/toon_encoder/README.md:9-50). Written from the
public format description: unquoted keys/simple values, `field[N]: a,b,c`
for primitive arrays, columnar `[N]{f1,f2}:` blocks for homogeneous object
arrays, 2-space indentation for nesting. Decoder included so tests can
assert lossless round-trip.
"""

from __future__ import annotations

import json
import re
from typing import Any, List, Optional, Tuple

_SIMPLE = re.compile(r"^[A-Za-z0-9_.@+\-]+\Z")  # \Z: $ would match before a trailing newline
_NUMERIC = re.compile(r"^-?\d+(\.\d+)?([eE][+-]?\d+)?\Z")


def _scalar(v: Any) -> str:
    if v is None:
        return "null"
    if v is True:
        return "true"
    if v is False:
        return "false"
    if isinstance(v, (int, float)):
        return json.dumps(v)
    s = str(v)
    # quote when ambiguous (looks like a literal/number, has separators, or is empty)
    if s == "" or not _SIMPLE.match(s) or s in ("null", "true", "false") or _NUMERIC.match(s):
        return json.dumps(s)
    return s


def _is_scalar(v: Any) -> bool:
    return v is None or isinstance(v, (str, int, float, bool))


def _uniform_object_array(arr: List[Any]) -> Optional[List[str]]:
    """Fields if arr is a non-empty list of flat dicts with identical scalar keys."""
    if not arr or not all(isinstance(x, dict) and x for x in arr):
        return None
    keys = sorted(arr[0].keys())
    # field names ride inside the `{a,b}` header — anything that needs
    # quoting (commas, quotes, newlines) forces the block-item form
    if not all(isinstance(k, str) and _SIMPLE.match(k) for k in keys):
        return None
    for x in arr:
        if sorted(x.keys()) != keys:
            return None
        if not all(_is_scalar(v) for v in x.values()):
            return None
    return keys


def _encode_value(key: Optional[str], value: Any, indent: int, out: List[str]) -> None:
    pad = "  " * indent
    label = f"{_scalar(key)}" if key is not None else None

    if _is_scalar(value):
        out.append(f"{pad}{label}: {_scalar(value)}" if label else f"{pad}{_scalar(value)}")
        return

    if isinstance(value, list):
        fields = _uniform_object_array(value)
        if fields is not None:
            head = f"{pad}{label}[{len(value)}]{{{','.join(fields)}}}:" if label else f"{pad}[{len(value)}]{{{','.join(fields)}}}:"
            out.append(head)
            for item in value:
                out.append(f"{pad}  " + ",".join(_scalar(item[f]) for f in fields))
            return
        if all(_is_scalar(x) for x in value):
            body = ",".join(_scalar(x) for x in value)
            out.append(f"{pad}{label}[{len(value)}]: {body}" if label else f"{pad}[{len(value)}]: {body}")
            return
        # mixed array: one element per line with '- '
        out.append(f"{pad}{label}[{len(value)}]:" if label else f"{pad}[{len(value)}]:")
        for item in value:
            if _is_scalar(item):
                out.append(f"{pad}  - {_scalar(item)}")
            else:
                out.append(f"{pad}  -")
                _encode_container(item, indent + 2, out)
        return

    if isinstance(value, dict):
        if label:
            out.append(f"{pad}{label}:")
            _encode_container(value, indent + 1, out)
        else:
            _encode_container(value, indent, out)
        return

    out.append(f"{pad}{label}: {_scalar(str(value))}")


def _encode_container(value: Any, indent: int, out: List[str]) -> None:
    if isinstance(value, dict):
        for k, v in value.items():
            _encode_value(str(k), v, indent, out)
    else:
        _encode_value(None, value, indent, out)


def encode(value: Any) -> str:
    out: List[str] = []
    _encode_value(None, value, 0, out)
    return "\n".join(out)


# ---------------------------------------------------------------------------
# Decoder (for round-trip tests)
# ---------------------------------------------------------------------------

def _parse_scalar(tok: str) -> Any:
    tok = tok.strip()
    if tok == "null":
        return None
    if tok == "true":
        return True
    if tok == "false":
        return False
    if tok.startswith('"'):
        try:
            return json.loads(tok)
        except Exception:
            return tok  # malformed quote: treat as raw text (defensive)
    if _NUMERIC.match(tok):
        return json.loads(tok)
    return tok


def _split_csv(s: str) -> List[str]:
    out, cur, q, esc = [], [], False, False
    for ch in s:
        if esc:
            cur.append(ch)
            esc = False
        elif ch == "\\" and q:
            cur.append(ch)
            esc = True
        elif ch == '"':
            q = not q
            cur.append(ch)
        elif ch == "," and not q:
            out.append("".join(cur))
            cur = []
        else:
            cur.append(ch)
    out.append("".join(cur))
    return out


_HEAD = re.compile(r"^(?P<key>\"(?:[^\"\\]|\\.)*\"|[^:\[{]+)?(?:\[(?P<n>\d+)\](?:\{(?P<fields>[^}]*)\})?)?:(?P<rest>.*)$")


def decode(text: str) -> Any:
    lines = [ln for ln in text.splitlines() if ln.strip()]
    # bare top-level scalar; a whole-line JSON string wins over the k:v
    # reading (the encoder quotes scalars like ":" that contain separators)
    if len(lines) == 1:
        t = lines[0].strip()
        if t.startswith('"'):
            try:
                return json.loads(t)
            except Exception:
                pass
        if not _HEAD.match(t):
            return _parse_scalar(t)
    if len(lines) == 1:
        m = _HEAD.match(lines[0].strip())
        if m and m.group("n") is None and m.group("key") is not None and ":" not in lines[0]:
            return _parse_scalar(lines[0].strip())
    pos = 0

    def indent_of(ln: str) -> int:
        return (len(ln) - len(ln.lstrip(" "))) // 2

    def parse_block(level: int) -> Any:
        nonlocal pos
        obj: dict = {}
        items: list = []
        is_array = False
        while pos < len(lines):
            ln = lines[pos]
            ind = indent_of(ln)
            if ind < level:
                break
            body = ln.strip()
            if body.startswith("- ") or body == "-":
                is_array = True
                pos += 1
                if body == "-":
                    # block item: contents are one level deeper than the dash
                    items.append(parse_block(ind + 1))
                else:
                    items.append(_parse_scalar(body[2:]))
                continue
            m = _HEAD.match(body)
            if not m:
                pos += 1
                continue
            key = m.group("key")
            key = _parse_scalar(key) if key is not None else None
            n = m.group("n")
            fields = m.group("fields")
            rest = m.group("rest").strip()
            pos += 1
            if fields is not None:
                rows = []
                fl = fields.split(",")
                for _ in range(int(n)):
                    row = _split_csv(lines[pos].strip())
                    rows.append({f: _parse_scalar(v) for f, v in zip(fl, row)})
                    pos += 1
                val: Any = rows
            elif n is not None and rest:
                val = [_parse_scalar(t) for t in _split_csv(rest)]
            elif n is not None:
                val = parse_block(level + 1)
                if isinstance(val, dict) and not val:
                    val = []
            elif rest:
                val = _parse_scalar(rest)
            else:
                val = parse_block(level + 1)
            if key is None:
                return val
            obj[key] = val
        return items if is_array else obj

    result = parse_block(0)
    return result


def savings(value: Any) -> Tuple[int, int, float]:
    """(json_bytes, toon_bytes, fraction_saved)."""
    j = len(json.dumps(value, separators=(",", ":")))
    t = len(encode(value))
    return j, t, (1.0 - t / j) if j else 0.0
