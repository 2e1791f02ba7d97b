"""JSONPath evaluator (reference: main.py:1281 jsonpath_modifier backed by
jsonpath-ng; tool rows carry a `jsonpath_filter` applied to results).

Covers the jsonpath-ng surface the reference exercises:

  $.a.b           dotted fields            $['a b']        quoted fields
  $[0]  $[-1]     indexing                 $[1:4] $[::2]   slices
  $.*   $[*]      wildcards                $..price        recursive descent
  $[0,2] $['a','b'] unions                 $[?(@.x == 'y')] filters

Filter predicates support ==, !=, <, <=, >, >=, =~ (regex), bare existence
(`?(@.x)`), dotted paths on `@`, and numeric/string/bool/null literals.

`evaluate` returns ALL matches (jsonpath-ng `find` semantics);
`jsonpath_filter` wraps it with the reference's modifier behavior: a
single-match path yields the value itself, multi-match constructs yield the
list, no match yields None.
"""

from __future__ import annotations

import json
import re
from typing import Any, List, Optional, Tuple

_FIELD = re.compile(r"[A-Za-z_][\w\-]*")
_NUM = re.compile(r"-?\d+")


class JSONPathError(ValueError):
    pass


def _parse(expr: str) -> List[Tuple[str, Any]]:
    """expr (after `$`) → list of ops: (kind, arg)."""
    ops: List[Tuple[str, Any]] = []
    i, n = 0, len(expr)
    while i < n:
        c = expr[i]
        if c == ".":
            if expr.startswith("..", i):
                i += 2
                if i < n and expr[i] == "[":
                    ops.append(("rdesc", None))   # $..[...] — bracket handled next loop
                    continue
                m = _FIELD.match(expr, i)
                if m:
                    ops.append(("rdesc_field", m.group(0)))
                    i = m.end()
                elif i < n and expr[i] == "*":
                    ops.append(("rdesc_wild", None))
                    i += 1
                else:
                    raise JSONPathError(f"bad recursive descent at {i} in {expr!r}")
            else:
                i += 1
                if i < n and expr[i] == "*":
                    ops.append(("wild", None))
                    i += 1
                else:
                    m = _FIELD.match(expr, i)
                    if not m:
                        raise JSONPathError(f"expected field at {i} in {expr!r}")
                    ops.append(("field", m.group(0)))
                    i = m.end()
        elif c == "[":
            j = _matching_bracket(expr, i)
            inner = expr[i + 1:j].strip()
            i = j + 1
            if inner == "*":
                ops.append(("wild", None))
            elif inner.startswith("?"):
                ops.append(("filter", _compile_filter(inner)))
            elif ":" in inner and not inner.startswith(("'", '"')):
                parts = inner.split(":")
                if len(parts) > 3:
                    raise JSONPathError(f"bad slice {inner!r}")
                vals = [int(p) if p.strip() else None for p in parts] + [None] * (3 - len(parts))
                ops.append(("slice", tuple(vals[:3])))
            else:
                items = _split_union(inner)
                parsed: List[Any] = []
                for it in items:
                    it = it.strip()
                    if it.startswith(("'", '"')):
                        parsed.append(("key", it[1:-1]))
                    elif _NUM.fullmatch(it):
                        parsed.append(("idx", int(it)))
                    else:
                        raise JSONPathError(f"bad bracket item {it!r}")
                ops.append(("union", parsed) if len(parsed) > 1 else parsed[0])
        else:
            raise JSONPathError(f"unexpected {c!r} at {i} in {expr!r}")
    return ops


def _matching_bracket(s: str, i: int) -> int:
    depth = 0
    quote: Optional[str] = None
    for j in range(i, len(s)):
        c = s[j]
        if quote:
            if c == quote and s[j - 1] != "\\":
                quote = None
        elif c in "'\"":
            quote = c
        elif c == "[":
            depth += 1
        elif c == "]":
            depth -= 1
            if depth == 0:
                return j
    raise JSONPathError(f"unbalanced bracket at {i} in {s!r}")


def _split_union(inner: str) -> List[str]:
    out, cur, quote = [], [], None
    for c in inner:
        if quote:
            cur.append(c)
            if c == quote:
                quote = None
        elif c in "'\"":
            quote = c
            cur.append(c)
        elif c == ",":
            out.append("".join(cur))
            cur = []
        else:
            cur.append(c)
    out.append("".join(cur))
    return out


_FILTER = re.compile(
    r"^\?\(\s*@(?P<path>(?:\.[\w\-]+|\[\d+\])*)\s*"
    r"(?:(?P<op>==|!=|<=|>=|<|>|=~)\s*(?P<lit>.+?))?\s*\)$")


def _compile_filter(inner: str):
    m = _FILTER.match(inner)
    if not m:
        raise JSONPathError(f"unsupported filter {inner!r}")
    path = m.group("path") or ""
    subops = _parse(path) if path else []
    op = m.group("op")
    lit_raw = m.group("lit")
    lit: Any = None
    if op is not None:
        lr = lit_raw.strip()
        if op == "=~":
            lit = re.compile(lr.strip("/").strip("'\""))
        elif lr.startswith("'") and lr.endswith("'"):
            lit = lr[1:-1]
        else:
            try:
                lit = json.loads(lr)
            except Exception as exc:
                raise JSONPathError(f"bad filter literal {lr!r}") from exc

    def pred(node: Any) -> bool:
        vals = _apply(subops, [node]) if subops else [node]
        if not vals:
            return False
        v = vals[0]
        try:
            if op is None:
                return v is not None and v is not False
            if op == "==":
                return v == lit
            if op == "!=":
                return v != lit
            if op == "=~":
                return isinstance(v, str) and bool(lit.search(v))
            if not isinstance(v, (int, float)) or isinstance(v, bool):
                return False
            return {"<": v < lit, "<=": v <= lit, ">": v > lit, ">=": v >= lit}[op]
        except TypeError:
            return False

    return pred


def _descend(node: Any, acc: List[Any]) -> None:
    acc.append(node)
    if isinstance(node, dict):
        for v in node.values():
            _descend(v, acc)
    elif isinstance(node, list):
        for v in node:
            _descend(v, acc)


def _apply(ops: List[Tuple[str, Any]], nodes: List[Any]) -> List[Any]:
    for kind, arg in ops:
        nxt: List[Any] = []
        for node in nodes:
            if kind == "field" or kind == "key":
                if isinstance(node, dict) and arg in node:
                    nxt.append(node[arg])
            elif kind == "idx":
                if isinstance(node, list) and -len(node) <= arg < len(node):
                    nxt.append(node[arg])
            elif kind == "wild":
                if isinstance(node, dict):
                    nxt.extend(node.values())
                elif isinstance(node, list):
                    nxt.extend(node)
            elif kind == "slice":
                if isinstance(node, list):
                    nxt.extend(node[slice(*arg)])
            elif kind == "union":
                for ik, iv in arg:
                    if ik == "key" and isinstance(node, dict) and iv in node:
                        nxt.append(node[iv])
                    elif ik == "idx" and isinstance(node, list) and -len(node) <= iv < len(node):
                        nxt.append(node[iv])
            elif kind == "filter":
                items = node if isinstance(node, list) else [node]
                nxt.extend(it for it in items if arg(it))
            elif kind == "rdesc_field":
                acc: List[Any] = []
                _descend(node, acc)
                nxt.extend(d[arg] for d in acc if isinstance(d, dict) and arg in d)
            elif kind in ("rdesc", "rdesc_wild"):
                acc = []
                _descend(node, acc)
                if kind == "rdesc_wild":
                    for d in acc:
                        if isinstance(d, dict):
                            nxt.extend(d.values())
                        elif isinstance(d, list):
                            nxt.extend(d)
                else:
                    nxt.extend(acc)
        nodes = nxt
        if not nodes:
            break
    return nodes


_MULTI_HINT = re.compile(r"\.\.|\[\s*\*\s*\]|\.\*|\[\?|,|\[[^]]*:")


def evaluate(value: Any, expr: str) -> List[Any]:
    """All matches of `expr` against `value` (jsonpath-ng find semantics)."""
    if not expr or not expr.startswith("$"):
        raise JSONPathError(f"expression must start with $: {expr!r}")
    return _apply(_parse(expr[1:]), [value])


def jsonpath_filter(value: Any, expr: Optional[str]) -> Any:
    """Reference jsonpath_modifier behavior (main.py:1281): apply the tool's
    jsonpath to its result. Single-match deterministic paths yield the value,
    multi-match constructs yield the match list, no match yields None."""
    if not expr or expr == "$":
        return value
    if not expr.startswith("$"):
        return value
    try:
        matches = evaluate(value, expr)
    except JSONPathError:
        return None
    if _MULTI_HINT.search(expr):
        return matches
    return matches[0] if matches else None
