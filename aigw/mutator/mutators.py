"""Header and body mutators.

Behavioral parity with internal/headermutator/header_mutator.go (set/remove
with original-header restore on retry — restore handled by the processor,
which re-applies mutations to a fresh copy of the original headers each try)
and internal/bodymutator/body_mutator.go:17-38 (sjson-style dotted-path JSON
field set/remove on the request body).
"""

from __future__ import annotations

from typing import Optional

from aigw.filterapi.config import BodyMutation, HeaderMutation


def apply_header_mutation(
    headers: dict[str, str], mut: Optional[HeaderMutation]
) -> dict[str, str]:
    """Return a new header dict with the mutation applied (keys lowercase)."""
    if mut is None:
        return headers
    out = dict(headers)
    for k in mut.remove:
        out.pop(k.lower(), None)
    for k, v in mut.set.items():
        out[k.lower()] = v
    return out


def _path_parts(path: str) -> list[str]:
    # sjson-compatible: dots separate keys; `\.` escapes a literal dot.
    parts: list[str] = []
    cur: list[str] = []
    i = 0
    while i < len(path):
        c = path[i]
        if c == "\\" and i + 1 < len(path):
            cur.append(path[i + 1])
            i += 2
            continue
        if c == ".":
            parts.append("".join(cur))
            cur = []
        else:
            cur.append(c)
        i += 1
    parts.append("".join(cur))
    return parts


def set_json_path(doc: dict, path: str, value: object) -> None:
    parts = _path_parts(path)
    node = doc
    for p in parts[:-1]:
        if isinstance(node, list):
            idx = int(p)
            while len(node) <= idx:
                node.append({})
            node = node[idx]
        else:
            nxt = node.get(p)
            if not isinstance(nxt, (dict, list)):
                nxt = {}
                node[p] = nxt
            node = nxt
    last = parts[-1]
    if isinstance(node, list):
        idx = int(last)
        while len(node) <= idx:
            node.append(None)
        node[idx] = value
    else:
        node[last] = value


def remove_json_path(doc: dict, path: str) -> None:
    parts = _path_parts(path)
    node = doc
    for p in parts[:-1]:
        if isinstance(node, list):
            try:
                node = node[int(p)]
            except (ValueError, IndexError):
                return
        elif isinstance(node, dict):
            node = node.get(p)
        else:
            return
        if node is None:
            return
    last = parts[-1]
    if isinstance(node, list):
        try:
            node.pop(int(last))
        except (ValueError, IndexError):
            return
    elif isinstance(node, dict):
        node.pop(last, None)


def apply_body_mutation(body: dict, mut: Optional[BodyMutation]) -> dict:
    """Apply set/remove paths to a parsed JSON body in place; returns body."""
    if mut is None:
        return body
    for p in mut.remove:
        remove_json_path(body, p)
    for p, v in mut.set.items():
        set_json_path(body, p, v)
    return body
