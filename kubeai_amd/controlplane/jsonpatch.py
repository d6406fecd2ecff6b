"""RFC-6902 JSON Patch subset (add/remove/replace/copy/move/test) —
the admin escape hatch the reference applies to every engine pod
(config `modelServerPods.jsonPatches`, internal/config/system.go:243-260,
applied at internal/modelcontroller/pod_plan.go:42-44 via
internal/modelcontroller/patch.go)."""
from __future__ import annotations

import copy
from typing import Any


class PatchError(ValueError):
    pass


def _split(pointer: str) -> list[str]:
    if pointer == "":
        return []
    if not pointer.startswith("/"):
        raise PatchError(f"invalid JSON pointer {pointer!r}")
    return [p.replace("~1", "/").replace("~0", "~")
            for p in pointer[1:].split("/")]


def _walk(doc: Any, parts: list[str]):
    """-> (parent, last_token) for the pointer."""
    cur = doc
    for p in parts[:-1]:
        if isinstance(cur, list):
            cur = cur[int(p)]
        elif isinstance(cur, dict):
            if p not in cur:
                raise PatchError(f"path segment {p!r} not found")
            cur = cur[p]
        else:
            raise PatchError(f"cannot traverse {type(cur).__name__}")
    return cur, parts[-1] if parts else None


def _get(doc: Any, pointer: str) -> Any:
    parts = _split(pointer)
    cur = doc
    for p in parts:
        cur = cur[int(p)] if isinstance(cur, list) else cur[p]
    return cur


def apply_patch(doc: Any, patch: list[dict]) -> Any:
    """Returns a patched deep copy; raises PatchError on any failure
    (the reference fails the pod build on a bad patch)."""
    doc = copy.deepcopy(doc)
    for op_obj in patch:
        op = op_obj.get("op")
        path = op_obj.get("path", "")
        parts = _split(path)
        if op in ("add", "replace"):
            value = copy.deepcopy(op_obj.get("value"))
            if not parts:
                doc = value
                continue
            parent, last = _walk(doc, parts)
            if isinstance(parent, list):
                if last == "-":
                    parent.append(value)
                elif op == "add":
                    parent.insert(int(last), value)
                else:
                    parent[int(last)] = value
            else:
                if op == "replace" and last not in parent:
                    raise PatchError(f"replace: {path!r} not found")
                parent[last] = value
        elif op == "remove":
            parent, last = _walk(doc, parts)
            if isinstance(parent, list):
                del parent[int(last)]
            elif last in parent:
                del parent[last]
            else:
                raise PatchError(f"remove: {path!r} not found")
        elif op in ("copy", "move"):
            src = op_obj.get("from")
            value = copy.deepcopy(_get(doc, src))
            if op == "move":
                doc = apply_patch(doc, [{"op": "remove", "path": src}])
            doc = apply_patch(
                doc, [{"op": "add", "path": path, "value": value}]
            )
        elif op == "test":
            if _get(doc, path) != op_obj.get("value"):
                raise PatchError(f"test failed at {path!r}")
        else:
            raise PatchError(f"unknown op {op!r}")
    return doc
