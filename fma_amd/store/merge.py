"""Merge-patch semantics shared by the stores and the nominal builder.

Two flavors, matching the Kubernetes content types:

- :func:`merge_patch` — RFC 7386 JSON merge patch
  (``application/merge-patch+json``): dicts merge recursively, ``null``
  deletes a key, lists REPLACE wholesale.
- :func:`strategic_merge` — the strategic subset Pod specs need
  (``application/strategic-merge-patch+json``; reference
  inference-server.go:1843-1947 applies the server-patch annotation this
  way client-side): like merge patch, but lists of objects carrying a
  ``name`` key merge element-wise by name (patchMergeKey=name), the way
  containers/volumes/env merge in a real apiserver.

Both are PURE: untouched branches of ``base`` are deep-copied, never
aliased, so mutating a result cannot corrupt the caller's input.
"""

from __future__ import annotations

import copy
from typing import Any


def strategic_merge(base: Any, patch: Any) -> Any:
    """Strategic merge patch for the Pod-shaped subset: dicts merge
    recursively; lists of objects with a ``name`` key merge by name;
    other lists replace; ``None`` deletes a key."""
    return _merge(base, patch, strategic=True)


def merge_patch(base: Any, patch: Any) -> Any:
    """RFC 7386 JSON merge patch: dicts merge recursively, ``None``
    deletes, every list replaces."""
    return _merge(base, patch, strategic=False)


def _merge(base: Any, patch: Any, strategic: bool) -> Any:
    if patch is None:
        return None
    if isinstance(base, dict) and isinstance(patch, dict):
        # pure merge: untouched branches are COPIED, never aliased, so
        # mutating the result can't corrupt the caller's base
        out = {k: copy.deepcopy(v) for k, v in base.items()
               if k not in patch}
        out.update({k: copy.deepcopy(base[k]) for k in patch
                    if k in base})
        for k, v in patch.items():
            if v is None:
                out.pop(k, None)
            elif k in out:
                merged = _merge(out[k], v, strategic)
                if merged is None:
                    out.pop(k, None)
                else:
                    out[k] = merged
            else:
                out[k] = copy.deepcopy(v)
        return out
    if strategic and isinstance(base, list) and isinstance(patch, list):
        if all(isinstance(e, dict) and "name" in e for e in base + patch):
            by_name = {e["name"]: copy.deepcopy(e) for e in base}
            order = [e["name"] for e in base]
            for e in patch:
                if e["name"] in by_name:
                    by_name[e["name"]] = _merge(by_name[e["name"]], e,
                                                strategic)
                else:
                    by_name[e["name"]] = copy.deepcopy(e)
                    order.append(e["name"])
            return [by_name[n] for n in order]
        return copy.deepcopy(patch)
    return copy.deepcopy(patch)
