"""In-process cluster state store with Kubernetes apiserver semantics.

This is the coordination substrate the MI355X stack's controllers run
against when no real apiserver is present (this stack targets single-node
and test deployments out of the box; the client surface is narrow enough to
be re-pointed at a real cluster). It reproduces the apiserver behaviors the
reference's controllers depend on:

- monotonically increasing resourceVersion per object; preconditioned
  updates/deletes (UID + resourceVersion) that fail with Conflict the way
  client-go does (reference pkg/controller/launcher-populator/populator.go:488-489)
- finalizer-gated deletion: delete sets deletionTimestamp, removal happens
  when the finalizer list empties (reference
  pkg/controller/dual-pods/inference-server.go:295-329 relies on this)
- generation bump on spec change; status updates do not bump generation
- watch with a global revision sequence and replayable history, the model
  informers are built on (reference uses SharedInformerFactory,
  pkg/controller/dual-pods/controller.go:197-360)
- validating-admission hooks, used to enforce the reference's CEL policies
  (reference config/validating-admission-policies/*.yaml) in-process.
"""

from __future__ import annotations

import threading
from typing import Any, Callable, Dict, Iterator, List, Optional, Tuple

from . import objects as ob


class ApiError(Exception):
    def __init__(self, code: int, message: str):
        super().__init__(f"{code}: {message}")
        self.code = code
        self.message = message


class Conflict(ApiError):
    def __init__(self, message: str):
        super().__init__(409, message)


class NotFound(ApiError):
    def __init__(self, message: str):
        super().__init__(404, message)


class AlreadyExists(ApiError):
    def __init__(self, message: str):
        super().__init__(409, message)


class Invalid(ApiError):
    def __init__(self, message: str):
        super().__init__(422, message)


class RevisionTooOld(ApiError):
    """Watch cursor predates the bounded event history (events were
    evicted unseen): the watcher must re-LIST and resume from the list
    revision. Mirrors Kubernetes' 410 Gone on stale resourceVersions."""

    def __init__(self, message: str):
        super().__init__(410, message)


class WatchEvent:
    __slots__ = ("revision", "type", "kind", "obj")

    def __init__(self, revision: int, type_: str, kind: str, obj: Dict[str, Any]):
        self.revision = revision
        self.type = type_  # ADDED | MODIFIED | DELETED
        self.kind = kind
        self.obj = obj

    def __repr__(self) -> str:  # pragma: no cover
        return f"WatchEvent({self.revision}, {self.type}, {self.kind}/{ob.name_of(self.obj)})"


#: admission hook: (operation, old_obj_or_None, new_obj_or_None, actor) -> None
#: raises Invalid to deny. Mirrors ValidatingAdmissionPolicy evaluation.
AdmissionHook = Callable[[str, Optional[Dict[str, Any]], Optional[Dict[str, Any]], str], None]


class MemStore:
    """Thread-safe in-memory object store keyed by (kind, namespace, name)."""

    def __init__(self) -> None:
        self._lock = threading.Condition()
        self._objects: Dict[Tuple[str, str, str], Dict[str, Any]] = {}
        self._revision = 0
        self._history: List[WatchEvent] = []
        self._history_cap = 100_000
        self._admission: List[AdmissionHook] = []
        #: (kind, index_name) -> (fn, {key: {(ns, name): None}})
        self._indexes: Dict[Tuple[str, str], Tuple[Callable, Dict]] = {}
        #: (kind, index_name) -> {(ns, name): [keys]}  (for removal)
        self._indexed_keys: Dict[Tuple[str, str], Dict] = {}

    # -- admission ----------------------------------------------------------

    def add_admission_hook(self, hook: AdmissionHook) -> None:
        with self._lock:
            self._admission.append(hook)

    def _admit(self, op: str, old: Optional[Dict[str, Any]],
               new: Optional[Dict[str, Any]], actor: str) -> None:
        for hook in self._admission:
            hook(op, old, new, actor)

    # -- indexes -------------------------------------------------------------

    def add_index(self, kind: str, name: str,
                  fn: Callable[[Dict[str, Any]], List[str]]) -> None:
        """Register an index (idempotent): fn(obj) -> list of keys. Mirrors
        the reference's informer indexers (controller.go:129-159); here
        the index lives store-side and is maintained on every mutation."""
        with self._lock:
            if (kind, name) in self._indexes:
                return
            idx: Dict[str, Dict] = {}
            rev: Dict = {}
            for (k, ns, nm), obj in self._objects.items():
                if k != kind:
                    continue
                keys = fn(obj) or []
                if keys:
                    rev[(ns, nm)] = keys
                    for key in keys:
                        idx.setdefault(key, {})[(ns, nm)] = None
            self._indexes[(kind, name)] = (fn, idx)
            self._indexed_keys[(kind, name)] = rev

    def index_get(self, kind: str, index_name: str, key: str,
                  namespace: Optional[str] = "default"
                  ) -> List[Dict[str, Any]]:
        """All objects whose index keys include ``key`` — a dict hit, not
        a scan. Raises KeyError for an unregistered index."""
        with self._lock:
            if (kind, index_name) not in self._indexes:
                raise KeyError(f"no index {index_name!r} on {kind}")
            _, idx = self._indexes[(kind, index_name)]
            out = []
            for (ns, nm) in idx.get(key, {}):
                if namespace is not None and ns != namespace:
                    continue
                obj = self._objects.get((kind, ns, nm))
                if obj is not None:
                    out.append(ob.deepcopy(obj))
            return out

    def _index_event(self, type_: str, kind: str, obj: Dict[str, Any]) -> None:
        ident = (ob.namespace_of(obj), ob.name_of(obj))
        for (k, iname), (fn, idx) in self._indexes.items():
            if k != kind:
                continue
            rev = self._indexed_keys[(k, iname)]
            for key in rev.pop(ident, []):
                bucket = idx.get(key)
                if bucket is not None:
                    bucket.pop(ident, None)
                    if not bucket:
                        idx.pop(key, None)
            if type_ != "DELETED":
                keys = fn(obj) or []
                if keys:
                    rev[ident] = keys
                    for key in keys:
                        idx.setdefault(key, {})[ident] = None

    # -- core CRUD ----------------------------------------------------------

    def _emit(self, type_: str, kind: str, obj: Dict[str, Any]) -> None:
        self._revision += 1
        ev = WatchEvent(self._revision, type_, kind, ob.deepcopy(obj))
        self._history.append(ev)
        if len(self._history) > self._history_cap:
            del self._history[: self._history_cap // 10]
        self._index_event(type_, kind, obj)
        self._lock.notify_all()

    def create(self, obj: Dict[str, Any], actor: str = "system") -> Dict[str, Any]:
        obj = ob.deepcopy(obj)
        kind = obj.get("kind", "")
        if not kind or not ob.name_of(obj):
            raise Invalid("object needs kind and metadata.name")
        key = (kind, ob.namespace_of(obj), ob.name_of(obj))
        with self._lock:
            if key in self._objects:
                raise AlreadyExists(f"{kind} {key[1]}/{key[2]} already exists")
            self._admit("CREATE", None, obj, actor)
            m = ob.meta(obj)
            m["uid"] = ob.generate_uid()
            m["creationTimestamp"] = ob.now()
            m["generation"] = 1
            self._revision_stamp(obj)
            self._objects[key] = obj
            self._emit("ADDED", kind, obj)
            return ob.deepcopy(obj)

    def _revision_stamp(self, obj: Dict[str, Any]) -> None:
        ob.meta(obj)["resourceVersion"] = str(self._revision + 1)

    def get(self, kind: str, name: str, namespace: str = "default") -> Dict[str, Any]:
        with self._lock:
            obj = self._objects.get((kind, namespace, name))
            if obj is None:
                raise NotFound(f"{kind} {namespace}/{name} not found")
            return ob.deepcopy(obj)

    def try_get(self, kind: str, name: str, namespace: str = "default") -> Optional[Dict[str, Any]]:
        try:
            return self.get(kind, name, namespace)
        except NotFound:
            return None

    def list(self, kind: str, namespace: Optional[str] = "default",
             label_selector: Optional[Dict[str, str]] = None) -> List[Dict[str, Any]]:
        with self._lock:
            out = []
            for (k, ns, _), obj in self._objects.items():
                if k != kind:
                    continue
                if namespace is not None and ns != namespace:
                    continue
                if label_selector:
                    lbls = ob.labels_of(obj)
                    if any(lbls.get(lk) != lv for lk, lv in label_selector.items()):
                        continue
                out.append(ob.deepcopy(obj))
            return out

    def list_revision(self) -> int:
        with self._lock:
            return self._revision

    def update(self, obj: Dict[str, Any], actor: str = "system",
               expect_uid: Optional[str] = None,
               expect_rv: Optional[str] = None,
               subresource: Optional[str] = None) -> Dict[str, Any]:
        """Full-object update with optimistic concurrency.

        The caller's object must carry the resourceVersion it read (as with
        client-go Update); a mismatch raises Conflict. ``subresource="status"``
        only applies .status and does not bump generation.
        """
        obj = ob.deepcopy(obj)
        kind = obj.get("kind", "")
        key = (kind, ob.namespace_of(obj), ob.name_of(obj))
        with self._lock:
            cur = self._objects.get(key)
            if cur is None:
                raise NotFound(f"{kind} {key[1]}/{key[2]} not found")
            if subresource == "status":
                new = ob.deepcopy(cur)
                new["status"] = obj.get("status", {})
            else:
                new = obj
                # immutable server-side fields
                nm = ob.meta(new)
                cm = ob.meta(cur)
                nm["uid"] = cm["uid"]
                nm["creationTimestamp"] = cm["creationTimestamp"]
                nm["generation"] = cm.get("generation", 1)
                nm["deletionTimestamp"] = cm.get("deletionTimestamp")
                if new.get("spec") != cur.get("spec"):
                    nm["generation"] = cm.get("generation", 1) + 1
            # admission BEFORE the optimistic-concurrency check: a real
            # apiserver runs (validating) admission ahead of the storage
            # commit where the RV conflict is detected, so a stale update
            # that also violates policy is denied as Invalid, not Conflict
            # (the apiserver double behaves the same; the differential
            # fuzz pins the order)
            self._admit("UPDATE", cur, new, actor)
            if expect_uid is not None and ob.uid_of(cur) != expect_uid:
                raise Conflict(f"uid mismatch on {key}")
            rv_expected = expect_rv if expect_rv is not None else ob.rv_of(obj)
            if rv_expected and rv_expected != ob.rv_of(cur):
                raise Conflict(
                    f"resourceVersion conflict on {key}: have {ob.rv_of(cur)}, caller {rv_expected}")
            self._revision_stamp(new)
            self._objects[key] = new
            self._emit("MODIFIED", kind, new)
            # finalizer-gated removal completes when the last finalizer goes
            if ob.is_deleting(new) and not ob.finalizers_of(new):
                self._remove_locked(key, actor)
            return ob.deepcopy(self._objects.get(key, new))

    def patch(self, kind: str, name: str, patch: Dict[str, Any],
              namespace: str = "default", actor: str = "system",
              strategic: bool = False) -> Dict[str, Any]:
        """Server-side merge patch (RFC 7386; ``strategic=True`` for
        kubernetes strategic-merge list semantics). Unlike update, a
        patch carries no resourceVersion: it applies onto whatever the
        current object is, so concurrent patches of disjoint fields never
        lose each other (the reference's notifier sidecar PATCHes the
        instance-signature annotation for exactly this reason). Runs the
        normal admission chain on the merged result."""
        from fma_amd.store.merge import merge_patch, strategic_merge
        fn = strategic_merge if strategic else merge_patch
        for _ in range(16):
            cur = self.get(kind, name, namespace)
            new = fn(cur, patch)
            if not isinstance(new, dict):
                raise Invalid("patch must produce an object")
            try:
                return self.update(new, actor=actor,
                                   expect_rv=ob.rv_of(cur))
            except Conflict:
                continue  # racer moved the object; re-read and re-merge
        raise Conflict(f"patch on {kind} {namespace}/{name} kept "
                       "conflicting after 16 attempts")

    def delete(self, kind: str, name: str, namespace: str = "default",
               actor: str = "system",
               expect_uid: Optional[str] = None,
               expect_rv: Optional[str] = None) -> None:
        key = (kind, namespace, name)
        with self._lock:
            cur = self._objects.get(key)
            if cur is None:
                raise NotFound(f"{kind} {namespace}/{name} not found")
            if expect_uid is not None and ob.uid_of(cur) != expect_uid:
                raise Conflict(f"uid mismatch deleting {key}")
            if expect_rv is not None and ob.rv_of(cur) != expect_rv:
                raise Conflict(f"resourceVersion conflict deleting {key}")
            self._admit("DELETE", cur, None, actor)
            if ob.finalizers_of(cur):
                if not ob.is_deleting(cur):
                    cur = ob.deepcopy(cur)
                    ob.meta(cur)["deletionTimestamp"] = ob.now()
                    self._revision_stamp(cur)
                    self._objects[key] = cur
                    self._emit("MODIFIED", kind, cur)
                return
            self._remove_locked(key, actor)

    def _remove_locked(self, key: Tuple[str, str, str], actor: str) -> None:
        cur = self._objects.pop(key, None)
        if cur is not None:
            self._emit("DELETED", key[0], cur)
            self._gc_owned_locked(cur, actor)

    def _gc_owned_locked(self, owner: Dict[str, Any], actor: str) -> None:
        """Minimal ownerReference garbage collection (the populator sets
        owner-refs from launcher Pods to their LauncherConfig; reference
        utils/pod-helper.go:205-300)."""
        owner_uid = ob.uid_of(owner)
        doomed = [k for k, o in self._objects.items()
                  if any(ref.get("uid") == owner_uid
                         for ref in ob.meta(o).get("ownerReferences", []))]
        for k in doomed:
            cur = self._objects.get(k)
            if cur is None:
                continue
            if ob.finalizers_of(cur):
                if not ob.is_deleting(cur):
                    ob.meta(cur)["deletionTimestamp"] = ob.now()
                    self._revision_stamp(cur)
                    self._emit("MODIFIED", k[0], cur)
            else:
                self._remove_locked(k, actor)

    # -- watch ---------------------------------------------------------------

    def watch(self, since: int = 0, kinds: Optional[List[str]] = None,
              stop: Optional[threading.Event] = None,
              timeout: Optional[float] = None) -> Iterator[WatchEvent]:
        """Yield events with revision > since; blocks for new ones.

        Generator exits when ``stop`` is set or ``timeout`` elapses with no
        new events (informers loop around it). Raises RevisionTooOld when
        history eviction overtakes the cursor (the watcher must re-LIST and
        resume from the list revision, as with Kubernetes 410 Gone).
        """
        cursor = since
        while True:
            batch: List[WatchEvent] = []
            with self._lock:
                while True:
                    if cursor and self._history \
                            and cursor + 1 < self._history[0].revision:
                        raise RevisionTooOld(
                            f"watch cursor {cursor} predates history; "
                            f"oldest is {self._history[0].revision}")
                    batch = [e for e in self._history if e.revision > cursor
                             and (kinds is None or e.kind in kinds)]
                    if batch:
                        break
                    if stop is not None and stop.is_set():
                        return
                    if not self._lock.wait(timeout=timeout if timeout is not None else 0.5):
                        if timeout is not None:
                            return
                        if stop is not None and stop.is_set():
                            return
            for ev in batch:
                cursor = max(cursor, ev.revision)
                yield ev
            if stop is not None and stop.is_set():
                return
