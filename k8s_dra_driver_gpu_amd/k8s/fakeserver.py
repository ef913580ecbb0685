"""In-memory Kubernetes API server with real API semantics.

The CPU-CI backbone: the analog of the reference's generated **fake
clientset** (``pkg/nvidia.com/clientset/versioned/fake``) plus the kind-based
mock harness — but implemented as one in-memory store with the real API
machinery the controllers depend on:

* resourceVersion bumping on every mutation, optimistic-concurrency checks
  on update,
* list + watch with label-selector filtering and client-go watch-cache
  semantics: list returns a collection resourceVersion, ``watch`` resumes
  from a given resourceVersion out of a bounded event history (a watch
  older than the retained window gets **410 Gone**, forcing a relist), and
  idle streams emit BOOKMARK events so clients' last-seen RV advances,
* finalizer semantics: DELETE on an object with finalizers sets
  ``deletionTimestamp``; the object disappears when the last finalizer is
  removed,
* namespaced and cluster-scoped resources, uid assignment.

Objects are plain dicts in k8s JSON shape; typed layers (api.types) decode on
top. Thread-safe.
"""

from __future__ import annotations

import copy
import queue
import threading
import time
import uuid as uuidlib
from typing import Any, Dict, Iterator, List, Optional, Tuple


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


class Gone(ApiError):
    """410: the requested resourceVersion fell out of the watch cache."""

    def __init__(self, message: str):
        super().__init__(410, message)


def _matches_selector(obj: Dict[str, Any], selector: Optional[Dict[str, str]]) -> bool:
    if not selector:
        return True
    labels = (obj.get("metadata") or {}).get("labels") or {}
    return all(labels.get(k) == v for k, v in selector.items())


class WatchEvent:
    def __init__(self, type_: str, obj: Dict[str, Any]):
        self.type = type_
        self.object = obj

    def __repr__(self):
        return f"WatchEvent({self.type}, {self.object.get('metadata', {}).get('name')})"


class Watch:
    def __init__(self, server: "FakeApiServer", resource: str, namespace: Optional[str],
                 selector: Optional[Dict[str, str]],
                 bookmark_interval: Optional[float] = None):
        self._q: "queue.Queue[Optional[WatchEvent]]" = queue.Queue()
        self._server = server
        self._resource = resource
        self._namespace = namespace
        self._selector = selector
        self._stopped = False
        # client-go-style watch bookmarks: when idle for this long, emit a
        # BOOKMARK event carrying only metadata.resourceVersion so the
        # client's last-seen RV advances without object traffic
        self._bookmark_interval = bookmark_interval
        self._last_emit = time.monotonic()

    def _deliver(self, event: WatchEvent) -> None:
        ns = (event.object.get("metadata") or {}).get("namespace", "")
        if self._namespace is not None and ns != self._namespace:
            return
        if not _matches_selector(event.object, self._selector):
            return
        self._q.put(event)

    def _bookmark(self) -> WatchEvent:
        self._last_emit = time.monotonic()
        return WatchEvent(
            "BOOKMARK",
            {"metadata": {"resourceVersion": self._server.current_rv()}},
        )

    def stop(self) -> None:
        self._stopped = True
        self._q.put(None)
        self._server._remove_watch(self._resource, self)

    def __iter__(self) -> Iterator[WatchEvent]:
        while True:
            if self._bookmark_interval is None:
                ev = self._q.get()
            else:
                try:
                    ev = self._q.get(timeout=self._bookmark_interval)
                except queue.Empty:
                    yield self._bookmark()
                    continue
            if ev is None:
                return
            self._last_emit = time.monotonic()
            yield ev

    def next(self, timeout: float = 1.0) -> Optional[WatchEvent]:
        try:
            ev = self._q.get(timeout=timeout)
            if ev is not None:
                self._last_emit = time.monotonic()
            return ev
        except queue.Empty:
            if (self._bookmark_interval is not None
                    and time.monotonic() - self._last_emit >= self._bookmark_interval):
                return self._bookmark()
            return None


class FakeApiServer:
    def __init__(self, history_limit: int = 1024):
        self._lock = threading.RLock()
        self._rv = 0
        # store[resource][(namespace, name)] = obj
        self._store: Dict[str, Dict[Tuple[str, str], Dict[str, Any]]] = {}
        self._watches: Dict[str, List[Watch]] = {}
        # bounded per-resource event history — the watch cache that lets a
        # client resume ``watch(resourceVersion=rv)``; events older than the
        # window are evicted and such resumes get 410 Gone (client-go
        # "too old resource version")
        self._history_limit = history_limit
        self._history: Dict[str, List[Tuple[int, str, Dict[str, Any]]]] = {}
        self._evicted_rv: Dict[str, int] = {}

    # -- helpers ----------------------------------------------------------

    def current_rv(self) -> str:
        with self._lock:
            return str(self._rv)

    def _bump(self, obj: Dict[str, Any]) -> None:
        self._rv += 1
        obj.setdefault("metadata", {})["resourceVersion"] = str(self._rv)

    def _key(self, obj: Dict[str, Any]) -> Tuple[str, str]:
        md = obj.get("metadata") or {}
        return (md.get("namespace", ""), md.get("name", ""))

    def _notify(self, resource: str, type_: str, obj: Dict[str, Any]) -> None:
        hist = self._history.setdefault(resource, [])
        hist.append((int(obj["metadata"]["resourceVersion"]), type_, copy.deepcopy(obj)))
        while len(hist) > self._history_limit:
            evicted = hist.pop(0)
            self._evicted_rv[resource] = max(
                self._evicted_rv.get(resource, 0), evicted[0]
            )
        for w in list(self._watches.get(resource, [])):
            w._deliver(WatchEvent(type_, copy.deepcopy(obj)))

    def _remove_watch(self, resource: str, watch: Watch) -> None:
        with self._lock:
            try:
                self._watches.get(resource, []).remove(watch)
            except ValueError:
                pass

    # -- CRUD ---------------------------------------------------------------

    def create(self, resource: str, obj: Dict[str, Any]) -> Dict[str, Any]:
        with self._lock:
            obj = copy.deepcopy(obj)
            md = obj.setdefault("metadata", {})
            key = self._key(obj)
            if not key[1]:
                if md.get("generateName"):
                    md["name"] = md["generateName"] + uuidlib.uuid4().hex[:5]
                    key = self._key(obj)
                else:
                    raise ApiError(400, "metadata.name required")
            table = self._store.setdefault(resource, {})
            if key in table:
                raise AlreadyExists(f"{resource} {key} already exists")
            md.setdefault("uid", str(uuidlib.uuid4()))
            md.setdefault(
                "creationTimestamp", time.strftime("%Y-%m-%dT%H:%M:%SZ", time.gmtime())
            )
            md.setdefault("generation", 1)
            self._bump(obj)
            table[key] = obj
            self._notify(resource, "ADDED", obj)
            return copy.deepcopy(obj)

    def get(self, resource: str, name: str, namespace: str = "") -> Dict[str, Any]:
        with self._lock:
            obj = self._store.get(resource, {}).get((namespace, name))
            if obj is None:
                raise NotFound(f"{resource} {namespace}/{name} not found")
            return copy.deepcopy(obj)

    def list(
        self,
        resource: str,
        namespace: Optional[str] = None,
        selector: Optional[Dict[str, str]] = None,
    ) -> List[Dict[str, Any]]:
        with self._lock:
            out = []
            for (ns, _), obj in sorted(self._store.get(resource, {}).items()):
                if namespace is not None and ns != namespace:
                    continue
                if not _matches_selector(obj, selector):
                    continue
                out.append(copy.deepcopy(obj))
            return out

    def update(self, resource: str, obj: Dict[str, Any]) -> Dict[str, Any]:
        with self._lock:
            obj = copy.deepcopy(obj)
            key = self._key(obj)
            table = self._store.setdefault(resource, {})
            existing = table.get(key)
            if existing is None:
                raise NotFound(f"{resource} {key} not found")
            rv = (obj.get("metadata") or {}).get("resourceVersion")
            if rv and rv != existing["metadata"].get("resourceVersion"):
                raise Conflict(
                    f"{resource} {key}: resourceVersion mismatch "
                    f"({rv} != {existing['metadata'].get('resourceVersion')})"
                )
            obj["metadata"]["uid"] = existing["metadata"]["uid"]
            obj["metadata"].setdefault(
                "creationTimestamp", existing["metadata"].get("creationTimestamp")
            )
            if existing["metadata"].get("deletionTimestamp"):
                obj["metadata"]["deletionTimestamp"] = existing["metadata"]["deletionTimestamp"]
            if obj.get("spec") != existing.get("spec"):
                obj["metadata"]["generation"] = existing["metadata"].get("generation", 1) + 1
            else:
                obj["metadata"]["generation"] = existing["metadata"].get("generation", 1)
            self._bump(obj)
            if obj["metadata"].get("deletionTimestamp") and not obj["metadata"].get("finalizers"):
                del table[key]
                self._notify(resource, "DELETED", obj)
                return copy.deepcopy(obj)
            table[key] = obj
            self._notify(resource, "MODIFIED", obj)
            return copy.deepcopy(obj)

    def patch(
        self, resource: str, name: str, patch: Dict[str, Any], namespace: str = ""
    ) -> Dict[str, Any]:
        """JSON merge patch (RFC 7386)."""
        with self._lock:
            existing = self._store.get(resource, {}).get((namespace, name))
            if existing is None:
                raise NotFound(f"{resource} {namespace}/{name} not found")
            merged = _merge_patch(copy.deepcopy(existing), patch)
            merged["metadata"]["name"] = name
            merged["metadata"]["namespace"] = namespace
            merged["metadata"]["resourceVersion"] = existing["metadata"]["resourceVersion"]
            return self.update(resource, merged)

    def delete(self, resource: str, name: str, namespace: str = "") -> None:
        with self._lock:
            table = self._store.get(resource, {})
            obj = table.get((namespace, name))
            if obj is None:
                raise NotFound(f"{resource} {namespace}/{name} not found")
            if obj["metadata"].get("finalizers"):
                if not obj["metadata"].get("deletionTimestamp"):
                    obj["metadata"]["deletionTimestamp"] = time.strftime(
                        "%Y-%m-%dT%H:%M:%SZ", time.gmtime()
                    )
                    self._bump(obj)
                    self._notify(resource, "MODIFIED", obj)
                return
            del table[(namespace, name)]
            self._bump(obj)
            self._notify(resource, "DELETED", obj)

    def list_with_rv(
        self,
        resource: str,
        namespace: Optional[str] = None,
        selector: Optional[Dict[str, str]] = None,
    ) -> Tuple[List[Dict[str, Any]], str]:
        """list + the collection resourceVersion a subsequent watch resumes
        from (k8s ListMeta.resourceVersion semantics)."""
        with self._lock:
            return self.list(resource, namespace, selector), str(self._rv)

    def watch(
        self,
        resource: str,
        namespace: Optional[str] = None,
        selector: Optional[Dict[str, str]] = None,
        send_initial: bool = True,
        resource_version: Optional[str] = None,
        bookmark_interval: Optional[float] = None,
    ) -> Watch:
        """Open a watch stream.

        * ``resource_version=None`` (legacy mode): current objects are
          replayed as ADDED when ``send_initial``.
        * ``resource_version="<rv>"``: resume — replay retained history
          events with rv strictly greater, then go live. Raises ``Gone``
          (410) when the requested rv precedes the retained window, exactly
          like a real apiserver's watch cache.
        """
        with self._lock:
            w = Watch(self, resource, namespace, selector,
                      bookmark_interval=bookmark_interval)
            if resource_version not in (None, ""):
                try:
                    rv = int(resource_version)
                except ValueError:
                    raise ApiError(400, f"invalid resourceVersion {resource_version!r}")
                if rv < self._evicted_rv.get(resource, 0):
                    raise Gone(
                        f"too old resource version: {rv} "
                        f"(oldest retained: {self._evicted_rv.get(resource, 0) + 1})"
                    )
                self._watches.setdefault(resource, []).append(w)
                for ev_rv, type_, obj in self._history.get(resource, []):
                    if ev_rv > rv:
                        w._deliver(WatchEvent(type_, copy.deepcopy(obj)))
            else:
                self._watches.setdefault(resource, []).append(w)
                if send_initial:
                    for obj in self.list(resource, namespace, selector):
                        w._q.put(WatchEvent("ADDED", obj))
            return w


def _merge_patch(target: Any, patch: Any) -> Any:
    if not isinstance(patch, dict):
        return copy.deepcopy(patch)
    if not isinstance(target, dict):
        target = {}
    for k, v in patch.items():
        if v is None:
            target.pop(k, None)
        else:
            target[k] = _merge_patch(target.get(k), v)
    return target
