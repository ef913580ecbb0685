"""Informer: list+watch with a local cache and event handlers.

The client-go informer analog the reference's controllers are built on
(``compute-domain-controller/computedomain.go:128-138`` etc.): a background
thread keeps a keyed cache in sync and dispatches add/update/delete handlers;
``wait_for_sync`` gates controller startup.

Reflector semantics follow client-go's watch-cache model: one list
establishes the collection resourceVersion, the watch resumes FROM that rv
(no replay, no dedupe heuristics), BOOKMARK events advance the last-seen rv
on idle streams, a closed stream re-watches from the last rv without
relisting, and a **410 Gone** (rv fell out of the server's retained window)
triggers a full relist that reconciles the cache — including DELETEDs that
happened while disconnected. Clients without rv support (``list_with_rv``
returns rv="") fall back to the legacy replay+dedupe mode.
"""

from __future__ import annotations

import logging
import threading
from typing import Any, Callable, Dict, List, Optional

from .client import Client

logger = logging.getLogger("amddra.informer")

Handler = Callable[[str, Dict[str, Any]], None]  # (event_type, object)


def obj_key(obj: Dict[str, Any]) -> str:
    md = obj.get("metadata") or {}
    ns = md.get("namespace", "")
    return f"{ns}/{md.get('name', '')}" if ns else md.get("name", "")


class Informer:
    def __init__(
        self,
        client: Client,
        resource: str,
        namespace: Optional[str] = None,
        selector: Optional[Dict[str, str]] = None,
    ):
        self.client = client
        self.resource = resource
        self.namespace = namespace
        self.selector = selector
        self._cache: Dict[str, Dict[str, Any]] = {}
        self._lock = threading.RLock()
        self._handlers: List[Handler] = []
        self._synced = threading.Event()
        self._stop = threading.Event()
        self._watch = None
        self._thread: Optional[threading.Thread] = None
        self._uid_index: Dict[str, str] = {}  # uid -> key
        # last-seen collection/event resourceVersion (advanced by events AND
        # idle bookmarks; the rv a re-watch resumes from)
        self.last_resource_version: str = ""

    def add_handler(self, handler: Handler) -> None:
        with self._lock:
            self._handlers.append(handler)
            for obj in self._cache.values():
                handler("ADDED", obj)

    def start(self) -> "Informer":
        self._thread = threading.Thread(target=self._run, daemon=True, name=f"inf-{self.resource}")
        self._thread.start()
        return self

    def _run(self) -> None:
        from .fakeserver import ApiError

        while not self._stop.is_set():
            try:
                objs, rv = self.client.list_with_rv(
                    self.resource, self.namespace, self.selector
                )
                self._reconcile(objs)
                self._synced.set()
                self.last_resource_version = rv
                if not rv:
                    # legacy client: replay-mode watch with rv-dedupe
                    self._legacy_watch_loop()
                    continue
                # watch loop: resume from rv; stream close -> re-watch from
                # the last-seen rv; 410 -> break to relist
                while not self._stop.is_set():
                    try:
                        self._watch = self.client.watch(
                            self.resource, self.namespace, self.selector,
                            resource_version=self.last_resource_version,
                            allow_bookmarks=True,
                        )
                    except ApiError as e:
                        if e.code == 410:
                            logger.info(
                                "informer %s: rv %s too old (410); relisting",
                                self.resource, self.last_resource_version,
                            )
                            break
                        raise
                    gone = False
                    for ev in self._watch:
                        if self._stop.is_set():
                            return
                        if ev.type == "BOOKMARK":
                            nrv = (ev.object.get("metadata") or {}).get("resourceVersion")
                            if nrv:
                                self.last_resource_version = nrv
                            continue
                        if ev.type == "ERROR":
                            if (ev.object or {}).get("code") == 410:
                                gone = True
                            break
                        self._apply_event(ev)
                        nrv = (ev.object.get("metadata") or {}).get("resourceVersion")
                        if nrv:
                            self.last_resource_version = nrv
                    if gone:
                        logger.info(
                            "informer %s: watch stream returned 410; relisting",
                            self.resource,
                        )
                        break
                    # stream closed normally (server timeoutSeconds or
                    # transient break): re-watch from last rv, no relist
            except Exception:
                if not self._stop.is_set():
                    logger.exception("informer %s: relisting after error", self.resource)
                    self._stop.wait(0.5)

    def _reconcile(self, objs) -> None:
        """Replace the cache with a fresh list, emitting the delta events —
        including DELETEDs for objects that vanished while disconnected."""
        with self._lock:
            old_keys = set(self._cache)
            new_keys = set()
            for obj in objs:
                k = obj_key(obj)
                new_keys.add(k)
                prev = self._cache.get(k)
                if prev is not None and prev.get("metadata", {}).get(
                    "resourceVersion"
                ) == obj.get("metadata", {}).get("resourceVersion"):
                    continue  # unchanged: no event
                self._cache[k] = obj
                self._index(obj, k)
                self._dispatch("ADDED" if prev is None else "MODIFIED", obj)
            for k in old_keys - new_keys:
                gone = self._cache.pop(k)
                self._unindex(gone)
                self._dispatch("DELETED", gone)

    def _apply_event(self, ev) -> None:
        k = obj_key(ev.object)
        with self._lock:
            if ev.type == "DELETED":
                self._cache.pop(k, None)
                self._unindex(ev.object)
            else:
                self._cache[k] = ev.object
                self._index(ev.object, k)
            self._dispatch(ev.type, ev.object)

    def _legacy_watch_loop(self) -> None:
        """Replay-mode watch for clients without resourceVersion support."""
        self._watch = self.client.watch(self.resource, self.namespace, self.selector)
        for ev in self._watch:
            if self._stop.is_set():
                return
            k = obj_key(ev.object)
            with self._lock:
                if ev.type == "DELETED":
                    self._cache.pop(k, None)
                    self._unindex(ev.object)
                else:
                    # replayed ADDED on (re)watch; dedupe by resourceVersion
                    prev = self._cache.get(k)
                    if prev is not None and prev.get("metadata", {}).get(
                        "resourceVersion"
                    ) == ev.object.get("metadata", {}).get("resourceVersion"):
                        continue
                    self._cache[k] = ev.object
                    self._index(ev.object, k)
                self._dispatch(ev.type, ev.object)

    def _index(self, obj, key):
        uid = (obj.get("metadata") or {}).get("uid")
        if uid:
            self._uid_index[uid] = key

    def _unindex(self, obj):
        uid = (obj.get("metadata") or {}).get("uid")
        if uid:
            self._uid_index.pop(uid, None)

    def _dispatch(self, type_: str, obj: Dict[str, Any]) -> None:
        for h in list(self._handlers):
            try:
                h(type_, obj)
            except Exception:
                logger.exception("informer handler failed for %s", self.resource)

    # -- cache access -------------------------------------------------------

    def wait_for_sync(self, timeout: float = 10.0) -> bool:
        return self._synced.wait(timeout)

    def get(self, key: str) -> Optional[Dict[str, Any]]:
        with self._lock:
            return self._cache.get(key)

    def get_by_uid(self, uid: str) -> Optional[Dict[str, Any]]:
        with self._lock:
            k = self._uid_index.get(uid)
            return self._cache.get(k) if k else None

    def items(self) -> List[Dict[str, Any]]:
        with self._lock:
            return list(self._cache.values())

    def stop(self) -> None:
        self._stop.set()
        if self._watch is not None:
            self._watch.stop()
        self._synced.set()
