"""Kubernetes API client interface.

Two implementations share one interface:

* ``FakeClient`` — wraps the in-memory ``FakeApiServer`` (CPU CI, bench),
* ``HttpClient`` — a real API-server client over HTTPS (httpx), speaking the
  standard conventions: GET/LIST with labelSelector, POST, PUT with
  optimistic concurrency, JSON merge-PATCH, DELETE, and chunked watch
  streams. In-cluster config comes from the service-account token
  (``/var/run/secrets/kubernetes.io/serviceaccount``), matching how the
  reference's controllers connect (``pkg/flags/kubeclient.go:44-72``).

Resource names are the plural REST path segments (``computedomains``,
``computedomaincliques``, ``resourceclaims``, ``resourceclaimtemplates``,
``resourceslices``, ``daemonsets``, ``pods``, ``nodes``, ``leases``).
"""

from __future__ import annotations

import json
import logging
import os
import threading
from typing import Any, Dict, List, Optional

from .fakeserver import ApiError, Conflict, FakeApiServer, NotFound, Watch

# resource -> (apiGroupVersion, namespaced)
RESOURCE_INFO: Dict[str, tuple] = {
    "computedomains": ("resource.amd.com/v1beta1", True),
    "computedomaincliques": ("resource.amd.com/v1beta1", False),
    "resourceclaims": ("resource.k8s.io/v1beta1", True),
    "resourceclaimtemplates": ("resource.k8s.io/v1beta1", True),
    "resourceslices": ("resource.k8s.io/v1beta1", False),
    "deviceclasses": ("resource.k8s.io/v1beta1", False),
    "daemonsets": ("apps/v1", True),
    "deployments": ("apps/v1", True),
    "pods": ("v1", True),
    "nodes": ("v1", False),
    "leases": ("coordination.k8s.io/v1", True),
    "jobs": ("batch/v1", True),
    "events": ("v1", True),
}


def retry_on_conflict(fn, attempts: int = 5, base_delay: float = 0.01):
    """Re-run ``fn()`` while it raises 409 ``Conflict``, sleeping between
    tries (client-go ``retry.RetryOnConflict`` with ``DefaultRetry``:
    5 steps, 10 ms base, jittered). ``fn`` must re-read the object each
    attempt so it picks up the fresh resourceVersion. Raises the final
    ``Conflict`` if every attempt loses the race."""
    import random
    import time

    last: Optional[Conflict] = None
    for i in range(attempts):
        try:
            return fn()
        except Conflict as e:
            last = e
            time.sleep(base_delay * (1 + random.random() * 0.1) * (i + 1))
    assert last is not None
    raise last


class Client:
    """Interface; see FakeClient / HttpClient."""

    def create(self, resource: str, obj: Dict[str, Any]) -> Dict[str, Any]:
        raise NotImplementedError

    def get(self, resource: str, name: str, namespace: str = "") -> Dict[str, Any]:
        raise NotImplementedError

    def list(self, resource: str, namespace: Optional[str] = None,
             selector: Optional[Dict[str, str]] = None) -> List[Dict[str, Any]]:
        raise NotImplementedError

    def update(self, resource: str, obj: Dict[str, Any]) -> Dict[str, Any]:
        raise NotImplementedError

    def patch(self, resource: str, name: str, patch: Dict[str, Any],
              namespace: str = "") -> Dict[str, Any]:
        raise NotImplementedError

    def delete(self, resource: str, name: str, namespace: str = "") -> None:
        raise NotImplementedError

    def watch(self, resource: str, namespace: Optional[str] = None,
              selector: Optional[Dict[str, str]] = None,
              resource_version: Optional[str] = None,
              allow_bookmarks: bool = False) -> Watch:
        """Watch; with ``resource_version`` the stream resumes from that RV
        (410 Gone when too old); ``allow_bookmarks`` requests periodic
        BOOKMARK events on idle streams."""
        raise NotImplementedError

    def list_with_rv(self, resource: str, namespace: Optional[str] = None,
                     selector: Optional[Dict[str, str]] = None):
        """(items, collection resourceVersion). Default shim returns rv=""
        for implementations without watch-cache semantics; the informer
        falls back to legacy replay mode then."""
        return self.list(resource, namespace, selector), ""

    def server_version(self) -> tuple:
        """(major, minor) of the API server; used to key feature availability
        such as KEP-4815 partitionable slices on k8s >= 1.33 (the reference's
        shouldUseSplitResourceSlices probe, driver.go:574-603)."""
        return (1, 33)

    # -- conveniences shared by implementations ---------------------------

    def get_or_none(self, resource: str, name: str, namespace: str = ""):
        try:
            return self.get(resource, name, namespace)
        except NotFound:
            return None

    def apply(self, resource: str, obj: Dict[str, Any]) -> Dict[str, Any]:
        """Create-or-update by name (server-side-apply-lite). Update races
        (another writer bumped resourceVersion between our read and PUT)
        are retried with a fresh read, client-go style."""
        md = obj.get("metadata") or {}

        def attempt():
            existing = self.get_or_none(
                resource, md.get("name", ""), md.get("namespace", "")
            )
            if existing is None:
                return self.create(resource, obj)
            fresh = dict(obj)
            fresh.setdefault("metadata", {})["resourceVersion"] = existing["metadata"][
                "resourceVersion"
            ]
            return self.update(resource, fresh)

        return retry_on_conflict(attempt)

    def add_finalizer(self, resource: str, name: str, namespace: str, finalizer: str):
        obj = self.get(resource, name, namespace)
        fins = obj["metadata"].get("finalizers") or []
        if finalizer in fins:
            return obj
        fins.append(finalizer)
        return self.patch(resource, name, {"metadata": {"finalizers": fins}}, namespace)

    def remove_finalizer(self, resource: str, name: str, namespace: str, finalizer: str):
        obj = self.get_or_none(resource, name, namespace)
        if obj is None:
            return None
        fins = [f for f in (obj["metadata"].get("finalizers") or []) if f != finalizer]
        return self.patch(
            resource, name, {"metadata": {"finalizers": fins or None}}, namespace
        )


class FakeClient(Client):
    # idle-bookmark cadence for allow_bookmarks watches (client-go asks the
    # apiserver for bookmarks roughly every few seconds to minutes)
    bookmark_interval = 5.0

    def __init__(self, server: Optional[FakeApiServer] = None):
        self.server = server or FakeApiServer()

    def create(self, resource, obj):
        return self.server.create(resource, obj)

    def get(self, resource, name, namespace=""):
        return self.server.get(resource, name, namespace)

    def list(self, resource, namespace=None, selector=None):
        return self.server.list(resource, namespace, selector)

    def update(self, resource, obj):
        return self.server.update(resource, obj)

    def patch(self, resource, name, patch, namespace=""):
        return self.server.patch(resource, name, patch, namespace)

    def delete(self, resource, name, namespace=""):
        return self.server.delete(resource, name, namespace)

    def watch(self, resource, namespace=None, selector=None,
              resource_version=None, allow_bookmarks=False):
        return self.server.watch(
            resource, namespace, selector,
            resource_version=resource_version,
            bookmark_interval=self.bookmark_interval if allow_bookmarks else None,
        )

    def list_with_rv(self, resource, namespace=None, selector=None):
        return self.server.list_with_rv(resource, namespace, selector)

    def server_version(self):
        return getattr(self.server, "version", (1, 33))


class HttpClient(Client):
    """Real API-server client. QPS limits follow the reference's defaults
    (QPS 5 / burst 10, kubeclient.go:53-68) via a token bucket."""

    SA_DIR = "/var/run/secrets/kubernetes.io/serviceaccount"

    def __init__(
        self,
        base_url: str = "",
        token: str = "",
        ca_cert: str = "",
        qps: float = 0.0,
        burst: int = 0,
    ):
        # reference defaults: QPS 5 / burst 10 (kubeclient.go:53-68),
        # overridable via env for high-churn deployments
        qps = qps or float(os.environ.get("AMDDRA_KUBE_QPS", "5"))
        burst = burst or int(os.environ.get("AMDDRA_KUBE_BURST", "10"))
        import httpx

        self.base_url = (
            base_url or os.environ.get("AMDDRA_API_SERVER") or self._in_cluster_url()
        )
        token = token or self._read_sa(os.path.join(self.SA_DIR, "token"))
        ca = ca_cert or os.path.join(self.SA_DIR, "ca.crt")
        headers = {"Accept": "application/json", "Content-Type": "application/json"}
        if token:
            headers["Authorization"] = f"Bearer {token}"
        self._http = httpx.Client(
            base_url=self.base_url,
            headers=headers,
            verify=ca if os.path.exists(ca) else False,
            timeout=30.0,
        )
        from ..utils.workqueue import RateLimiter

        self._limiter = RateLimiter(base_delay=0.0, max_delay=0.0, qps=qps, burst=burst)
        self._lock = threading.Lock()

    @staticmethod
    def _in_cluster_url() -> str:
        host = os.environ.get("KUBERNETES_SERVICE_HOST", "kubernetes.default.svc")
        port = os.environ.get("KUBERNETES_SERVICE_PORT", "443")
        return f"https://{host}:{port}"

    @staticmethod
    def _read_sa(path: str) -> str:
        try:
            with open(path) as f:
                return f.read().strip()
        except OSError:
            return ""

    def _path(self, resource: str, namespace: Optional[str], name: str = "") -> str:
        gv, namespaced = RESOURCE_INFO[resource]
        prefix = f"/api/{gv}" if "/" not in gv else f"/apis/{gv}"
        p = prefix
        if namespaced and namespace:
            p += f"/namespaces/{namespace}"
        p += f"/{resource}"
        if name:
            p += f"/{name}"
        return p

    def _throttle(self):
        import time

        d = self._limiter.when("req")
        self._limiter.forget("req")
        if d > 0:
            time.sleep(d)

    def _check(self, r):
        if r.status_code == 404:
            raise NotFound(r.text)
        if r.status_code == 409:
            raise Conflict(r.text)
        if r.status_code >= 400:
            raise ApiError(r.status_code, r.text)
        return r.json() if r.content else None

    def _get_with_retry(self, path: str, params=None, attempts: int = 4):
        """GET is idempotent: retry 429 (honoring Retry-After, capped) and
        5xx with linear backoff, mirroring client-go's transport-level
        handling of apiserver overload (priority&fairness shedding)."""
        import time

        for i in range(attempts):
            r = self._http.get(path, params=params)
            if r.status_code == 429 or r.status_code >= 500:
                if i == attempts - 1:
                    break
                delay = 0.1 * (i + 1)
                ra = r.headers.get("Retry-After")
                if ra:
                    try:
                        delay = min(float(ra), 5.0)
                    except ValueError:
                        pass
                time.sleep(delay)
                continue
            break
        return self._check(r)

    def create(self, resource, obj):
        self._throttle()
        ns = (obj.get("metadata") or {}).get("namespace", "")
        return self._check(self._http.post(self._path(resource, ns), json=obj))

    def get(self, resource, name, namespace=""):
        self._throttle()
        return self._get_with_retry(self._path(resource, namespace, name))

    def list(self, resource, namespace=None, selector=None):
        self._throttle()
        params = {}
        if selector:
            params["labelSelector"] = ",".join(f"{k}={v}" for k, v in selector.items())
        data = self._get_with_retry(self._path(resource, namespace), params=params)
        return data.get("items", [])

    def list_with_rv(self, resource, namespace=None, selector=None):
        self._throttle()
        params = {}
        if selector:
            params["labelSelector"] = ",".join(f"{k}={v}" for k, v in selector.items())
        data = self._get_with_retry(self._path(resource, namespace), params=params)
        rv = ((data.get("metadata") or {}).get("resourceVersion")) or ""
        return data.get("items", []), rv

    def update(self, resource, obj):
        self._throttle()
        md = obj.get("metadata") or {}
        return self._check(
            self._http.put(
                self._path(resource, md.get("namespace", ""), md.get("name", "")), json=obj
            )
        )

    def patch(self, resource, name, patch, namespace=""):
        self._throttle()
        return self._check(
            self._http.request(
                "PATCH",
                self._path(resource, namespace, name),
                content=json.dumps(patch),
                headers={"Content-Type": "application/merge-patch+json"},
            )
        )

    def delete(self, resource, name, namespace=""):
        self._throttle()
        self._check(self._http.delete(self._path(resource, namespace, name)))

    def server_version(self):
        try:
            data = self._check(self._http.get("/version"))
            return (int(data.get("major", "1")),
                    int("".join(ch for ch in data.get("minor", "33") if ch.isdigit()) or 33))
        except Exception:
            return (1, 33)

    def watch(self, resource, namespace=None, selector=None,
              resource_version=None, allow_bookmarks=False):
        """Streamed watch; returns a Watch-like iterator thread. A server
        410 (resourceVersion too old) surfaces as an ERROR WatchEvent whose
        object carries code=410, which the informer turns into a relist."""
        from .fakeserver import WatchEvent

        w = Watch.__new__(Watch)
        import queue as _q

        w._q = _q.Queue()
        w._stopped = False
        w._server = None
        w._resource = resource
        w._bookmark_interval = None
        w._last_emit = 0.0

        def run():
            # bounded watch (client-go style): ask the server to close the
            # stream after 5 min (timeoutSeconds) and cap the client read
            # slightly above it — a silently dead connection otherwise hangs
            # the informer forever instead of triggering its relist
            import httpx

            wt = os.environ.get("AMDDRA_WATCH_TIMEOUT", "300")
            params = {"watch": "true", "timeoutSeconds": wt}
            if resource_version not in (None, ""):
                params["resourceVersion"] = str(resource_version)
            if allow_bookmarks:
                params["allowWatchBookmarks"] = "true"
            if selector:
                params["labelSelector"] = ",".join(f"{k}={v}" for k, v in selector.items())
            try:
                with self._http.stream(
                    "GET", self._path(resource, namespace), params=params,
                    timeout=httpx.Timeout(connect=30.0, read=float(wt) + 30.0,
                                          write=30.0, pool=30.0),
                ) as r:
                    if r.status_code == 410:
                        w._q.put(WatchEvent("ERROR", {"code": 410}))
                        return
                    if r.status_code >= 400:
                        w._q.put(WatchEvent("ERROR", {"code": r.status_code}))
                        return
                    for line in r.iter_lines():
                        if w._stopped:
                            return
                        if not line:
                            continue
                        ev = json.loads(line)
                        w._q.put(WatchEvent(ev["type"], ev["object"]))
            except Exception:
                # stream broke: the informer relists + rewatches
                logging.getLogger("amddra.client").debug(
                    "watch stream ended", exc_info=True)
            finally:
                w._q.put(None)

        t = threading.Thread(target=run, daemon=True)
        t.start()

        def stop():
            w._stopped = True
            w._q.put(None)

        w.stop = stop
        return w
