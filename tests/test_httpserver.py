"""Mini API server (HTTP facade) + HttpClient round trips: the real-cluster
client path exercised in CI, including watch streaming, informers over HTTP,
and a kubelet plugin running as a SEPARATE PROCESS against the server."""

import os
import subprocess
import sys
import time

import pytest

from k8s_dra_driver_gpu_amd.k8s.client import HttpClient
from k8s_dra_driver_gpu_amd.k8s.fakeserver import ApiError, NotFound
from k8s_dra_driver_gpu_amd.k8s.httpserver import MiniApiServer
from k8s_dra_driver_gpu_amd.k8s.informer import Informer

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


@pytest.fixture
def server():
    srv = MiniApiServer()
    srv.start()
    yield srv
    srv.stop()


@pytest.fixture
def client(server):
    return HttpClient(base_url=f"http://127.0.0.1:{server.port}")


def cd(name="cd1", ns="default"):
    return {
        "apiVersion": "resource.amd.com/v1beta1",
        "kind": "ComputeDomain",
        "metadata": {"name": name, "namespace": ns},
        "spec": {"numNodes": 1},
    }


class TestHttpClient:
    def test_crud_round_trip(self, client):
        obj = client.create("computedomains", cd())
        assert obj["metadata"]["uid"]
        got = client.get("computedomains", "cd1", "default")
        assert got["spec"]["numNodes"] == 1
        got["spec"]["numNodes"] = 2
        updated = client.update("computedomains", got)
        assert updated["spec"]["numNodes"] == 2
        patched = client.patch("computedomains", "cd1",
                               {"status": {"status": "Ready"}}, "default")
        assert patched["status"]["status"] == "Ready"
        assert len(client.list("computedomains")) == 1
        client.delete("computedomains", "cd1", "default")
        assert client.get_or_none("computedomains", "cd1", "default") is None

    def test_conflict_and_notfound(self, client):
        a = client.create("computedomains", cd())
        b = client.get("computedomains", "cd1", "default")
        a["spec"]["numNodes"] = 3
        client.update("computedomains", a)
        b["spec"]["numNodes"] = 4
        with pytest.raises(ApiError):
            client.update("computedomains", b)
        with pytest.raises(NotFound):
            client.get("computedomains", "nope", "default")

    def test_cluster_scoped_and_selector(self, client):
        client.create("computedomaincliques",
                      {"metadata": {"name": "u.h.0", "labels": {"a": "b"}}, "daemons": []})
        client.create("computedomaincliques",
                      {"metadata": {"name": "u.h.1", "labels": {"a": "c"}}, "daemons": []})
        assert len(client.list("computedomaincliques")) == 2
        sel = client.list("computedomaincliques", selector={"a": "b"})
        assert [o["metadata"]["name"] for o in sel] == ["u.h.0"]

    def test_watch_stream(self, client):
        w = client.watch("computedomains")
        time.sleep(0.2)
        client.create("computedomains", cd("live"))
        deadline = time.monotonic() + 5
        seen = None
        while time.monotonic() < deadline:
            ev = w.next(1.0)
            if ev is not None and ev.object["metadata"]["name"] == "live":
                seen = ev
                break
        assert seen is not None and seen.type == "ADDED"
        w.stop()

    def test_informer_over_http(self, client):
        client.create("computedomains", cd("pre"))
        inf = Informer(client, "computedomains").start()
        assert inf.wait_for_sync(10.0)
        assert inf.get("default/pre") is not None
        client.create("computedomains", cd("post"))
        deadline = time.monotonic() + 5
        while time.monotonic() < deadline and inf.get("default/post") is None:
            time.sleep(0.05)
        assert inf.get("default/post") is not None
        inf.stop()

    def test_server_version(self, client):
        assert client.server_version() >= (1, 33)


class TestSeparateProcessPlugin:
    def test_gpu_plugin_against_http_server(self, server, client, tmp_path):
        """The GPU kubelet plugin as its own process publishing slices to the
        mini API server over HTTP — the full multi-process deployment shape."""
        from k8s_dra_driver_gpu_amd.device.mock import MockTree

        tree = MockTree(root=str(tmp_path / "mock"), num_gpus=2)
        tree.setup()
        env = dict(os.environ)
        env.update({
            "PYTHONPATH": REPO,
            "AMDDRA_API_SERVER": f"http://127.0.0.1:{server.port}",
            "AMDDRA_SYSFS_ROOT": tree.sysfs_root,
            "AMDDRA_DEV_ROOT": tree.dev_root,
            "PLUGIN_DIR": str(tmp_path / "plugin"),
            "PLUGINS_REGISTRY_DIR": str(tmp_path / "registry"),
            "CDI_ROOT": str(tmp_path / "cdi"),
            "NODE_NAME": "proc-node",
            "HEALTHCHECK_PORT": "0",
        })
        proc = subprocess.Popen(
            [sys.executable, "-m", "k8s_dra_driver_gpu_amd.cmd.gpu_kubelet_plugin"],
            env=env, cwd=REPO, stdout=subprocess.PIPE, stderr=subprocess.STDOUT, text=True,
        )
        try:
            deadline = time.monotonic() + 20
            slices = []
            while time.monotonic() < deadline:
                assert proc.poll() is None, proc.stdout.read()
                slices = client.list("resourceslices")
                if slices:
                    break
                time.sleep(0.2)
            assert slices, "plugin did not publish ResourceSlices over HTTP"
            assert slices[0]["spec"]["nodeName"] == "proc-node"
            names = [d["name"] for d in slices[0]["spec"]["devices"]]
            assert "gpu-0" in names
            # partitionable auto mode resolved via /version (>=1.33)
            assert any("-cpx-" in n for n in names)
        finally:
            proc.terminate()
            proc.wait(timeout=10)


class TestMultiProcessCluster:
    def test_apiserver_scheduler_plugin_processes(self, server, client, tmp_path):
        """The no-kind mock cluster: apiserver (in-proc) + scheduler stub +
        GPU plugin as separate processes; a claim created via HTTP gets
        allocated by the scheduler process and prepared by the plugin
        process over its gRPC socket."""
        from k8s_dra_driver_gpu_amd.device.mock import MockTree
        from k8s_dra_driver_gpu_amd.dra import api as dra
        import yaml

        from k8s_dra_driver_gpu_amd.utils.helmlite import chart_deviceclasses

        chart = os.path.join(REPO, "deployments", "helm", "amd-dra-driver")
        for doc in chart_deviceclasses(chart):
            client.create("deviceclasses", doc)
        tree = MockTree(root=str(tmp_path / "mock"), num_gpus=1)
        tree.setup()
        env = dict(os.environ)
        env.update({
            "PYTHONPATH": REPO,
            "AMDDRA_API_SERVER": f"http://127.0.0.1:{server.port}",
            "AMDDRA_SYSFS_ROOT": tree.sysfs_root,
            "AMDDRA_DEV_ROOT": tree.dev_root,
            "PLUGIN_DIR": str(tmp_path / "plugin"),
            "PLUGINS_REGISTRY_DIR": str(tmp_path / "registry"),
            "CDI_ROOT": str(tmp_path / "cdi"),
            "NODE_NAME": "mp-node",
        })
        plugin = subprocess.Popen(
            [sys.executable, "-m", "k8s_dra_driver_gpu_amd.cmd.gpu_kubelet_plugin"],
            env=env, cwd=REPO, stdout=subprocess.PIPE, stderr=subprocess.STDOUT, text=True,
        )
        sched = subprocess.Popen(
            [sys.executable, "-m", "k8s_dra_driver_gpu_amd.cmd.scheduler"],
            env=env, cwd=REPO, stdout=subprocess.PIPE, stderr=subprocess.STDOUT, text=True,
        )
        try:
            # wait for slice publication
            deadline = time.monotonic() + 20
            while time.monotonic() < deadline and not client.list("resourceslices"):
                time.sleep(0.2)
            assert client.list("resourceslices")
            claim = client.create("resourceclaims", {
                "apiVersion": "resource.k8s.io/v1beta1",
                "kind": "ResourceClaim",
                "metadata": {"name": "mp-claim", "namespace": "default"},
                "spec": {"devices": {"requests": [
                    {"name": "r0", "deviceClassName": "gpu.amd.com"}]}},
            })
            deadline = time.monotonic() + 20
            alloc = None
            while time.monotonic() < deadline:
                got = client.get("resourceclaims", "mp-claim", "default")
                alloc = (got.get("status") or {}).get("allocation")
                if alloc:
                    break
                time.sleep(0.2)
            assert alloc, "scheduler process did not allocate the claim"
            assert alloc["devices"]["results"][0]["device"] == "gpu-0"
            # kubelet-style prepare against the plugin process's socket
            cli = dra.DRAPluginClient(f"unix://{tmp_path / 'plugin' / 'dra.sock'}")
            uid = got["metadata"]["uid"]
            resp = cli.prepare([dra.Claim(namespace="default", name="mp-claim", uid=uid)])
            assert resp.claims[uid].error == "", resp.claims[uid].error
            cli.unprepare([dra.Claim(namespace="default", name="mp-claim", uid=uid)])
            cli.close()
            # free the GPU for the next scenario
            client.delete("resourceclaims", "mp-claim", "default")
            # DRAExtendedResource through the scheduler PROCESS: legacy
            # limits pod -> special claim allocated -> pod deletion GCs it
            client.create("pods", {
                "apiVersion": "v1", "kind": "Pod",
                "metadata": {"name": "legacy-mp", "namespace": "default"},
                "spec": {"containers": [{"name": "c", "resources": {
                    "limits": {"amd.com/gpu": 1}}}]},
            })
            deadline = time.monotonic() + 20
            ercs = None
            while time.monotonic() < deadline:
                pod = client.get("pods", "legacy-mp", "default")
                ercs = (pod.get("status") or {}).get("extendedResourceClaimStatus")
                if ercs:
                    break
                time.sleep(0.2)
            assert ercs, "scheduler process did not satisfy the legacy limit"
            cname = ercs["resourceClaimName"]
            got = client.get("resourceclaims", cname, "default")
            assert (got.get("status") or {}).get("allocation")
            client.delete("pods", "legacy-mp", "default")
            deadline = time.monotonic() + 20
            while time.monotonic() < deadline:
                if client.get_or_none("resourceclaims", cname, "default") is None:
                    break
                time.sleep(0.2)
            assert client.get_or_none("resourceclaims", cname, "default") is None, \
                "extended claim not GC'd after pod deletion"
        finally:
            for pr in (plugin, sched):
                pr.terminate()
                try:
                    pr.wait(timeout=10)
                except subprocess.TimeoutExpired:
                    pr.kill()


class TestWatchTimeout:
    def test_server_closes_watch_after_timeout_seconds(self, server):
        """Standard watch semantics: the server ends the stream after
        timeoutSeconds; the client's iterator terminates (informers then
        relist + rewatch)."""
        import httpx

        t0 = time.time()
        lines = 0
        with httpx.Client(base_url=f"http://127.0.0.1:{server.port}") as http:
            with http.stream(
                "GET", "/apis/resource.amd.com/v1beta1/computedomains",
                params={"watch": "true", "timeoutSeconds": "1"}, timeout=10.0,
            ) as r:
                for _line in r.iter_lines():
                    lines += 1
        took = time.time() - t0
        assert took < 6, took  # closed by the server, not the client timeout


class TestWatchRecycleChurn:
    def test_allocation_survives_rapid_watch_recycles(self, server, client, tmp_path):
        """Claims keep allocating while the scheduler's watch streams recycle
        every second (the missed-DELETED race found by the 20-min soak is
        healed by the periodic resync within SCHED_RESYNC_INTERVAL)."""
        import yaml

        from k8s_dra_driver_gpu_amd.utils.helmlite import chart_deviceclasses

        chart = os.path.join(REPO, "deployments", "helm", "amd-dra-driver")
        for doc in chart_deviceclasses(chart):
            client.create("deviceclasses", doc)
        client.create("resourceslices", {
            "apiVersion": "resource.k8s.io/v1beta1", "kind": "ResourceSlice",
            "metadata": {"name": "wr-gpu"},
            "spec": {"driver": "gpu.amd.com",
                     "pool": {"name": "wr", "generation": 1,
                              "resourceSliceCount": 1},
                     "nodeName": "wr",
                     "devices": [{"name": "gpu-0", "basic": {"attributes": {
                         "type": {"string": "gpu"}}}}]},
        })
        env = dict(os.environ)
        env.update({
            "PYTHONPATH": REPO,
            "AMDDRA_API_SERVER": f"http://127.0.0.1:{server.port}",
            "AMDDRA_WATCH_TIMEOUT": "1",
            "SCHED_RESYNC_INTERVAL": "0.3",
            "SCHED_POLL_INTERVAL": "0.05",
            "AMDDRA_KUBE_QPS": "2000", "AMDDRA_KUBE_BURST": "2000",
        })
        sched = subprocess.Popen(
            [sys.executable, "-m", "k8s_dra_driver_gpu_amd.cmd.scheduler"],
            env=env, cwd=REPO, stdout=subprocess.DEVNULL,
            stderr=subprocess.DEVNULL,
        )
        try:
            ok_cycles = 0
            deadline = time.time() + 12
            i = 0
            while time.time() < deadline:
                name = f"wr-{i}"
                i += 1
                client.create("resourceclaims", {
                    "apiVersion": "resource.k8s.io/v1beta1",
                    "kind": "ResourceClaim",
                    "metadata": {"name": name, "namespace": "default"},
                    "spec": {"devices": {"requests": [
                        {"name": "r0", "deviceClassName": "gpu.amd.com"}]}},
                })
                t0 = time.time()
                while time.time() - t0 < 10:
                    got = client.get("resourceclaims", name, "default")
                    if (got.get("status") or {}).get("allocation"):
                        break
                    time.sleep(0.01)
                else:
                    raise AssertionError(
                        f"claim {name} not allocated within 10s "
                        f"(cycle {ok_cycles}, sched alive={sched.poll() is None})")
                client.delete("resourceclaims", name, "default")
                ok_cycles += 1
            assert ok_cycles > 10, ok_cycles
            assert sched.poll() is None
        finally:
            sched.terminate()
            try:
                sched.wait(timeout=10)
            except subprocess.TimeoutExpired:
                sched.kill()


class TestTransientRetry:
    """GETs are retried through apiserver overload shedding (429 with
    Retry-After, 5xx) — the client-go transport analog."""

    def _flaky_server(self, statuses):
        import http.server
        import json as _json
        import threading

        seen = []

        class H(http.server.BaseHTTPRequestHandler):
            def do_GET(self):
                seen.append(self.path)
                if len(seen) <= len(statuses):
                    code = statuses[len(seen) - 1]
                    self.send_response(code)
                    self.send_header("Retry-After", "0.01")
                    self.end_headers()
                    return
                body = _json.dumps({"metadata": {"name": "cd", "namespace":
                                    "default", "resourceVersion": "1"},
                                    "spec": {"numNodes": 1}}).encode()
                self.send_response(200)
                self.send_header("Content-Type", "application/json")
                self.send_header("Content-Length", str(len(body)))
                self.end_headers()
                self.wfile.write(body)

            def log_message(self, *a):
                pass

        srv = http.server.ThreadingHTTPServer(("127.0.0.1", 0), H)
        threading.Thread(target=srv.serve_forever, daemon=True).start()
        return srv, seen

    def test_get_retries_429_and_5xx(self):
        srv, seen = self._flaky_server([429, 503])
        try:
            c = HttpClient(base_url=f"http://127.0.0.1:{srv.server_address[1]}",
                           qps=10000, burst=10000)
            got = c.get("computedomains", "cd", "default")
            assert got["spec"]["numNodes"] == 1
            assert len(seen) == 3  # 429 -> 503 -> 200
        finally:
            srv.shutdown()

    def test_get_gives_up_after_persistent_503(self):
        srv, seen = self._flaky_server([503] * 10)
        try:
            c = HttpClient(base_url=f"http://127.0.0.1:{srv.server_address[1]}",
                           qps=10000, burst=10000)
            with pytest.raises(ApiError):
                c.get("computedomains", "cd", "default")
            assert len(seen) == 4  # bounded attempts
        finally:
            srv.shutdown()

    def test_stale_put_raises_conflict_type(self, client):
        from k8s_dra_driver_gpu_amd.k8s.fakeserver import Conflict

        a = client.create("computedomains", cd("cdc"))
        fresh = client.get("computedomains", "cdc", "default")
        fresh["spec"]["numNodes"] = 2
        client.update("computedomains", fresh)
        a["spec"]["numNodes"] = 3
        with pytest.raises(Conflict):
            client.update("computedomains", a)


class TestExtendedResourceOverHttp:
    """schedule_extended_resources against the HTTP facade: pods list +
    status patch + claim create all over the wire (the path the scheduler
    PROCESS uses in a real cluster)."""

    def test_pod_with_legacy_limit(self, server, client):
        from k8s_dra_driver_gpu_amd.k8s.scheduler import SchedulerStub

        client.create("deviceclasses", {
            "apiVersion": "resource.k8s.io/v1beta1", "kind": "DeviceClass",
            "metadata": {"name": "gpu.amd.com"},
            "spec": {"extendedResourceName": "amd.com/gpu", "selectors": [
                {"cel": {"expression": 'device.driver == "gpu.amd.com"'}}]},
        })
        client.create("resourceslices", {
            "apiVersion": "resource.k8s.io/v1beta1", "kind": "ResourceSlice",
            "metadata": {"name": "node-a-gpu"},
            "spec": {"driver": "gpu.amd.com",
                     "pool": {"name": "node-a", "generation": 1,
                              "resourceSliceCount": 1},
                     "nodeName": "node-a",
                     "devices": [{"name": "gpu-0", "basic": {"attributes": {}}}]},
        })
        client.create("pods", {
            "apiVersion": "v1", "kind": "Pod",
            "metadata": {"name": "legacy", "namespace": "default"},
            "spec": {"containers": [{"name": "c", "resources": {
                "limits": {"amd.com/gpu": 1}}}]},
        })
        sched = SchedulerStub(client)
        assert sched.schedule_pending() >= 1
        pod = client.get("pods", "legacy", "default")
        ercs = pod["status"]["extendedResourceClaimStatus"]
        claim = client.get("resourceclaims", ercs["resourceClaimName"], "default")
        res = claim["status"]["allocation"]["devices"]["results"]
        assert res[0]["device"] == "gpu-0"


class TestHttpWatchCache:
    """resourceVersion resume, 410 Gone and bookmarks over the HTTP wire."""

    def _pod(self, client, name):
        return client.create("pods", {"apiVersion": "v1", "kind": "Pod",
                                      "metadata": {"name": name,
                                                   "namespace": "default"}})

    def test_list_with_rv_over_http(self, client):
        self._pod(client, "a")
        items, rv = client.list_with_rv("pods")
        assert len(items) == 1 and rv and int(rv) >= 1

    def test_watch_resume_over_http(self, client):
        self._pod(client, "a")
        _, rv = client.list_with_rv("pods")
        self._pod(client, "b")
        w = client.watch("pods", resource_version=rv)
        ev = w.next(timeout=5.0)
        assert ev is not None and ev.type == "ADDED"
        assert ev.object["metadata"]["name"] == "b"
        w.stop()

    def test_410_over_http_surfaces_as_error_event(self):
        from k8s_dra_driver_gpu_amd.k8s.fakeserver import FakeApiServer

        srv = MiniApiServer(api=FakeApiServer(history_limit=2))
        srv.start()
        try:
            client = HttpClient(base_url=f"http://127.0.0.1:{srv.port}")
            for i in range(6):
                self._pod(client, f"p{i}")
            w = client.watch("pods", resource_version="1")
            ev = w.next(timeout=5.0)
            assert ev is not None and ev.type == "ERROR"
            assert ev.object.get("code") == 410
            w.stop()
        finally:
            srv.stop()

    def test_informer_over_http_survives_410(self):
        from k8s_dra_driver_gpu_amd.k8s.fakeserver import FakeApiServer

        srv = MiniApiServer(api=FakeApiServer(history_limit=2))
        srv.start()
        try:
            client = HttpClient(base_url=f"http://127.0.0.1:{srv.port}")
            self._pod(client, "victim")
            inf = Informer(client, "pods").start()
            assert inf.wait_for_sync(10.0)
            deadline = time.monotonic() + 5
            while time.monotonic() < deadline and inf._watch is None:
                time.sleep(0.02)
            inf._watch.stop()
            client.delete("pods", "victim", "default")
            for i in range(6):
                self._pod(client, f"churn{i}")

            def converged():
                names = {o["metadata"]["name"] for o in inf.items()}
                return "victim" not in names and "churn5" in names

            deadline = time.monotonic() + 15
            while time.monotonic() < deadline and not converged():
                time.sleep(0.1)
            assert converged(), [o["metadata"]["name"] for o in inf.items()]
            inf.stop()
        finally:
            srv.stop()


class TestHttpBookmarks:
    def test_idle_http_watch_emits_bookmarks(self, client):
        client.create("pods", {"apiVersion": "v1", "kind": "Pod",
                               "metadata": {"name": "a", "namespace": "default"}})
        _, rv = client.list_with_rv("pods")
        w = client.watch("pods", resource_version=rv, allow_bookmarks=True)
        deadline = time.monotonic() + 15
        ev = None
        while time.monotonic() < deadline:
            ev = w.next(timeout=2.0)
            if ev is not None:
                break
        assert ev is not None and ev.type == "BOOKMARK", ev
        assert ev.object["metadata"]["resourceVersion"] == rv
        w.stop()


class TestHttpInformerRandomizedConvergence:
    """The randomized watch-cache convergence test over the REAL HTTP path
    (MiniApiServer + HttpClient + Informer): mutations through the HTTP
    client, tiny watch-cache history (eviction -> 410 Gone -> relist), and
    a 2 s forced stream recycle via AMDDRA_WATCH_TIMEOUT."""

    def test_converges_under_eviction_and_recycling(self, monkeypatch):
        import random

        from k8s_dra_driver_gpu_amd.k8s.fakeserver import FakeApiServer

        monkeypatch.setenv("AMDDRA_WATCH_TIMEOUT", "1")
        srv = MiniApiServer(api=FakeApiServer(history_limit=4))
        srv.start()
        c = HttpClient(base_url=f"http://127.0.0.1:{srv.port}")
        inf = Informer(c, "computedomains").start()
        try:
            assert inf.wait_for_sync(5.0)
            rng = random.Random(99)
            live = set()
            for step in range(60):
                op = rng.random()
                if op < 0.5 or not live:
                    name = f"cd-{rng.randrange(25)}"
                    if name not in live:
                        c.create("computedomains", cd(name))
                        live.add(name)
                elif op < 0.75:
                    name = rng.choice(sorted(live))
                    cur = c.get("computedomains", name, "default")
                    cur["spec"]["numNodes"] = rng.randrange(1, 9)
                    c.update("computedomains", cur)
                else:
                    name = rng.choice(sorted(live))
                    c.delete("computedomains", name, "default")
                    live.discard(name)
                if step == 30:
                    time.sleep(1.2)  # span at least one forced recycle
            deadline = time.monotonic() + 8.0
            while time.monotonic() < deadline:
                want = {o["metadata"]["name"]: o["metadata"]["resourceVersion"]
                        for o in c.list("computedomains")}
                have = {o["metadata"]["name"]: o["metadata"]["resourceVersion"]
                        for o in inf.items()}
                if have == want:
                    break
                time.sleep(0.05)
            assert have == want, f"diverged: {len(have)} vs {len(want)}"
            assert sorted(have) == sorted(live)
        finally:
            inf.stop()
            srv.stop()
