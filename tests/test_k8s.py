"""Tests for the k8s machinery: fake API server semantics, informers, leader
election."""

import threading
import time

import pytest

from k8s_dra_driver_gpu_amd.k8s.client import FakeClient
from k8s_dra_driver_gpu_amd.k8s.fakeserver import (
    AlreadyExists,
    Conflict,
    FakeApiServer,
    NotFound,
)
from k8s_dra_driver_gpu_amd.k8s.informer import Informer, obj_key
from k8s_dra_driver_gpu_amd.k8s.leaderelection import LeaderElector


def cd(name="cd1", ns="default", **spec):
    return {
        "apiVersion": "resource.amd.com/v1beta1",
        "kind": "ComputeDomain",
        "metadata": {"name": name, "namespace": ns},
        "spec": {"numNodes": 1, **spec},
    }


class TestFakeServer:
    def test_create_get_list_delete(self):
        s = FakeApiServer()
        obj = s.create("computedomains", cd())
        assert obj["metadata"]["uid"]
        assert obj["metadata"]["resourceVersion"] == "1"
        got = s.get("computedomains", "cd1", "default")
        assert got["spec"]["numNodes"] == 1
        assert len(s.list("computedomains")) == 1
        assert s.list("computedomains", namespace="other") == []
        s.delete("computedomains", "cd1", "default")
        with pytest.raises(NotFound):
            s.get("computedomains", "cd1", "default")

    def test_duplicate_create(self):
        s = FakeApiServer()
        s.create("computedomains", cd())
        with pytest.raises(AlreadyExists):
            s.create("computedomains", cd())

    def test_optimistic_concurrency(self):
        s = FakeApiServer()
        a = s.create("computedomains", cd())
        b = s.get("computedomains", "cd1", "default")
        a["spec"]["numNodes"] = 2
        s.update("computedomains", a)
        b["spec"]["numNodes"] = 3
        with pytest.raises(Conflict):
            s.update("computedomains", b)

    def test_generation_bumps_on_spec_change(self):
        s = FakeApiServer()
        a = s.create("computedomains", cd())
        assert a["metadata"]["generation"] == 1
        a["spec"]["numNodes"] = 4
        a = s.update("computedomains", a)
        assert a["metadata"]["generation"] == 2
        a.setdefault("status", {})["status"] = "Ready"
        a = s.update("computedomains", a)
        assert a["metadata"]["generation"] == 2  # status-only change

    def test_merge_patch(self):
        s = FakeApiServer()
        s.create("computedomains", cd())
        s.patch(
            "computedomains",
            "cd1",
            {"metadata": {"labels": {"a": "b"}}, "status": {"status": "Ready"}},
            "default",
        )
        got = s.get("computedomains", "cd1", "default")
        assert got["metadata"]["labels"] == {"a": "b"}
        assert got["status"]["status"] == "Ready"
        # null deletes a key
        s.patch("computedomains", "cd1", {"metadata": {"labels": {"a": None}}}, "default")
        assert s.get("computedomains", "cd1", "default")["metadata"]["labels"] == {}

    def test_finalizer_semantics(self):
        s = FakeApiServer()
        obj = cd()
        obj["metadata"]["finalizers"] = ["amd.com/cd-finalizer"]
        s.create("computedomains", obj)
        s.delete("computedomains", "cd1", "default")
        got = s.get("computedomains", "cd1", "default")  # still there
        assert got["metadata"]["deletionTimestamp"]
        got["metadata"]["finalizers"] = []
        s.update("computedomains", got)
        with pytest.raises(NotFound):
            s.get("computedomains", "cd1", "default")

    def test_watch_events(self):
        s = FakeApiServer()
        s.create("computedomains", cd("pre"))
        w = s.watch("computedomains")
        ev = w.next(1.0)
        assert ev.type == "ADDED" and ev.object["metadata"]["name"] == "pre"
        s.create("computedomains", cd("live"))
        ev = w.next(1.0)
        assert ev.type == "ADDED" and ev.object["metadata"]["name"] == "live"
        s.patch("computedomains", "live", {"status": {"status": "Ready"}}, "default")
        ev = w.next(1.0)
        assert ev.type == "MODIFIED"
        s.delete("computedomains", "live", "default")
        ev = w.next(1.0)
        assert ev.type == "DELETED"
        w.stop()

    def test_watch_selector(self):
        s = FakeApiServer()
        w = s.watch("pods", selector={"app": "x"})
        s.create("pods", {"metadata": {"name": "p1", "namespace": "d", "labels": {"app": "x"}}})
        s.create("pods", {"metadata": {"name": "p2", "namespace": "d", "labels": {"app": "y"}}})
        ev = w.next(0.5)
        assert ev.object["metadata"]["name"] == "p1"
        assert w.next(0.2) is None
        w.stop()

    def test_generate_name(self):
        s = FakeApiServer()
        o = s.create("pods", {"metadata": {"generateName": "worker-", "namespace": "d"}})
        assert o["metadata"]["name"].startswith("worker-")


class TestClientHelpers:
    def test_apply_create_then_update(self):
        c = FakeClient()
        c.apply("computedomains", cd())
        obj = c.apply("computedomains", cd(numNodes=2))
        assert obj["spec"]["numNodes"] == 2 or obj["spec"].get("numNodes") == 2
        assert len(c.list("computedomains")) == 1

    def test_finalizer_helpers(self):
        c = FakeClient()
        c.create("computedomains", cd())
        c.add_finalizer("computedomains", "cd1", "default", "f1")
        c.add_finalizer("computedomains", "cd1", "default", "f1")  # idempotent
        got = c.get("computedomains", "cd1", "default")
        assert got["metadata"]["finalizers"] == ["f1"]
        c.delete("computedomains", "cd1", "default")
        c.remove_finalizer("computedomains", "cd1", "default", "f1")
        assert c.get_or_none("computedomains", "cd1", "default") is None


class TestInformer:
    def test_sync_and_events(self):
        c = FakeClient()
        c.create("computedomains", cd("a"))
        inf = Informer(c, "computedomains").start()
        assert inf.wait_for_sync(5.0)
        assert inf.get("default/a") is not None
        events = []
        done = threading.Event()

        def handler(t, obj):
            events.append((t, obj["metadata"]["name"]))
            if t == "DELETED":
                done.set()

        inf.add_handler(handler)
        c.create("computedomains", cd("b"))
        c.delete("computedomains", "b", "default")
        assert done.wait(5.0)
        names = [n for _, n in events]
        assert "a" in names and "b" in names
        inf.stop()

    def test_uid_index(self):
        c = FakeClient()
        obj = c.create("computedomains", cd("a"))
        inf = Informer(c, "computedomains").start()
        inf.wait_for_sync(5.0)
        assert inf.get_by_uid(obj["metadata"]["uid"])["metadata"]["name"] == "a"
        inf.stop()


class TestLeaderElection:
    def test_single_leader_and_failover(self):
        c = FakeClient()
        e1 = LeaderElector(c, "lock", "kube-system", "pod-1",
                           lease_duration=0.6, retry_period=0.05)
        e2 = LeaderElector(c, "lock", "kube-system", "pod-2",
                           lease_duration=0.6, retry_period=0.05)
        e1.run()
        assert e1.is_leader.wait(5.0)
        e2.run()
        time.sleep(0.3)
        assert not e2.is_leader.is_set()
        # release-on-cancel -> fast failover
        e1.stop()
        assert e2.is_leader.wait(6.0)
        e2.stop()

    def test_expired_lease_takeover(self):
        c = FakeClient()
        e1 = LeaderElector(c, "lock", "ns", "p1", lease_duration=0.2, retry_period=0.05)
        e1.run()
        assert e1.is_leader.wait(5.0)
        # kill without release (crash)
        e1._stop.set()
        e1._thread.join()
        e2 = LeaderElector(c, "lock", "ns", "p2", lease_duration=0.2, retry_period=0.05)
        e2.run()
        assert e2.is_leader.wait(6.0)
        e2.stop()


class TestRetryOnConflict:
    """client-go retry.RetryOnConflict analog (kubeclient consumers retry
    optimistic-concurrency races instead of failing the reconcile)."""

    def test_succeeds_after_transient_conflicts(self):
        from k8s_dra_driver_gpu_amd.k8s.client import retry_on_conflict
        from k8s_dra_driver_gpu_amd.k8s.fakeserver import Conflict

        calls = []

        def fn():
            calls.append(1)
            if len(calls) < 3:
                raise Conflict("rv mismatch")
            return "ok"

        assert retry_on_conflict(fn, base_delay=0.001) == "ok"
        assert len(calls) == 3

    def test_raises_after_exhausting_attempts(self):
        from k8s_dra_driver_gpu_amd.k8s.client import retry_on_conflict
        from k8s_dra_driver_gpu_amd.k8s.fakeserver import Conflict

        with pytest.raises(Conflict):
            retry_on_conflict(lambda: (_ for _ in ()).throw(Conflict("x")).__next__(),
                              attempts=2, base_delay=0.001)

    def test_apply_wins_against_racing_writer(self):
        # A writer that bumps resourceVersion right after apply's read: the
        # first PUT loses with 409, the retry re-reads and wins.
        from k8s_dra_driver_gpu_amd.k8s.fakeserver import Conflict

        class RacyClient(FakeClient):
            def __init__(self):
                super().__init__()
                self.races_left = 2

            def get_or_none(self, resource, name, namespace=""):
                obj = super().get_or_none(resource, name, namespace)
                if obj is not None and self.races_left > 0:
                    self.races_left -= 1
                    racer = dict(obj)
                    racer["metadata"] = dict(obj["metadata"])
                    racer.setdefault("spec", {})
                    self.server.update(resource, racer)  # bumps rv
                return obj

        c = RacyClient()
        c.create("computedomains", {
            "apiVersion": "resource.amd.com/v1beta1", "kind": "ComputeDomain",
            "metadata": {"name": "cd", "namespace": "default"},
            "spec": {"numNodes": 1},
        })
        out = c.apply("computedomains", {
            "apiVersion": "resource.amd.com/v1beta1", "kind": "ComputeDomain",
            "metadata": {"name": "cd", "namespace": "default"},
            "spec": {"numNodes": 8},
        })
        assert out["spec"]["numNodes"] == 8
        assert c.races_left == 0

    def test_apply_without_retry_would_conflict(self):
        # Sanity: a direct stale update (no retry helper) still surfaces 409.
        from k8s_dra_driver_gpu_amd.k8s.fakeserver import Conflict

        c = FakeClient()
        c.create("computedomains", {
            "apiVersion": "resource.amd.com/v1beta1", "kind": "ComputeDomain",
            "metadata": {"name": "cd", "namespace": "default"},
            "spec": {"numNodes": 1},
        })
        stale = c.get("computedomains", "cd", "default")
        fresh = c.get("computedomains", "cd", "default")
        fresh["spec"]["numNodes"] = 2
        c.update("computedomains", fresh)
        stale["spec"]["numNodes"] = 3
        with pytest.raises(Conflict):
            c.update("computedomains", stale)


class TestExtendedResourceSkips:
    def _setup(self):
        c = FakeClient()
        c.create("deviceclasses", {
            "apiVersion": "resource.k8s.io/v1beta1", "kind": "DeviceClass",
            "metadata": {"name": "gpu.amd.com"},
            "spec": {"extendedResourceName": "amd.com/gpu", "selectors": [
                {"cel": {"expression": 'device.driver == "gpu.amd.com"'}}]},
        })
        c.create("resourceslices", {
            "apiVersion": "resource.k8s.io/v1beta1", "kind": "ResourceSlice",
            "metadata": {"name": "n-gpu"},
            "spec": {"driver": "gpu.amd.com",
                     "pool": {"name": "n", "generation": 1,
                              "resourceSliceCount": 1},
                     "nodeName": "n",
                     "devices": [{"name": "gpu-0", "basic": {"attributes": {}}}]},
        })
        return c

    def test_terminal_pod_gets_no_claim(self):
        from k8s_dra_driver_gpu_amd.k8s.scheduler import SchedulerStub

        c = self._setup()
        c.create("pods", {
            "apiVersion": "v1", "kind": "Pod",
            "metadata": {"name": "done", "namespace": "default"},
            "spec": {"containers": [{"name": "c", "resources": {
                "limits": {"amd.com/gpu": 1}}}]},
            "status": {"phase": "Succeeded"},
        })
        sched = SchedulerStub(c)
        assert sched.schedule_extended_resources() == 0
        assert c.list("resourceclaims") == []

    def test_running_pod_gets_claim(self):
        from k8s_dra_driver_gpu_amd.k8s.scheduler import SchedulerStub

        c = self._setup()
        c.create("pods", {
            "apiVersion": "v1", "kind": "Pod",
            "metadata": {"name": "run", "namespace": "default"},
            "spec": {"containers": [{"name": "c", "resources": {
                "limits": {"amd.com/gpu": 1}}}]},
            "status": {"phase": "Pending"},
        })
        sched = SchedulerStub(c)
        assert sched.schedule_extended_resources() == 1
        assert len(c.list("resourceclaims")) == 1


class TestInformerRecovery:
    def test_recovers_after_watch_break(self):
        """A broken watch stream must trigger relist + rewatch (client-go
        reflector behavior) — events created after the break still arrive."""
        from k8s_dra_driver_gpu_amd.k8s.informer import Informer

        c = FakeClient()
        inf = Informer(c, "computedomains")
        seen = []
        inf.add_handler(lambda t, o: seen.append((t, o["metadata"]["name"])))
        inf.start()
        assert inf.wait_for_sync()

        def cd(name):
            return {"apiVersion": "resource.amd.com/v1beta1",
                    "kind": "ComputeDomain",
                    "metadata": {"name": name, "namespace": "default"},
                    "spec": {"numNodes": 1}}

        c.create("computedomains", cd("w1"))
        deadline = time.time() + 10
        while time.time() < deadline and ("ADDED", "w1") not in seen:
            time.sleep(0.02)
        assert ("ADDED", "w1") in seen
        inf._watch.stop()  # simulate the stream dying server-side
        time.sleep(0.3)
        c.create("computedomains", cd("w2"))
        deadline = time.time() + 10
        while time.time() < deadline and not any(n == "w2" for _, n in seen):
            time.sleep(0.02)
        assert any(n == "w2" for _, n in seen), seen
        assert inf.get("default/w2") is not None
        inf.stop()


class TestLeaderElectionOutage:
    def test_apiserver_outage_steps_down_after_deadline(self):
        """An unreachable apiserver must not kill the elector thread; the
        leader keeps leading through transient failures and steps down only
        after renew_deadline (client-go RenewDeadline semantics)."""
        from k8s_dra_driver_gpu_amd.k8s.leaderelection import LeaderElector

        class OutageClient(FakeClient):
            def __init__(self):
                super().__init__()
                self.down = False

            def _check(self):
                if self.down:
                    raise RuntimeError("apiserver unreachable")

            def get_or_none(self, resource, name, namespace=""):
                if resource == "leases":
                    self._check()
                return super().get_or_none(resource, name, namespace)

            def create(self, resource, obj):
                if resource == "leases":
                    self._check()
                return super().create(resource, obj)

            def update(self, resource, obj):
                if resource == "leases":
                    self._check()
                return super().update(resource, obj)

        c = OutageClient()
        e = LeaderElector(c, "lock", "ns", "p1", lease_duration=2.0,
                          renew_deadline=0.6, retry_period=0.1)
        e.run()
        assert e.is_leader.wait(5.0)
        c.down = True
        # within the renew window the leader holds on
        time.sleep(0.25)
        assert e.is_leader.is_set()
        # past renew_deadline it steps down, and the thread is still alive
        deadline = time.time() + 5
        while time.time() < deadline and e.is_leader.is_set():
            time.sleep(0.05)
        assert not e.is_leader.is_set()
        assert e._thread.is_alive()
        # apiserver returns: leadership is re-acquired by the SAME elector
        c.down = False
        assert e.is_leader.wait(5.0)
        e.stop()


class TestSchedulerMissedDelete:
    def test_resync_releases_stale_allocation(self):
        """A DELETED event lost in the informer relist->rewatch gap leaves
        the stub's bookkeeping pinning the device; resync() (run
        periodically by cmd/scheduler) must free it from live state."""
        from k8s_dra_driver_gpu_amd.k8s.scheduler import SchedulerStub

        c = FakeClient()
        c.create("deviceclasses", {
            "apiVersion": "resource.k8s.io/v1beta1", "kind": "DeviceClass",
            "metadata": {"name": "gpu.amd.com"},
            "spec": {"selectors": [
                {"cel": {"expression": 'device.driver == "gpu.amd.com"'}}]},
        })
        c.create("resourceslices", {
            "apiVersion": "resource.k8s.io/v1beta1", "kind": "ResourceSlice",
            "metadata": {"name": "n-gpu"},
            "spec": {"driver": "gpu.amd.com",
                     "pool": {"name": "n", "generation": 1,
                              "resourceSliceCount": 1},
                     "nodeName": "n",
                     "devices": [{"name": "gpu-0", "basic": {"attributes": {}}}]},
        })

        def claim(name):
            return c.create("resourceclaims", {
                "apiVersion": "resource.k8s.io/v1beta1", "kind": "ResourceClaim",
                "metadata": {"name": name, "namespace": "default"},
                "spec": {"devices": {"requests": [
                    {"name": "r0", "deviceClassName": "gpu.amd.com"}]}},
            })

        sched = SchedulerStub(c)
        a = claim("a")
        assert sched.schedule_pending() == 1  # steady state (stub synced)
        # the DELETED event is LOST: delete without sched.release(...)
        c.delete("resourceclaims", "a", "default")
        claim("b")
        assert sched.schedule_pending() == 0  # stale entry pins gpu-0
        sched.resync()  # what cmd/scheduler now does every few seconds
        assert sched.schedule_pending() == 1
        got = c.get("resourceclaims", "b", "default")
        assert got["status"]["allocation"]["devices"]["results"][0]["device"] == "gpu-0"


class TestWatchCacheSemantics:
    """client-go watch-cache model (VERDICT round-1 item 9):
    resourceVersion resume, 410 Gone, BOOKMARK events."""

    def _mk(self, s, name, ns="default"):
        return s.create("pods", {"apiVersion": "v1", "kind": "Pod",
                                 "metadata": {"name": name, "namespace": ns}})

    def test_list_with_rv_and_resume(self):
        from k8s_dra_driver_gpu_amd.k8s.fakeserver import FakeApiServer

        s = FakeApiServer()
        self._mk(s, "a")
        items, rv = s.list_with_rv("pods")
        assert len(items) == 1 and rv == items[0]["metadata"]["resourceVersion"]
        self._mk(s, "b")
        s.delete("pods", "a", "default")
        w = s.watch("pods", resource_version=rv)
        evs = [w.next(timeout=0.5) for _ in range(2)]
        assert [(e.type, e.object["metadata"]["name"]) for e in evs] == [
            ("ADDED", "b"), ("DELETED", "a")]
        assert w.next(timeout=0.1) is None  # nothing replayed from before rv
        w.stop()

    def test_too_old_rv_gets_410_gone(self):
        from k8s_dra_driver_gpu_amd.k8s.fakeserver import FakeApiServer, Gone

        s = FakeApiServer(history_limit=2)
        for i in range(6):
            self._mk(s, f"p{i}")
        with pytest.raises(Gone):
            s.watch("pods", resource_version="1")
        # a recent rv still resumes fine
        _, rv = s.list_with_rv("pods")
        w = s.watch("pods", resource_version=rv)
        assert w.next(timeout=0.1) is None
        w.stop()

    def test_bookmark_on_idle_stream(self):
        from k8s_dra_driver_gpu_amd.k8s.fakeserver import FakeApiServer

        s = FakeApiServer()
        self._mk(s, "a")
        _, rv = s.list_with_rv("pods")
        w = s.watch("pods", resource_version=rv, bookmark_interval=0.1)
        time.sleep(0.15)
        ev = w.next(timeout=0.3)
        assert ev is not None and ev.type == "BOOKMARK"
        assert ev.object["metadata"]["resourceVersion"] == rv
        w.stop()

    def test_informer_survives_410_without_missing_deletes(self):
        """Forced 410: the informer is disconnected while the event window
        is evicted (including a DELETED it never saw); the relist driven by
        410 reconciles the cache — no controller-level resync crutch."""
        from k8s_dra_driver_gpu_amd.k8s.client import FakeClient
        from k8s_dra_driver_gpu_amd.k8s.fakeserver import FakeApiServer
        from k8s_dra_driver_gpu_amd.k8s.informer import Informer

        s = FakeApiServer(history_limit=2)
        c = FakeClient(s)
        self._mk(s, "victim")
        events = []
        inf = Informer(c, "pods").start()
        assert inf.wait_for_sync(5.0)
        inf.add_handler(lambda t, o: events.append((t, o["metadata"]["name"])))
        # simulate a network partition: kill the informer's live stream
        deadline = time.monotonic() + 5
        while time.monotonic() < deadline and inf._watch is None:
            time.sleep(0.02)
        inf._watch.stop()
        # while disconnected: the victim is deleted and enough churn evicts
        # the event from the 2-entry history window
        s.delete("pods", "victim", "default")
        for i in range(6):
            self._mk(s, f"churn{i}")

        def converged():
            names = {o["metadata"]["name"] for o in inf.items()}
            return "victim" not in names and "churn5" in names

        deadline = time.monotonic() + 10
        while time.monotonic() < deadline and not converged():
            time.sleep(0.05)
        assert converged(), [o["metadata"]["name"] for o in inf.items()]
        assert ("DELETED", "victim") in events
        inf.stop()

    def test_informer_rewatches_without_relist_when_rv_retained(self):
        """A plain stream break with the rv still in the window must NOT
        trigger a relist — the re-watch resumes from last_resource_version."""
        from k8s_dra_driver_gpu_amd.k8s.client import FakeClient
        from k8s_dra_driver_gpu_amd.k8s.fakeserver import FakeApiServer
        from k8s_dra_driver_gpu_amd.k8s.informer import Informer

        s = FakeApiServer()
        c = FakeClient(s)
        lists = []
        real = c.list_with_rv
        c.list_with_rv = lambda *a, **k: (lists.append(1), real(*a, **k))[1]
        self._mk(s, "a")
        inf = Informer(c, "pods").start()
        assert inf.wait_for_sync(5.0)
        assert len(lists) == 1
        deadline = time.monotonic() + 5
        while time.monotonic() < deadline and inf._watch is None:
            time.sleep(0.02)
        inf._watch.stop()  # stream break, rv still retained
        self._mk(s, "b")
        deadline = time.monotonic() + 10
        while time.monotonic() < deadline and inf.get("default/b") is None:
            time.sleep(0.05)
        assert inf.get("default/b") is not None
        assert len(lists) == 1, "re-watch must resume from rv, not relist"
        inf.stop()

    def test_bookmark_advances_informer_rv(self):
        """Idle informer's last-seen rv advances via bookmarks, so a later
        re-watch does not resume from a stale (evictable) rv."""
        from k8s_dra_driver_gpu_amd.k8s.client import FakeClient
        from k8s_dra_driver_gpu_amd.k8s.fakeserver import FakeApiServer
        from k8s_dra_driver_gpu_amd.k8s.informer import Informer

        s = FakeApiServer()
        c = FakeClient(s)
        c.bookmark_interval = 0.1
        self._mk(s, "a")
        inf = Informer(c, "pods").start()
        assert inf.wait_for_sync(5.0)
        rv0 = int(inf.last_resource_version)
        # churn on a DIFFERENT resource bumps the global rv; the pods
        # watch sees no object events, only bookmarks
        for i in range(5):
            s.create("nodes", {"apiVersion": "v1", "kind": "Node",
                               "metadata": {"name": f"n{i}"}})
        deadline = time.monotonic() + 5
        while time.monotonic() < deadline and int(inf.last_resource_version) <= rv0:
            time.sleep(0.05)
        assert int(inf.last_resource_version) > rv0
        inf.stop()


class TestInformerRandomizedConvergence:
    """Randomized model test: drive the fake API server through hundreds of
    create/update/delete ops with an aggressively small watch-cache history
    (so the informer repeatedly hits 410 Gone and must relist), verifying
    at interleaved checkpoints that the informer cache converges exactly to
    the server's state. Seeded for reproducibility."""

    def _converged(self, inf, server, timeout=5.0):
        import time as _t

        deadline = _t.monotonic() + timeout
        want = None
        while _t.monotonic() < deadline:
            objs = server.list("computedomains")
            want = {
                f"{o['metadata'].get('namespace','')}/{o['metadata']['name']}":
                    o["metadata"]["resourceVersion"]
                for o in objs
            }
            have = {
                f"{o['metadata'].get('namespace','')}/{o['metadata']['name']}":
                    o["metadata"]["resourceVersion"]
                for o in inf.items()
            }
            if have == want:
                return True
            _t.sleep(0.02)
        return False

    def test_converges_under_eviction_pressure(self):
        import random

        from k8s_dra_driver_gpu_amd.k8s.fakeserver import FakeApiServer
        from k8s_dra_driver_gpu_amd.k8s.informer import Informer

        rng = random.Random(1234)
        server = FakeApiServer(history_limit=4)
        inf = Informer(server, "computedomains").start()
        try:
            assert inf.wait_for_sync(5.0)
            live = {}
            for step in range(300):
                op = rng.random()
                if op < 0.45 or not live:
                    name = f"cd-{rng.randrange(40)}"
                    if name not in live:
                        obj = {
                            "apiVersion": "resource.amd.com/v1beta1",
                            "kind": "ComputeDomain",
                            "metadata": {"name": name, "namespace": "default"},
                            "spec": {"numNodes": 1},
                        }
                        live[name] = server.create("computedomains", obj)
                elif op < 0.75:
                    name = rng.choice(list(live))
                    cur = server.get("computedomains", name, "default")
                    cur["spec"]["numNodes"] = rng.randrange(1, 9)
                    live[name] = server.update("computedomains", cur)
                else:
                    name = rng.choice(list(live))
                    server.delete("computedomains", name, "default")
                    del live[name]
                if step in (75, 150, 225):
                    assert self._converged(inf, server), (
                        f"informer diverged at step {step}"
                    )
            assert self._converged(inf, server), "informer diverged at end"
            assert len(inf.items()) == len(live)
        finally:
            inf.stop()
