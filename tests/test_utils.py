"""Unit tests for shared infrastructure (flock, workqueue, featuregates, bootid).

Mirrors the reference's pkg/-level unit tests (pkg/workqueue/workqueue_test.go,
pkg/featuregates/featuregates_test.go:33-547, pkg/bootid/bootid_test.go).
"""

import os
import threading
import time

import pytest

from k8s_dra_driver_gpu_amd.utils.bootid import read_boot_id
from k8s_dra_driver_gpu_amd.utils.featuregates import (
    FeatureGates,
    Gate,
    Stage,
    VersionedSpec,
    new_default_feature_gates,
)
from k8s_dra_driver_gpu_amd.utils.flock import Flock, FlockTimeout
from k8s_dra_driver_gpu_amd.utils.workqueue import (
    RateLimiter,
    WorkQueue,
    cd_daemon_limiter,
    prepare_unprepare_limiter,
)


class TestFlock:
    def test_acquire_release(self, tmp_path):
        lock = Flock(str(tmp_path / "a.lock"))
        with lock.acquire(timeout=1.0):
            assert os.path.exists(tmp_path / "a.lock")
        # reacquirable after release
        with lock.acquire(timeout=1.0):
            pass

    def test_contention_timeout(self, tmp_path):
        path = str(tmp_path / "b.lock")
        l1 = Flock(path)
        l2 = Flock(path, poll_interval=0.005)
        l1.acquire(timeout=1.0)
        try:
            t0 = time.monotonic()
            with pytest.raises(FlockTimeout):
                l2.acquire(timeout=0.1)
            assert time.monotonic() - t0 >= 0.1
        finally:
            l1.release()
        # now acquirable
        with l2.acquire(timeout=1.0):
            pass

    def test_cross_thread_exclusion(self, tmp_path):
        path = str(tmp_path / "c.lock")
        order = []

        def worker(name):
            lk = Flock(path, poll_interval=0.001)
            with lk.acquire(timeout=5.0):
                order.append((name, "in"))
                time.sleep(0.02)
                order.append((name, "out"))

        ts = [threading.Thread(target=worker, args=(i,)) for i in range(3)]
        for t in ts:
            t.start()
        for t in ts:
            t.join()
        # critical sections must not interleave
        for i in range(0, len(order), 2):
            assert order[i][0] == order[i + 1][0]
            assert order[i][1] == "in" and order[i + 1][1] == "out"


class TestRateLimiter:
    def test_exponential_backoff(self):
        rl = RateLimiter(base_delay=0.1, max_delay=1.0)
        assert rl.when("k") == pytest.approx(0.1)
        assert rl.when("k") == pytest.approx(0.2)
        assert rl.when("k") == pytest.approx(0.4)
        for _ in range(10):
            rl.when("k")
        assert rl.when("k") == pytest.approx(1.0)  # capped
        rl.forget("k")
        assert rl.when("k") == pytest.approx(0.1)

    def test_jitter_bounds(self):
        rl = cd_daemon_limiter()
        d = rl.when("x")
        assert 0.005 * 0.75 <= d <= 0.005 * 1.25

    def test_global_bucket(self):
        # burst of 10 distinct keys passes at per-item delay; the 11th+ waits
        # on the bucket (5 rps) and successive overflows space out at 1/qps
        rl = RateLimiter(base_delay=0.001, max_delay=0.001, qps=5.0, burst=10)
        delays = [rl.when(f"k{i}") for i in range(13)]
        assert all(d == pytest.approx(0.001) for d in delays[:10])
        assert delays[10] == pytest.approx(0.2, abs=0.05)
        assert delays[11] == pytest.approx(0.4, abs=0.05)
        assert delays[12] == pytest.approx(0.6, abs=0.05)

    def test_preset_prepare_limiter(self):
        rl = prepare_unprepare_limiter()
        assert rl.when("a") == pytest.approx(0.25)
        assert rl.when("a") == pytest.approx(0.5)
        for _ in range(5):
            rl.when("a")
        assert rl.when("a") == pytest.approx(3.0)  # capped at 3 s

    def test_per_key_independent(self):
        rl = RateLimiter(base_delay=0.1, max_delay=5.0)
        rl.when("a")
        rl.when("a")
        assert rl.when("b") == pytest.approx(0.1)


class TestWorkQueue:
    def test_runs_work(self):
        wq = WorkQueue(name="t1")
        done = threading.Event()
        wq.enqueue("k", done.set)
        assert done.wait(2.0)
        wq.shutdown()

    def test_latest_enqueue_wins(self):
        wq = WorkQueue(name="t2")
        results = []
        gate = threading.Event()
        # first item blocks the single worker so later enqueues stay queued
        wq.enqueue("blocker", lambda: gate.wait(2.0))
        wq.enqueue("k", lambda: results.append("old"))
        wq.enqueue("k", lambda: results.append("new"))
        gate.set()
        assert wq.wait_idle(5.0)
        wq.shutdown()
        assert results == ["new"]

    def test_retry_on_failure(self):
        wq = WorkQueue(limiter=RateLimiter(base_delay=0.01, max_delay=0.02), name="t3")
        attempts = []

        def flaky():
            attempts.append(1)
            if len(attempts) < 3:
                raise RuntimeError("boom")

        wq.enqueue("k", flaky)
        deadline = time.monotonic() + 5.0
        while len(attempts) < 3 and time.monotonic() < deadline:
            time.sleep(0.01)
        wq.shutdown()
        assert len(attempts) == 3

    def test_supersede_cancels_retry(self):
        wq = WorkQueue(limiter=RateLimiter(base_delay=0.05, max_delay=0.05), name="t4")
        log = []

        def failing():
            log.append("fail")
            raise RuntimeError("x")

        wq.enqueue("k", failing)
        time.sleep(0.01)  # let it fail once
        wq.enqueue("k", lambda: log.append("ok"))
        assert wq.wait_idle(5.0)
        wq.shutdown()
        assert log[-1] == "ok"
        assert log.count("ok") == 1

    def test_per_key_serialization_with_many_workers(self):
        """A re-enqueued key must never run concurrently with its in-flight
        predecessor (client-go dirty/processing semantics); other keys keep
        running in parallel."""
        wq = WorkQueue(name="t5", workers=4)
        lock = threading.Lock()
        running = {"k": 0}
        max_concurrent = [0]
        first_started = threading.Event()
        release_first = threading.Event()
        other_done = threading.Event()

        def work_k(block):
            with lock:
                running["k"] += 1
                max_concurrent[0] = max(max_concurrent[0], running["k"])
            first_started.set()
            if block:
                release_first.wait(2.0)
            with lock:
                running["k"] -= 1

        wq.enqueue("k", lambda: work_k(True))
        assert first_started.wait(2.0)
        # queued while the predecessor is still executing
        wq.enqueue("k", lambda: work_k(False))
        # an unrelated key is NOT blocked by k's serialization
        wq.enqueue("other", other_done.set)
        assert other_done.wait(2.0)
        release_first.set()
        assert wq.wait_idle(5.0)
        wq.shutdown()
        assert max_concurrent[0] == 1

    def test_deferred_key_runs_after_predecessor(self):
        """The deferred enqueue is not lost: it runs once the in-flight run
        finishes."""
        wq = WorkQueue(name="t6", workers=2)
        order = []
        gate = threading.Event()
        started = threading.Event()

        def first():
            started.set()
            gate.wait(2.0)
            order.append("first")

        wq.enqueue("k", first)
        assert started.wait(2.0)
        wq.enqueue("k", lambda: order.append("second"))
        gate.set()
        assert wq.wait_idle(5.0)
        wq.shutdown()
        assert order == ["first", "second"]


class TestFeatureGates:
    def test_defaults(self):
        fg = new_default_feature_gates()
        assert fg.enabled("TimeSlicingSettings") is True
        assert fg.enabled("DynamicPartitioning") is False
        fg.validate()

    def test_set_and_parse(self):
        fg = new_default_feature_gates()
        fg.set_from_string("DynamicPartitioning=true, DeviceHealthCheck=false")
        assert fg.enabled("DynamicPartitioning") is True
        assert fg.enabled("DeviceHealthCheck") is False

    def test_unknown_gate(self):
        fg = new_default_feature_gates()
        with pytest.raises(KeyError):
            fg.set("NoSuchGate", True)
        with pytest.raises(ValueError):
            fg.set_from_string("TimeSlicingSettings=maybe")

    def test_dependency_validation(self):
        fg = new_default_feature_gates()
        fg.set("FabricDaemonsWithDNSNames", True)
        fg.set("ComputeDomainCliques", False)
        with pytest.raises(ValueError, match="requires"):
            fg.validate()
        fg.set("ComputeDomainCliques", True)
        fg.validate()

    def test_conflict_validation(self):
        fg = FeatureGates()
        fg.register(Gate("A", [VersionedSpec("1.0", Stage.BETA, True)], conflicts=("B",)))
        fg.register(Gate("B", [VersionedSpec("1.0", Stage.BETA, True)]))
        with pytest.raises(ValueError, match="mutually exclusive"):
            fg.validate()
        fg.set("B", False)
        fg.validate()

    def test_versioned_lifecycle(self):
        gate = Gate(
            "G",
            [
                VersionedSpec("1.0", Stage.ALPHA, False),
                VersionedSpec("2.0", Stage.BETA, True),
                VersionedSpec("3.0", Stage.GA, True, locked=True),
            ],
        )
        fg1 = FeatureGates("1.5")
        fg1.register(gate)
        assert fg1.enabled("G") is False
        fg2 = FeatureGates("2.0")
        fg2.register(gate)
        assert fg2.enabled("G") is True
        fg3 = FeatureGates("3.1")
        fg3.register(gate)
        with pytest.raises(ValueError, match="locked"):
            fg3.set("G", False)

    def test_to_map_round_trip(self):
        fg = new_default_feature_gates()
        fg2 = new_default_feature_gates()
        fg.set("DynamicPartitioning", True)
        fg2.set_from_string(fg.to_string())
        assert fg2.to_map() == fg.to_map()


class TestBootID:
    def test_read_real_or_empty(self):
        # on linux CI this file exists; either way no exception
        bid = read_boot_id()
        assert isinstance(bid, str)

    def test_read_custom_path(self, tmp_path):
        p = tmp_path / "boot_id"
        p.write_text("abc-123\n")
        assert read_boot_id(str(p)) == "abc-123"

    def test_missing_file(self, tmp_path):
        assert read_boot_id(str(tmp_path / "nope")) == ""


class TestDebugHelpers:
    def test_stack_dump_http(self):
        import urllib.request

        from k8s_dra_driver_gpu_amd.utils.debug import start_debug_http

        port = start_debug_http()
        with urllib.request.urlopen(f"http://127.0.0.1:{port}/debug/stacks", timeout=5) as r:
            body = r.read().decode()
        assert "MainThread" in body

    def test_sigusr2_dump(self, tmp_path):
        import signal

        from k8s_dra_driver_gpu_amd.utils.debug import install_stack_dump_handler

        path = str(tmp_path / "stacks.dump")
        install_stack_dump_handler(path)
        os.kill(os.getpid(), signal.SIGUSR2)
        time.sleep(0.2)
        assert os.path.exists(path)
        assert "MainThread" in open(path).read()


class TestLogConfig:
    def test_json_formatter(self, capsys):
        import json as _json
        import logging as _logging

        from k8s_dra_driver_gpu_amd.utils.logconfig import JsonFormatter

        rec = _logging.LogRecord("t", _logging.INFO, __file__, 1, "hello %s", ("x",), None)
        out = _json.loads(JsonFormatter().format(rec))
        assert out["msg"] == "hello x"
        assert out["level"] == "info"

    def test_setup_logging_levels(self):
        import logging as _logging

        from k8s_dra_driver_gpu_amd.utils.logconfig import setup_logging

        setup_logging(verbosity=6)
        assert _logging.getLogger().level == _logging.DEBUG
        setup_logging(verbosity=4)
        assert _logging.getLogger().level == _logging.INFO


class TestUnixSocketPathGuard:
    def test_long_path_rejected_loudly(self):
        import pytest as _pt

        from k8s_dra_driver_gpu_amd.utils.paths import check_unix_socket_path

        ok = "/var/lib/kubelet/plugins/gpu.amd.com/dra.sock"
        assert check_unix_socket_path(ok) == ok
        with _pt.raises(ValueError, match="AF_UNIX"):
            check_unix_socket_path("/tmp/" + "x" * 110)


class TestHelmliteRange:
    def test_range_splitlist(self):
        from k8s_dra_driver_gpu_amd.utils.helmlite import render_template

        out = render_template(
            '{{- range $n := splitList "," "a,b" }}\nx: {{ $n | quote }}\n{{- end }}', {})
        assert out == '\nx: "a"\nx: "b"'

    def test_range_variable_scoping_restores(self):
        from k8s_dra_driver_gpu_amd.utils.helmlite import (
            HelmliteError, render_template,
        )
        import pytest as _pt

        # $n is undefined outside the range body
        with _pt.raises(HelmliteError, match="undefined variable"):
            render_template(
                '{{- range $n := splitList "," "a" }}{{ $n }}{{- end }}{{ $n }}', {})

    def test_nested_if_inside_range(self):
        from k8s_dra_driver_gpu_amd.utils.helmlite import render_template

        out = render_template(
            '{{- range $n := splitList "," "a,b" }}'
            '{{- if eq $n "a" }}[{{ $n }}]{{- else }}({{ $n }}){{- end }}'
            '{{- end }}', {})
        assert out == "[a](b)"
