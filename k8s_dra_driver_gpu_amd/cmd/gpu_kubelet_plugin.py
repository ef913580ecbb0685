"""GPU kubelet plugin entrypoint (the ``cmd/gpu-kubelet-plugin/main.go``
analog): flag/env wiring, startup config dump, SIGUSR2 stack dumps, device
enumeration, startup reconciliation, slice publication, health monitoring,
cleanup loops, serving until signalled."""

from __future__ import annotations

import argparse
import logging
import os
import signal
import threading

from .. import GPU_DRIVER_NAME
from ..cdi.spec import CdiHandler
from ..device.devicelib import DeviceLib
from ..k8s.client import FakeClient, HttpClient
from ..metrics.dra import DraMetrics
from ..plugin.checkpoint import CheckpointManager
from ..plugin.cleanup import CheckpointCleanupManager
from ..plugin.device_health import HealthMonitor, TaintTracker
from ..plugin.device_state import DeviceState
from ..plugin.driver import GpuDriver, k8s_claim_resolver
from ..plugin.health_svc import HealthServer
from ..plugin.resourceslice import ResourceSliceGenerator
from ..utils.debug import dump_config, install_stack_dump_handler
from ..utils.featuregates import new_default_feature_gates

logger = logging.getLogger("amddra.cmd.gpu")


def parse_args(argv=None):
    p = argparse.ArgumentParser("gpu-kubelet-plugin")
    env = os.environ.get
    p.add_argument("--node-name", default=env("NODE_NAME", "node"))
    p.add_argument("--plugin-dir",
                   default=env("PLUGIN_DIR", f"/var/lib/kubelet/plugins/{GPU_DRIVER_NAME}"))
    p.add_argument("--registry-dir",
                   default=env("PLUGINS_REGISTRY_DIR", "/var/lib/kubelet/plugins_registry"))
    p.add_argument("--cdi-root", default=env("CDI_ROOT", "/var/run/cdi"))
    p.add_argument("--rocm-driver-root", default=env("ROCM_DRIVER_ROOT", "/opt/rocm"),
                   help="host ROCm install the CDI specs mount into containers "
                        "(the nvidia-driver-root analog)")
    p.add_argument("--feature-gates", default=env("FEATURE_GATES", ""))
    p.add_argument("--healthcheck-port", type=int, default=int(env("HEALTHCHECK_PORT", "0")))
    p.add_argument("--metrics-port", type=int, default=int(env("METRICS_PORT", "0")))
    p.add_argument("--additional-events-to-ignore",
                   default=env("ADDITIONAL_EVENTS_TO_IGNORE", ""))
    p.add_argument("--partitionable-slices", choices=["auto", "true", "false"],
                   default=env("PARTITIONABLE_SLICES", "auto"))
    p.add_argument("--in-cluster", action="store_true",
                   default=env("KUBERNETES_SERVICE_HOST", "") != "")
    p.add_argument("-v", "--verbosity", type=int, default=int(env("LOG_VERBOSITY", "4")))
    p.add_argument("--log-json", action="store_true",
                   default=env("LOG_FORMAT", "") == "json")
    return p.parse_args(argv)


def main(argv=None) -> int:
    args = parse_args(argv)
    from ..utils.logconfig import setup_logging

    setup_logging(args.verbosity, args.log_json)
    install_stack_dump_handler()
    stop = threading.Event()
    for sig in (signal.SIGTERM, signal.SIGINT):
        signal.signal(sig, lambda *_: stop.set())
    dump_config("gpu-kubelet-plugin", vars(args))

    gates = new_default_feature_gates()
    if args.feature_gates:
        gates.set_from_string(args.feature_gates)
    gates.validate()

    client = (HttpClient() if (args.in_cluster or os.environ.get("AMDDRA_API_SERVER"))
              else FakeClient())
    devicelib = DeviceLib()
    gpus = devicelib.gpus()
    logger.info("enumerated %d GPU(s): %s", len(gpus),
                ", ".join(f"{g.canonical_name}({g.product_name})" for g in gpus))

    state_dir = args.plugin_dir
    metrics = DraMetrics()
    state = DeviceState(
        devicelib=devicelib,
        cdi=CdiHandler(cdi_root=args.cdi_root, rocm_root=args.rocm_driver_root),
        checkpoints=CheckpointManager(state_dir),
        state_dir=state_dir,
    )
    if gates.enabled("DynamicPartitioning"):
        reset = state.destroy_unknown_partitions()
        if reset:
            logger.warning("startup reconciliation reset %d GPU(s) to SPX", reset)

    driver = GpuDriver(
        state=state,
        claim_resolver=k8s_claim_resolver(client),
        node_name=args.node_name,
        metrics=metrics,
    )
    socks = driver.start(plugin_dir=args.plugin_dir, registry_dir=args.registry_dir)
    logger.info("serving DRA on %s", socks["dra"])

    # KEP-4815 partitionable slices: keyed on API-server version when auto
    # (the shouldUseSplitResourceSlices probe analog, ref driver.go:574-603)
    if args.partitionable_slices == "auto":
        partitionable = client.server_version() >= (1, 33)
    else:
        partitionable = args.partitionable_slices == "true"
    logger.info("partitionable ResourceSlices: %s", partitionable)

    gen = ResourceSliceGenerator(
        devicelib, node_name=args.node_name,
        partitionable=partitionable,
        extended_metadata=gates.enabled("DeviceMetadata"),
        vfio=gates.enabled("PassthroughSupport"),
    )

    def publish(taints=None):
        # one generator instance: the pool generation increments per publish
        gen.taints = taints or gen.taints
        for sl in gen.generate():
            client.apply("resourceslices", sl)

    publish()

    # periodic re-enumeration + republish: picks up device topology changes
    # (hotplug, partition drift) even without health events
    def refresh_loop():
        while not stop.wait(300):
            try:
                devicelib.invalidate()
                publish()
            except Exception:
                logger.exception("periodic slice refresh failed")

    threading.Thread(target=refresh_loop, daemon=True, name="slice-refresh").start()

    stoppables = []
    if gates.enabled("DeviceHealthCheck"):
        tracker = TaintTracker(devicelib, publish)
        skip = {s.strip() for s in args.additional_events_to_ignore.split(",") if s.strip()}
        monitor = HealthMonitor(devicelib, tracker.on_events, additional_skip=skip).start()
        stoppables.append(monitor)
    cleanup = CheckpointCleanupManager(state, client).start()
    stoppables.append(cleanup)
    health = None
    if args.healthcheck_port >= 0:
        health = HealthServer(socks["dra"], socks.get("registration", ""))
        port = health.start(args.healthcheck_port)
        logger.info("healthcheck on 127.0.0.1:%d", port)
    if args.metrics_port:
        metrics.serve(args.metrics_port)

    stop.wait()
    for s in stoppables:
        s.stop()
    if health:
        health.stop()
    driver.stop()
    return 0


if __name__ == "__main__":
    raise SystemExit(main())
