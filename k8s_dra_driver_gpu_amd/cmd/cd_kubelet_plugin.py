"""ComputeDomain kubelet plugin entrypoint (the
``cmd/compute-domain-kubelet-plugin/main.go`` analog)."""

from __future__ import annotations

import argparse
import logging
import os
import signal
import threading
from concurrent import futures

import grpc

from .. import COMPUTE_DOMAIN_DRIVER_NAME
from ..cdplugin.plugin import ComputeDomainPlugin
from ..device.devicelib import DeviceLib
from ..dra import api as dra
from ..k8s.client import FakeClient, HttpClient
from ..utils.debug import dump_config, install_stack_dump_handler
from ..utils.paths import check_unix_socket_path

logger = logging.getLogger("amddra.cmd.cd")


def parse_args(argv=None):
    p = argparse.ArgumentParser("compute-domain-kubelet-plugin")
    env = os.environ.get
    p.add_argument("--node-name", default=env("NODE_NAME", "node"))
    p.add_argument(
        "--plugin-dir",
        default=env("PLUGIN_DIR", f"/var/lib/kubelet/plugins/{COMPUTE_DOMAIN_DRIVER_NAME}"),
    )
    p.add_argument("--registry-dir",
                   default=env("PLUGINS_REGISTRY_DIR", "/var/lib/kubelet/plugins_registry"))
    p.add_argument("--cdi-root", default=env("CDI_ROOT", "/var/run/cdi"))
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
    dump_config("compute-domain-kubelet-plugin", vars(args))

    client = (HttpClient() if (args.in_cluster or os.environ.get("AMDDRA_API_SERVER"))
              else FakeClient())
    plugin = ComputeDomainPlugin(
        client=client,
        devicelib=DeviceLib(),
        state_dir=args.plugin_dir,
        node_name=args.node_name,
    )
    logger.info("clique id on this node: %r", plugin.clique_id())

    os.makedirs(args.plugin_dir, exist_ok=True)
    dra_sock = os.path.join(args.plugin_dir, "dra.sock")
    try:
        os.unlink(dra_sock)
    except FileNotFoundError:
        pass
    server = grpc.server(futures.ThreadPoolExecutor(max_workers=8))
    plugin.add_to_server(server)
    server.add_insecure_port(f"unix://{check_unix_socket_path(dra_sock)}")
    server.start()

    registration = dra.RegistrationServicer(
        name=COMPUTE_DOMAIN_DRIVER_NAME, endpoint=dra_sock
    )
    os.makedirs(args.registry_dir, exist_ok=True)
    reg_sock = os.path.join(args.registry_dir, f"{COMPUTE_DOMAIN_DRIVER_NAME}-reg.sock")
    try:
        os.unlink(reg_sock)
    except FileNotFoundError:
        pass
    reg_server = grpc.server(futures.ThreadPoolExecutor(max_workers=2))
    registration.add_to_server(reg_server)
    reg_server.add_insecure_port(f"unix://{check_unix_socket_path(reg_sock)}")
    reg_server.start()

    client.apply("resourceslices", plugin.resource_slice())
    logger.info("serving DRA on %s", dra_sock)

    # 10-min stale domain-dir cleanup (ref computedomain.go:384-439)
    def cleanup_loop():
        while not stop.wait(600):
            try:
                plugin.cleanup_stale_domain_dirs()
                plugin.cleanup_stale_claims()
            except Exception:
                logger.exception("stale domain cleanup failed")

    threading.Thread(target=cleanup_loop, daemon=True).start()
    stop.wait()
    server.stop(1)
    reg_server.stop(1)
    return 0


if __name__ == "__main__":
    raise SystemExit(main())
