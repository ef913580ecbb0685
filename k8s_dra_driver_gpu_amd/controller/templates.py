"""Rendered object templates for the ComputeDomain controller.

The analog of the reference's runtime-rendered ``templates/*.tmpl.yaml``
(``compute-domain-daemon.tmpl.yaml``,
``compute-domain-{daemon,workload}-claim-template.tmpl.yaml``): per-CD
DaemonSet + two ResourceClaimTemplates, built as dicts.
"""

from __future__ import annotations

from typing import Any, Dict

from .. import API_GROUP, COMPUTE_DOMAIN_DRIVER_NAME

CD_LABEL_KEY = f"{API_GROUP}/computeDomain"  # node + object label carrying CD UID
DAEMON_DEVICE_CLASS = f"compute-domain-daemon.{API_GROUP.split('.', 1)[-1]}"


def cd_label(cd_uid: str) -> Dict[str, str]:
    return {CD_LABEL_KEY: cd_uid}


def daemon_set(
    cd_name: str,
    cd_uid: str,
    namespace: str,
    image: str = "amd-dra-driver:latest",
    daemon_rct_name: str = "",
    feature_gates: str = "",
    max_nodes: int = 8,
) -> Dict[str, Any]:
    """Per-CD DaemonSet running the fabric daemon supervisor
    (ref compute-domain-daemon.tmpl.yaml + daemonset.go:190-253)."""
    daemon_rct_name = daemon_rct_name or f"{cd_name}-daemon-claim"
    labels = {**cd_label(cd_uid), "app.kubernetes.io/name": "compute-domain-daemon"}
    return {
        "apiVersion": "apps/v1",
        "kind": "DaemonSet",
        "metadata": {
            "name": f"{cd_name}-daemon",
            "namespace": namespace,
            "labels": dict(labels),
        },
        "spec": {
            "selector": {"matchLabels": dict(labels)},
            "template": {
                "metadata": {"labels": dict(labels)},
                "spec": {
                    # the chart creates this SA in every namespace the
                    # controller may deploy daemons into
                    # (rbac-compute-domain-daemon.yaml)
                    "serviceAccountName": "amd-dra-cd-daemon",
                    # scheduled only onto nodes labeled into this CD
                    "nodeSelector": cd_label(cd_uid),
                    "containers": [
                        {
                            "name": "compute-domain-daemon",
                            "image": image,
                            "command": ["python", "-m",
                                        "k8s_dra_driver_gpu_amd.daemon.main", "run"],
                            "env": [
                                {"name": "CD_UID", "value": cd_uid},
                                {"name": "CD_NAME", "value": cd_name},
                                {"name": "CD_NAMESPACE", "value": namespace},
                                {"name": "CD_MAX_NODES", "value": str(max_nodes)},
                                {"name": "FEATURE_GATES", "value": feature_gates},
                                {"name": "NODE_NAME",
                                 "valueFrom": {"fieldRef": {"fieldPath": "spec.nodeName"}}},
                                {"name": "POD_IP",
                                 "valueFrom": {"fieldRef": {"fieldPath": "status.podIP"}}},
                            ],
                            # probes exec the check subcommand; budget 20 min
                            # (ref compute-domain-daemon.tmpl.yaml:78-98)
                            "startupProbe": {
                                "exec": {"command": ["python", "-m",
                                                     "k8s_dra_driver_gpu_amd.daemon.main",
                                                     "check"]},
                                "periodSeconds": 1,
                                "failureThreshold": 1200,
                            },
                            "readinessProbe": {
                                "exec": {"command": ["python", "-m",
                                                     "k8s_dra_driver_gpu_amd.daemon.main",
                                                     "check"]},
                                "periodSeconds": 5,
                            },
                        }
                    ],
                    "resourceClaims": [
                        {"name": "daemon-device",
                         "resourceClaimTemplateName": daemon_rct_name}
                    ],
                },
            },
        },
    }


def daemon_claim_template(cd_name: str, cd_uid: str, namespace: str) -> Dict[str, Any]:
    """RCT for the daemon pods (ref resourceclaimtemplate.go:281-338)."""
    from ..api.configs import APIVERSION

    return {
        "apiVersion": "resource.k8s.io/v1beta1",
        "kind": "ResourceClaimTemplate",
        "metadata": {
            "name": f"{cd_name}-daemon-claim",
            "namespace": namespace,
            "labels": cd_label(cd_uid),
        },
        "spec": {
            "spec": {
                "devices": {
                    "requests": [
                        {
                            "name": "daemon",
                            "deviceClassName": f"compute-domain-daemon.{API_GROUP}",
                        }
                    ],
                    "config": [
                        {
                            "requests": ["daemon"],
                            "opaque": {
                                "driver": COMPUTE_DOMAIN_DRIVER_NAME,
                                "parameters": {
                                    "apiVersion": APIVERSION,
                                    "kind": "ComputeDomainDaemonConfig",
                                    "domainID": cd_uid,
                                },
                            },
                        }
                    ],
                }
            }
        },
    }


def workload_claim_template(
    cd_name: str, cd_uid: str, namespace: str, rct_name: str, allocation_mode: str = "Single"
) -> Dict[str, Any]:
    """RCT for workload pods' channel claims (ref resourceclaimtemplate.go:341-399)."""
    from ..api.configs import APIVERSION

    return {
        "apiVersion": "resource.k8s.io/v1beta1",
        "kind": "ResourceClaimTemplate",
        "metadata": {
            "name": rct_name,
            "namespace": namespace,
            "labels": cd_label(cd_uid),
        },
        "spec": {
            "spec": {
                "devices": {
                    "requests": [
                        {
                            "name": "channel",
                            "deviceClassName": f"compute-domain-default-channel.{API_GROUP}",
                        }
                    ],
                    "config": [
                        {
                            "requests": ["channel"],
                            "opaque": {
                                "driver": COMPUTE_DOMAIN_DRIVER_NAME,
                                "parameters": {
                                    "apiVersion": APIVERSION,
                                    "kind": "ComputeDomainChannelConfig",
                                    "domainID": cd_uid,
                                    "allocationMode": allocation_mode,
                                },
                            },
                        }
                    ],
                }
            }
        },
    }
