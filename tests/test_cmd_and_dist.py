"""Entry-point smoke tests + multi-process (gloo, world_size=2) distributed
coverage of the bench harness and the RCCL validation workload."""

import json
import os
import signal
import subprocess
import sys
import time

import pytest

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def _mock_env(tmp_path, num_gpus=1):
    from k8s_dra_driver_gpu_amd.device.mock import MockTree

    tree = MockTree(root=str(tmp_path / "mock"), num_gpus=num_gpus)
    tree.setup()
    env = dict(os.environ)
    env.update(
        {
            "AMDDRA_SYSFS_ROOT": tree.sysfs_root,
            "AMDDRA_DEV_ROOT": tree.dev_root,
            "PYTHONPATH": REPO,
        }
    )
    return tree, env


class TestEntryPoints:
    def test_gpu_plugin_starts_and_stops(self, tmp_path):
        tree, env = _mock_env(tmp_path)
        env.update(
            {
                "PLUGIN_DIR": str(tmp_path / "plugin"),
                "PLUGINS_REGISTRY_DIR": str(tmp_path / "registry"),
                "CDI_ROOT": str(tmp_path / "cdi"),
                "HEALTHCHECK_PORT": "0",
            }
        )
        proc = subprocess.Popen(
            [sys.executable, "-m", "k8s_dra_driver_gpu_amd.cmd.gpu_kubelet_plugin"],
            env=env, cwd=REPO, stdout=subprocess.PIPE, stderr=subprocess.STDOUT, text=True,
        )
        try:
            # the DRA socket is served first, the registration socket right
            # after (driver.py start order) — wait for BOTH within the deadline
            deadline = time.monotonic() + 30
            sock = tmp_path / "plugin" / "dra.sock"

            def regs():
                return list((tmp_path / "registry").glob("*-reg.sock"))

            while time.monotonic() < deadline and not (sock.exists() and regs()):
                assert proc.poll() is None, proc.stdout.read()
                time.sleep(0.1)
            assert sock.exists()
            assert regs()
            # SIGUSR2 -> live stack dump (ref test_basics.bats "SIGUSR2
            # handler"); must not kill or destabilize the process
            proc.send_signal(signal.SIGUSR2)
            time.sleep(0.5)
            assert proc.poll() is None
            proc.send_signal(signal.SIGTERM)
            assert proc.wait(timeout=10) == 0
            out = proc.stdout.read()
            assert "SIGUSR2 stack dump written" in out, out[-1500:]
        finally:
            if proc.poll() is None:
                proc.kill()

    def test_cd_plugin_starts_and_stops(self, tmp_path):
        import tempfile

        tree, env = _mock_env(tmp_path)
        # socket dirs must stay short: AF_UNIX caps the path at 107 bytes
        # and xdist's nested tmp_path exceeds it (the plugin now fails
        # loudly for this; see utils/paths.check_unix_socket_path)
        short = tempfile.mkdtemp(prefix="cdp-", dir="/tmp")
        env.update(
            {
                "PLUGIN_DIR": os.path.join(short, "p"),
                "PLUGINS_REGISTRY_DIR": os.path.join(short, "r"),
            }
        )
        proc = subprocess.Popen(
            [sys.executable, "-m", "k8s_dra_driver_gpu_amd.cmd.cd_kubelet_plugin"],
            env=env, cwd=REPO, stdout=subprocess.PIPE, stderr=subprocess.STDOUT, text=True,
        )
        try:
            deadline = time.monotonic() + 30
            from pathlib import Path
            sock = Path(short) / "p" / "dra.sock"
            while time.monotonic() < deadline and not sock.exists():
                assert proc.poll() is None, proc.stdout.read()
                time.sleep(0.1)
            assert sock.exists()
            proc.send_signal(signal.SIGTERM)
            rc = proc.wait(timeout=10)
            assert rc == 0, f"exit {rc}: {proc.stdout.read()[-1500:]}"
        finally:
            if proc.poll() is None:
                proc.kill()

    def test_prestart_validates_rocm_root(self, tmp_path):
        host = tmp_path / "host"
        (host / "opt/rocm/lib").mkdir(parents=True)
        (host / "opt/rocm/lib/libamdhip64.so").touch()
        (host / "opt/rocm/lib/libhsa-runtime64.so").touch()
        parent = tmp_path / "parent"
        parent.mkdir()
        env = dict(os.environ, HOST_ROOT=str(host), DRIVER_ROOT_PARENT=str(parent),
                   PYTHONPATH=REPO)
        r = subprocess.run(
            [sys.executable, "-m", "k8s_dra_driver_gpu_amd.cmd.prestart"],
            env=env, cwd=REPO, capture_output=True, text=True,
        )
        assert r.returncode == 0, r.stderr
        assert (parent / "driver-root").is_symlink()

    def test_prestart_rejects_bad_root(self, tmp_path):
        env = dict(os.environ, HOST_ROOT=str(tmp_path), PYTHONPATH=REPO)
        r = subprocess.run(
            [sys.executable, "-m", "k8s_dra_driver_gpu_amd.cmd.prestart"],
            env=env, cwd=REPO, capture_output=True, text=True,
        )
        assert r.returncode == 1
        assert "invalid" in r.stderr


def _torchrun(module_or_script, env, nproc=2, extra=None, timeout=240):
    cmd = [
        sys.executable, "-m", "torch.distributed.run",
        "--standalone", "--local-addr", "127.0.0.1",
        f"--nproc-per-node={nproc}",
    ] + (module_or_script if isinstance(module_or_script, list) else [module_or_script]) + (
        extra or []
    )
    return subprocess.run(cmd, env=env, cwd=REPO, capture_output=True, text=True,
                          timeout=timeout)


class TestDistributedCpu:
    def test_bench_two_ranks_gloo(self, tmp_path):
        """bench.py under torchrun with 2 CPU ranks (gloo): the distributed
        code path the driver uses for the round-end scaling runs."""
        _, env = _mock_env(tmp_path, num_gpus=2)
        r = _torchrun("bench.py", env, nproc=2, extra=["--steps", "20", "--warmup", "2"])
        assert r.returncode == 0, r.stderr[-3000:]
        line = [l for l in r.stdout.splitlines() if l.startswith("{")][-1]
        out = json.loads(line)
        assert out["n_gpus"] == 2
        assert out["metric"] == "resourceclaim_pods_per_sec"
        assert out["value"] > 0
        assert out["config"]["parallelism"] == "dp2"
        assert out["config"]["fabric"]["rccl_allreduce_ok"] is True

    def test_rccl_validate_two_ranks_gloo(self, tmp_path):
        _, env = _mock_env(tmp_path, num_gpus=2)
        r = _torchrun(["-m", "k8s_dra_driver_gpu_amd.fabric.rccl_validate"], env, nproc=2)
        assert r.returncode == 0, r.stderr[-3000:]
        assert "RESULT bandwidth:" in r.stdout
        results = json.loads(
            [l for l in r.stdout.splitlines() if l.startswith("RESULTS:")][-1][len("RESULTS:"):]
        )
        assert results["allreduce_correct"] is True


class TestInspectCli:
    def test_inspect_table_and_json(self, tmp_path):
        tree, env = _mock_env(tmp_path, num_gpus=2)
        r = subprocess.run(
            [sys.executable, "-m", "k8s_dra_driver_gpu_amd.cmd.inspect"],
            env=env, cwd=REPO, capture_output=True, text=True, timeout=60,
        )
        assert r.returncode == 0, r.stderr
        assert "gpu-0: AMD Instinct MI355X" in r.stdout
        assert "clique=hive-" in r.stdout
        r = subprocess.run(
            [sys.executable, "-m", "k8s_dra_driver_gpu_amd.cmd.inspect", "--json"],
            env=env, cwd=REPO, capture_output=True, text=True, timeout=60,
        )
        out = json.loads(r.stdout)
        assert len(out["gpus"]) == 2


class TestHelmChart:
    """Chart rendered through helmlite (the in-repo Helm-subset renderer;
    no helm binary in the image) — the structural verification the
    reference gets from `helm template` in CI (tests/bats/test_basics.bats),
    plus golden checks for the templated knobs."""

    CHART = os.path.join(REPO, "deployments", "helm", "amd-dra-driver")

    def _render(self, overrides=None, **kw):
        from k8s_dra_driver_gpu_amd.utils.helmlite import render_chart

        return render_chart(self.CHART, overrides or {}, **kw)

    def _docs(self, overrides=None, **kw):
        import glob

        import yaml

        docs = []
        for text in self._render(overrides, **kw).values():
            for d in yaml.safe_load_all(text):
                if d:
                    docs.append(d)
        for f in glob.glob(os.path.join(self.CHART, "crds", "*.yaml")):
            for d in yaml.safe_load_all(open(f)):
                if d:
                    docs.append(d)
        return docs

    def test_all_manifests_parse_and_have_kind(self):
        docs = self._docs()
        assert len(docs) >= 12
        for d in docs:
            assert d.get("kind"), d
            assert (d.get("metadata") or {}).get("name"), d

    def test_webhook_service_backs_the_vwc(self):
        docs = self._docs()
        vwc = next(d for d in docs if d["kind"] == "ValidatingWebhookConfiguration")
        svc_ref = vwc["webhooks"][0]["clientConfig"]["service"]
        services = [d for d in docs if d["kind"] == "Service"]
        assert any(s["metadata"]["name"] == svc_ref["name"] for s in services)
        svc = next(s for s in services if s["metadata"]["name"] == svc_ref["name"])
        assert any(p["port"] == svc_ref["port"] for p in svc["spec"]["ports"])
        # int-typed k8s fields must hold real ints (quoted Go-templating is
        # only valid inside string-typed fields like env values)
        for p in svc["spec"]["ports"]:
            assert isinstance(p["port"], int)
            assert isinstance(p["targetPort"], int)
        assert isinstance(svc_ref["port"], int)
        # the Service selector matches the webhook Deployment's pod labels
        dep = next(d for d in docs if d["kind"] == "Deployment"
                   and d["metadata"]["name"] == svc_ref["name"])
        pod_labels = dep["spec"]["template"]["metadata"]["labels"]
        for k, v in svc["spec"]["selector"].items():
            assert pod_labels.get(k) == v

    def test_crds_and_deviceclasses_complete(self):
        docs = self._docs()
        crds = {d["metadata"]["name"] for d in docs
                if d["kind"] == "CustomResourceDefinition"}
        assert crds == {"computedomains.resource.amd.com",
                        "computedomaincliques.resource.amd.com"}
        dcs = {d["metadata"]["name"] for d in docs if d["kind"] == "DeviceClass"}
        assert {"gpu.amd.com", "partition.gpu.amd.com", "vfio.gpu.amd.com"} <= dcs
        # default (v1beta1): extendedResourceName must NOT appear — it is a
        # v1-only field (ref deviceclass-gpu.yaml:12-14)
        gpu_dc = next(d for d in docs if d["kind"] == "DeviceClass"
                      and d["metadata"]["name"] == "gpu.amd.com")
        assert gpu_dc["apiVersion"] == "resource.k8s.io/v1beta1"
        assert "extendedResourceName" not in gpu_dc["spec"]

    def test_resource_api_version_templating(self):
        """resourceApiVersion=v1: classes serve under resource.k8s.io/v1 and
        gpu.amd.com gains extendedResourceName."""
        docs = self._docs({"resourceApiVersion": "v1"})
        dcs = [d for d in docs if d["kind"] == "DeviceClass"]
        assert all(d["apiVersion"] == "resource.k8s.io/v1" for d in dcs)
        gpu_dc = next(d for d in dcs if d["metadata"]["name"] == "gpu.amd.com")
        assert gpu_dc["spec"]["extendedResourceName"] == "amd.com/gpu"

    def test_network_policies_gated_and_shaped(self):
        docs = self._docs()
        assert not any(d["kind"] == "NetworkPolicy" for d in docs)
        docs = self._docs({
            "controller": {"networkPolicy": {"enabled": True}},
            "kubeletPlugin": {"networkPolicy": {"enabled": True}},
            "webhook": {"networkPolicy": {"enabled": True}},
        })
        nps = {d["metadata"]["name"]: d for d in docs if d["kind"] == "NetworkPolicy"}
        assert set(nps) == {"amd-dra-controller", "amd-dra-kubelet-plugin",
                            "amd-dra-webhook"}
        kp = nps["amd-dra-kubelet-plugin"]
        ports = {p["port"] for rule in kp["spec"]["egress"] for p in rule["ports"]}
        assert {443, 6443, 50000, 50005} <= ports  # fabricd mesh ports
        wh = nps["amd-dra-webhook"]
        assert "Ingress" in wh["spec"]["policyTypes"]

    def test_webhook_cert_manager_mode(self):
        docs = self._docs()  # default: cert-manager + selfsigned
        kinds = {d["kind"] for d in docs}
        assert {"Issuer", "Certificate"} <= kinds
        cert = next(d for d in docs if d["kind"] == "Certificate")
        assert cert["spec"]["secretName"] == "amd-dra-webhook-cert"
        assert cert["spec"]["issuerRef"]["name"] == "amd-dra-webhook-issuer"
        assert any("svc" in n for n in cert["spec"]["dnsNames"])
        vwc = next(d for d in docs if d["kind"] == "ValidatingWebhookConfiguration")
        assert "cert-manager.io/inject-ca-from" in vwc["metadata"]["annotations"]
        dep = next(d for d in docs if d["kind"] == "Deployment"
                   and d["metadata"]["name"] == "amd-dra-webhook")
        vol = dep["spec"]["template"]["spec"]["volumes"][0]
        assert vol["secret"]["secretName"] == "amd-dra-webhook-cert"

    def test_webhook_secret_mode(self):
        docs = self._docs({"webhook": {"tls": {
            "mode": "secret", "secretName": "my-tls", "caBundle": "QUJD"}}})
        kinds = {d["kind"] for d in docs}
        assert "Issuer" not in kinds and "Certificate" not in kinds
        vwc = next(d for d in docs if d["kind"] == "ValidatingWebhookConfiguration")
        assert vwc["webhooks"][0]["clientConfig"]["caBundle"] == "QUJD"
        dep = next(d for d in docs if d["kind"] == "Deployment"
                   and d["metadata"]["name"] == "amd-dra-webhook")
        assert dep["spec"]["template"]["spec"]["volumes"][0]["secret"]["secretName"] == "my-tls"

    def test_webhook_disabled_removes_all_webhook_objects(self):
        docs = self._docs({"webhook": {"enabled": False}})
        for d in docs:
            assert "webhook" not in d["metadata"]["name"], d["metadata"]["name"]

    def test_validation_rejects_bad_values(self):
        from k8s_dra_driver_gpu_amd.utils.helmlite import HelmliteError

        with pytest.raises(HelmliteError, match="tls.mode"):
            self._render({"webhook": {"tls": {"mode": "bogus"}}})
        with pytest.raises(HelmliteError, match="resourceApiVersion"):
            self._render({"resourceApiVersion": "v2"})

    def test_scheduling_knobs(self):
        # kubelet plugin must outrank workload pods (system-node-critical);
        # tolerations and imagePullSecrets render from values lists
        import yaml

        out = self._render({
            "kubeletPlugin": {"tolerations": [{"operator": "Exists"}]},
            "imagePullSecrets": [{"name": "cred"}],
        })
        kp = list(yaml.safe_load_all(out["kubeletplugin.yaml"]))[0]
        spec = kp["spec"]["template"]["spec"]
        assert spec["priorityClassName"] == "system-node-critical"
        assert spec["tolerations"] == [{"operator": "Exists"}]
        assert spec["imagePullSecrets"] == [{"name": "cred"}]
        dep = [d for d in yaml.safe_load_all(self._render()["controller.yaml"])
               if d and d["kind"] == "Deployment"][0]
        assert (dep["spec"]["template"]["spec"]["priorityClassName"]
                == "system-cluster-critical")

    def test_default_namespace_rejected(self):
        from k8s_dra_driver_gpu_amd.utils.helmlite import HelmliteError

        with pytest.raises(HelmliteError, match="default.*namespace"):
            self._render({"namespace": "default"})
        # explicit override allows it
        self._render({"namespace": "default", "allowDefaultNamespace": True})

    def test_standard_labels_on_all_objects(self):
        docs = self._docs()
        for d in docs:
            if d["kind"] == "CustomResourceDefinition":
                continue  # crds/ ship unrendered
            labels = d["metadata"].get("labels") or {}
            assert labels.get("app.kubernetes.io/name") == "amd-dra-driver", d["metadata"]
            assert labels.get("app.kubernetes.io/managed-by") == "Helm"
            assert "helm.sh/chart" in labels

    def test_namespace_override(self):
        docs = self._docs({"namespaceOverride": "custom-ns"})
        namespaced = [d for d in docs if d["kind"] in
                      ("Deployment", "DaemonSet", "Service", "ServiceAccount")]
        assert namespaced
        for d in namespaced:
            assert d["metadata"]["namespace"] == "custom-ns", d["metadata"]

    def test_template_value_refs_resolve(self):
        """Every `.Values.x.y` referenced by a template must exist in
        values.yaml — the drift `helm lint` would catch in the reference's
        CI."""
        import glob
        import re

        import yaml

        vals = yaml.safe_load(open(os.path.join(self.CHART, "values.yaml")))
        refs = set()
        for f in glob.glob(os.path.join(self.CHART, "templates", "*.yaml")):
            refs.update(re.findall(r"\.Values\.([a-zA-Z0-9_.]+)", open(f).read()))
        assert refs, "no templated values found (templates moved?)"
        for r in sorted(refs):
            cur = vals
            for part in r.split("."):
                assert isinstance(cur, dict) and part in cur, \
                    f".Values.{r} not defined in values.yaml"
                cur = cur[part]

    def test_cd_daemon_rbac_and_serviceaccounts(self):
        # the per-CD daemon pods need their own SA in EVERY namespace the
        # controller may create DaemonSets in (release ns +
        # controller.additionalNamespaces), bound to a ClusterRole covering
        # cliques, CDs and pod labeling (ref rbac-compute-domain-daemon)
        import yaml

        docs = [d for d in yaml.safe_load_all(
            self._render({"controller": {"additionalNamespaces": "t1,t2"}})[
                "rbac-compute-domain-daemon.yaml"]) if d]
        sas = [d for d in docs if d["kind"] == "ServiceAccount"]
        assert {d["metadata"]["namespace"] for d in sas} == {
            "amd-dra-driver", "t1", "t2"}
        assert all(d["metadata"]["name"] == "amd-dra-cd-daemon" for d in sas)
        role = [d for d in docs if d["kind"] == "ClusterRole"][0]
        got = {(g, r): set(rule["verbs"]) for rule in role["rules"]
               for g in rule["apiGroups"] for r in rule["resources"]}
        assert "create" in got[("resource.amd.com", "computedomaincliques")]
        assert "patch" in got[("", "pods")]
        assert "update" in got[("resource.amd.com", "computedomains/status")]
        crb = [d for d in docs if d["kind"] == "ClusterRoleBinding"][0]
        assert {s["namespace"] for s in crb["subjects"]} == {
            "amd-dra-driver", "t1", "t2"}
        # the runtime-rendered DaemonSet runs as exactly that SA
        from k8s_dra_driver_gpu_amd.controller.templates import daemon_set

        ds = daemon_set("cd1", "uid1", "default")
        assert (ds["spec"]["template"]["spec"]["serviceAccountName"]
                == "amd-dra-cd-daemon")
        # default (no additional namespaces): single SA in the release ns
        docs1 = [d for d in yaml.safe_load_all(
            self._render()["rbac-compute-domain-daemon.yaml"]) if d]
        assert len([d for d in docs1 if d["kind"] == "ServiceAccount"]) == 1

    def test_rbac_covers_driver_resources(self):
        docs = self._docs()
        rules = []
        for d in docs:
            if d["kind"] in ("ClusterRole", "Role"):
                rules.extend(d.get("rules") or [])
        covered = set()
        for r in rules:
            for res in r.get("resources") or []:
                covered.add(res)
        for need in ("resourceclaims", "resourceslices", "computedomains",
                     "computedomaincliques", "leases"):
            assert need in covered, f"RBAC missing {need}"


class TestBenchContract:
    """bench.py is the driver's measurement contract: one JSON line on
    stdout with the BASELINE metric/config keys. Guard the schema so a
    refactor can't silently break the round-end benchmark run."""

    def test_default_invocation_json_line(self):
        proc = subprocess.run(
            [sys.executable, "bench.py", "--steps", "30", "--warmup", "5"],
            cwd=REPO, capture_output=True, text=True, timeout=240,
        )
        assert proc.returncode == 0, proc.stderr[-2000:]
        line = proc.stdout.strip().splitlines()[-1]
        out = json.loads(line)
        assert out["metric"] == "resourceclaim_pods_per_sec"
        assert out["unit"] == "pods/s"
        assert out["n_gpus"] == 1
        assert out["steps"] == 30 and out["warmup"] == 5
        assert isinstance(out["value"], (int, float)) and out["value"] > 0
        assert isinstance(out["ms_per_step"], (int, float))
        assert out["higher_is_better"] is True
        assert out["scaling"] == "weak"
        assert out["vs_baseline"] is None  # no published reference number
        assert out["dtype"] == "n/a"  # control-plane metric: no compute dtype
        assert "synthetic" in out["data"]
        cfg = out["config"]
        assert cfg["model"] == "dra-claim-churn"
        assert cfg["parallelism"] == "dp1"
        assert cfg["global_batch"] == 30
        assert "p50_alloc_latency_ms" in cfg and "p99_alloc_latency_ms" in cfg
        assert "cd_bringup_s" in cfg


class TestBenchMeshBringup:
    def test_mesh_bringup_two_daemons(self):
        """bench's N-daemon ComputeDomain mesh measurement (reported at
        N>1): controller + 2 real fabricd daemons reach Ready."""
        import importlib.util

        spec = importlib.util.spec_from_file_location(
            "benchmod", os.path.join(REPO, "bench.py"))
        m = importlib.util.module_from_spec(spec)
        spec.loader.exec_module(m)
        t = m._measure_mesh_bringup(2)
        assert 0 < t < 60, t
