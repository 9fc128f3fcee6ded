"""Manifest-contract tests: the deploy/ YAML surface must keep the exact
metric-name/label couplings the reference stack depends on (SURVEY.md §1:
"every layer couples to the next only through a metric name and a label
schema"). These tests parse the shipped manifests and evaluate the shipped
PromQL against the in-repo evaluator, so manifest drift breaks CI, not a
cluster."""

from pathlib import Path

import pytest
import yaml

from mi355x_gpu_hpa.control import Sample, evaluate_scalar

DEPLOY = Path(__file__).resolve().parent.parent / "deploy"


def load_all(path):
    return list(yaml.safe_load_all(path.read_text()))


class TestExporterManifest:
    def setup_method(self):
        self.docs = load_all(DEPLOY / "mi355x-exporter.yaml")
        self.ds = next(d for d in self.docs if d["kind"] == "DaemonSet")
        self.svc = next(d for d in self.docs if d["kind"] == "Service")

    def test_port_9400_contract(self):
        # reference: dcgm-exporter.yaml:31-32,75-77 — port 9400 end to end
        c = self.ds["spec"]["template"]["spec"]["containers"][0]
        assert c["ports"][0]["containerPort"] == 9400
        assert self.svc["spec"]["ports"][0]["port"] == 9400

    def test_no_sysadmin_no_nvidia(self):
        # check live YAML content (comments cite the reference and may name
        # NVIDIA paths; the actual spec must not). Unlike the reference
        # (dcgm-exporter.yaml:42-48) there is no SYS_ADMIN capability;
        # `privileged: true` IS expected — runc's device cgroup blocks
        # open() of hostPath char devices in non-privileged containers
        # (ADVICE round 1, medium).
        text = "\n".join(line for line in
                         (DEPLOY / "mi355x-exporter.yaml").read_text().splitlines()
                         if not line.lstrip().startswith("#"))
        for line in text.splitlines():
            stripped = line.split("#")[0].lower()
            assert "sys_admin" not in stripped
            assert "nvidia" not in stripped, line

    def test_privileged_for_device_cgroup(self):
        sc = self.ds["spec"]["template"]["spec"]["containers"][0]["securityContext"]
        assert sc.get("privileged") is True
        # and no added capabilities — privilege is device-cgroup passthrough
        # only, not a capability grant
        assert "capabilities" not in sc

    def test_amd_device_paths(self):
        vols = {v["name"]: v for v in self.ds["spec"]["template"]["spec"]["volumes"]}
        assert vols["dev-kfd"]["hostPath"]["path"] == "/dev/kfd"
        assert vols["dev-dri"]["hostPath"]["path"] == "/dev/dri"
        assert vols["pod-resources"]["hostPath"]["path"] == \
            "/var/lib/kubelet/pod-resources"

    def test_probes_defined(self):
        # the gap SURVEY.md §5.3 flags in the reference
        c = self.ds["spec"]["template"]["spec"]["containers"][0]
        assert c["readinessProbe"]["httpGet"]["path"] == "/readyz"
        assert c["livenessProbe"]["httpGet"]["path"] == "/healthz"

    def test_node_selector(self):
        sel = self.ds["spec"]["template"]["spec"]["nodeSelector"]
        assert sel == {"accelerator": "amd-gpu"}

    def test_service_selector_matches_pods(self):
        labels = self.ds["spec"]["template"]["metadata"]["labels"]
        assert self.svc["spec"]["selector"].items() <= labels.items()


class TestWorkloadManifest:
    def test_names_and_labels_are_reference_identical(self):
        d = load_all(DEPLOY / "cuda-test-deployment.yaml")[0]
        # contract: deployment name + app label join key (ref :13-14, rule :13)
        assert d["metadata"]["name"] == "cuda-test"
        assert d["spec"]["template"]["metadata"]["labels"]["app"] == "cuda-test"
        c = d["spec"]["template"]["spec"]["containers"][0]
        assert c["resources"]["limits"]["amd.com/gpu"] == 1


class TestRuleManifests:
    def eval_rule(self, expr, samples):
        return evaluate_scalar(expr, samples)

    def test_reference_rule_expression_evaluates(self):
        rule = load_all(DEPLOY / "cuda-test-prometheusrule.yaml")[0]
        g = rule["spec"]["groups"][0]
        r = g["rules"][0]
        assert r["record"] == "cuda_test_gpu_avg"
        assert r["labels"] == {"namespace": "default", "deployment": "cuda-test"}
        samples = [
            Sample("dcgm_gpu_utilization",
                   {"gpu": "0", "pod": "cuda-test-1", "node": "n0",
                    "namespace": "default"}, 42.0),
            Sample("kube_pod_labels",
                   {"pod": "cuda-test-1", "label_app": "cuda-test"}, 1.0),
        ]
        assert self.eval_rule(r["expr"], samples) == 42.0

    def test_multi_metric_rules_evaluate(self):
        rule = load_all(DEPLOY / "multi-metric" /
                        "gpu-metrics-prometheusrule.yaml")[0]
        rules = rule["spec"]["groups"][0]["rules"]
        by_name = {r["record"]: r for r in rules}
        assert set(by_name) == {"cuda_test_hbm_bw_avg",
                                "cuda_test_xgmi_link_util_avg"}
        samples = [
            Sample("amd_hbm_bandwidth_utilization",
                   {"gpu": "0", "pod": "cuda-test-1", "node": "n0",
                    "namespace": "default"}, 61.0),
            Sample("amd_xgmi_link_utilization",
                   {"gpu": "0", "pod": "cuda-test-1", "node": "n0",
                    "namespace": "default", "link": "0"}, 30.0),
            Sample("amd_xgmi_link_utilization",
                   {"gpu": "0", "pod": "cuda-test-1", "node": "n0",
                    "namespace": "default", "link": "1"}, 10.0),
            Sample("kube_pod_labels",
                   {"pod": "cuda-test-1", "label_app": "cuda-test"}, 1.0),
        ]
        assert self.eval_rule(by_name["cuda_test_hbm_bw_avg"]["expr"],
                              samples) == 61.0
        # busiest link wins via the inner max
        assert self.eval_rule(by_name["cuda_test_xgmi_link_util_avg"]["expr"],
                              samples) == 30.0


class TestHpaManifests:
    def test_hpa_targets_reference_values(self):
        h = load_all(DEPLOY / "cuda-test-hpa.yaml")[0]
        assert h["apiVersion"] == "autoscaling/v2"
        assert h["spec"]["minReplicas"] == 1
        assert h["spec"]["maxReplicas"] == 3  # ref cuda-test-hpa.yaml:12
        m = h["spec"]["metrics"][0]["object"]
        assert m["metric"]["name"] == "cuda_test_gpu_avg"
        assert m["target"]["value"] == "5"  # ref :21

    def test_multi_metric_hpa(self):
        h = load_all(DEPLOY / "multi-metric" / "cuda-test-hpa-multi.yaml")[0]
        assert h["spec"]["maxReplicas"] == 8
        names = {m["object"]["metric"]["name"] for m in h["spec"]["metrics"]}
        assert names == {"cuda_test_gpu_avg", "cuda_test_hbm_bw_avg"}


class TestScrapeValues:
    def test_scrape_job_contract(self):
        v = load_all(DEPLOY / "kube-prometheus-stack-values.yaml")[0]
        jobs = v["prometheus"]["prometheusSpec"]["additionalScrapeConfigs"]
        job = jobs[0]
        assert job["job_name"] == "gpu-metrics"
        assert job["scrape_interval"] == "1s"  # ref values.yaml:5
        relabels = job["relabel_configs"]
        node = next(r for r in relabels if r.get("target_label") == "node")
        assert node["source_labels"] == ["__meta_kubernetes_pod_node_name"]

    def test_ksm_label_allowlist(self):
        # kube-state-metrics >=2.0 exports no
        # label_* labels unless allowlisted — without this the rule's
        # `on(pod) group_left(label_app)` join is empty in a real cluster
        # and the HPA reads <unknown>.
        v = load_all(DEPLOY / "kube-prometheus-stack-values.yaml")[0]
        args = v["kube-state-metrics"]["extraArgs"]
        allow = next(a for a in args
                     if a.startswith("--metric-labels-allowlist="))
        spec = allow.split("=", 1)[1]
        # must cover pods' `app` label, the join key of the recording rule
        assert "pods=[" in spec and "app" in spec.split("pods=[", 1)[1]


class TestKustomize:
    def test_base_lists_existing_resources(self):
        k = yaml.safe_load((DEPLOY / "kustomization.yaml").read_text())
        for res in k["resources"]:
            assert (DEPLOY / res).exists(), res
        # every standalone top-level manifest is included
        listed = set(k["resources"])
        present = {p.name for p in DEPLOY.glob("*.yaml")
                   if p.name not in ("kustomization.yaml",
                                     "kube-prometheus-stack-values.yaml")}
        assert listed == present

    def test_overlay(self):
        k = yaml.safe_load(
            (DEPLOY / "multi-metric" / "kustomization.yaml").read_text())
        assert "../" in k["resources"]
        for res in k["resources"]:
            if res != "../":
                assert (DEPLOY / "multi-metric" / res).exists(), res


class TestKindHarnessScript:
    """deploy/kind/run.sh is the executable form of the runbook: its
    referenced manifests must exist and its probes must match the
    reference's verification steps (it cannot run here — no Docker — but
    it must not drift from the files it applies)."""

    def setup_method(self):
        self.script = (DEPLOY / "kind" / "run.sh").read_text()

    def test_applied_files_exist(self):
        import re
        for m in re.finditer(r"apply -f (\S+)", self.script):
            path = (DEPLOY / "kind" / m.group(1)).resolve()
            assert path.exists(), m.group(1)

    def test_values_file_referenced(self):
        assert "../kube-prometheus-stack-values.yaml" in self.script

    def test_reference_probes_present(self):
        # the four reference verification seams
        assert "dcgm_gpu_temp" in self.script              # README.md:42-47
        assert "cuda_test_gpu_avg" in self.script          # README.md:80-88
        assert "custom.metrics.k8s.io" in self.script      # README.md:98-102
        assert "hpa_scaled_up" in self.script              # README.md:112-122

    def test_fails_loudly(self):
        assert "set -euo pipefail" in self.script
        assert "fail()" in self.script
