"""Every shipped manifest validates against the pinned k8s OpenAPI subset.

kubectl's server-side validation is what the reference relies on at apply
time (reference README.md steps 2/6/7/10); with no cluster available, the
pinned-schema validator (mi355x_gpu_hpa/k8s_schema.py) is its automated
offline equivalent — strict about unknown fields, enum values, required
fields, and the admission-time cross-field invariants (selector/template
match, probe ports, HPA metric stanzas).
"""

from pathlib import Path

import pytest
import yaml

from mi355x_gpu_hpa.k8s_schema import validate_manifest

DEPLOY = Path(__file__).resolve().parent.parent / "deploy"

# kustomization + Helm values files are not k8s objects; everything else is
SKIP = {"kustomization.yaml", "kube-prometheus-stack-values.yaml"}

MANIFESTS = sorted(
    p for p in list(DEPLOY.glob("*.yaml")) + list(DEPLOY.glob("*/*.yaml"))
    if p.name not in SKIP
)


def _docs(path):
    return [d for d in yaml.safe_load_all(path.read_text()) if d]


@pytest.mark.parametrize("path", MANIFESTS, ids=lambda p: str(p.relative_to(DEPLOY)))
def test_manifest_validates(path):
    for doc in _docs(path):
        errors = validate_manifest(doc)
        assert not errors, "\n".join(str(e) for e in errors)


def test_all_manifests_found():
    # the parametrization must actually cover the shipped surface
    names = {p.name for p in MANIFESTS}
    for expected in ("mi355x-exporter.yaml", "cuda-test-deployment.yaml",
                     "cuda-test-hpa.yaml", "cuda-test-prometheusrule.yaml",
                     "cuda-test-hpa-multi.yaml"):
        assert expected in names


class TestValidatorCatchesRealMistakes:
    """The validator must be able to FAIL — each case is a plausible edit
    mistake kubectl would reject."""

    def _exporter_ds(self):
        docs = _docs(DEPLOY / "mi355x-exporter.yaml")
        return next(d for d in docs if d["kind"] == "DaemonSet")

    def test_unknown_field_rejected(self):
        ds = self._exporter_ds()
        ds["spec"]["template"]["spec"]["containers"][0]["tolerations"] = []
        errs = validate_manifest(ds)
        assert any("unknown field 'tolerations'" in str(e) for e in errs)

    def test_selector_template_mismatch_rejected(self):
        ds = self._exporter_ds()
        ds["spec"]["template"]["metadata"]["labels"] = {"app": "other"}
        errs = validate_manifest(ds)
        assert any("not matched by template labels" in str(e) for e in errs)

    def test_v2beta1_rejected(self):
        # the reference's apiVersion (cuda-test-hpa.yaml:1) is gone from
        # k8s >=1.26; shipping it would break apply
        hpa = _docs(DEPLOY / "cuda-test-hpa.yaml")[0]
        hpa["apiVersion"] = "autoscaling/v2beta1"
        errs = validate_manifest(hpa)
        assert any("autoscaling/v2beta1" in str(e) for e in errs)

    def test_object_metric_without_stanza_rejected(self):
        hpa = _docs(DEPLOY / "cuda-test-hpa.yaml")[0]
        del hpa["spec"]["metrics"][0]["object"]
        errs = validate_manifest(hpa)
        assert any("requires a 'object' stanza" in str(e) for e in errs)

    def test_bad_probe_port_rejected(self):
        ds = self._exporter_ds()
        ds["spec"]["template"]["spec"]["containers"][0][
            "readinessProbe"]["httpGet"]["port"] = 9999
        errs = validate_manifest(ds)
        assert any("probe port" in str(e) for e in errs)

    def test_min_over_max_rejected(self):
        hpa = _docs(DEPLOY / "cuda-test-hpa.yaml")[0]
        hpa["spec"]["minReplicas"] = 99
        errs = validate_manifest(hpa)
        assert any("maxReplicas" in str(e) for e in errs)

    def test_bad_enum_rejected(self):
        ds = self._exporter_ds()
        ds["spec"]["updateStrategy"]["type"] = "Rolling"
        errs = validate_manifest(ds)
        assert any("'Rolling' not in" in str(e) for e in errs)

    def test_unpinned_kind_rejected(self):
        errs = validate_manifest({"kind": "FancyNewThing"})
        assert errs and "no pinned schema" in str(errs[0])


class TestReferenceManifestsAgainstModernSchema:
    """The validator applied to the REFERENCE's own manifests: its HPA uses
    autoscaling/v2beta1 (reference cuda-test-hpa.yaml:1), which modern
    clusters (>=1.26) reject — exactly the apply-time failure this stack's
    rewritten manifest avoids. The validator must reproduce that verdict."""

    REF = Path("/root/reference")

    def test_reference_hpa_rejected_for_v2beta1(self):
        if not (self.REF / "cuda-test-hpa.yaml").exists():
            pytest.skip("reference tree not present")
        doc = yaml.safe_load((self.REF / "cuda-test-hpa.yaml").read_text())
        errors = validate_manifest(doc)
        assert any("autoscaling/v2beta1" in str(e) for e in errors), errors

    def test_reference_deployment_validates(self):
        # the workload manifest itself is schema-clean (only the GPU
        # resource name differs from ours)
        if not (self.REF / "cuda-test-deployment.yaml").exists():
            pytest.skip("reference tree not present")
        doc = yaml.safe_load(
            (self.REF / "cuda-test-deployment.yaml").read_text())
        assert validate_manifest(doc) == []


def test_cli_validates_shipped_manifests(tmp_path):
    """`python -m mi355x_gpu_hpa.k8s_schema deploy/*.yaml` — the operator
    CLI exits 0 on the shipped surface and 1 on a broken manifest."""
    import subprocess
    import sys

    files = [str(p) for p in MANIFESTS]
    p = subprocess.run([sys.executable, "-m", "mi355x_gpu_hpa.k8s_schema"]
                       + files, capture_output=True, timeout=60,
                       cwd=DEPLOY.parent)
    assert p.returncode == 0, p.stdout.decode()
    assert p.stdout.decode().count("OK") >= len(files)

    f = tmp_path / "bad_manifest.yaml"
    f.write_text("apiVersion: autoscaling/v2beta1\nkind: HorizontalPodAutoscaler\n"
                 "metadata: {name: x}\nspec: {maxReplicas: 3,\n"
                 "  scaleTargetRef: {kind: Deployment, name: x}}\n")
    p = subprocess.run([sys.executable, "-m", "mi355x_gpu_hpa.k8s_schema",
                        str(f)], capture_output=True, timeout=60,
                       cwd=DEPLOY.parent)
    assert p.returncode == 1
    assert b"v2beta1" in p.stdout
