"""k8s_schema.py — pinned-subset Kubernetes OpenAPI validation, offline.

kubectl performs server-side (OpenAPI) validation on apply; this build
environment has no cluster and no network, so the next-best proof that the
shipped manifests are applyable is a structural
validator against schemas pinned from the Kubernetes v1.29 OpenAPI spec and
the prometheus-operator monitoring.coreos.com/v1 CRD. The subset covers
every field the manifests under deploy/ actually use, plus the invariants
kubectl/apiserver enforce that are easiest to break silently:

  * apiVersion/kind pairing (e.g. HPA must be autoscaling/v2 on k8s >=1.26
    — the reference's v2beta1, cuda-test-hpa.yaml:1, no longer applies)
  * required fields per object (metadata.name, spec.selector, ...)
  * field types and enum values (imagePullPolicy, update strategy, ...)
  * label-selector consistency: spec.selector must match template labels
    (apps/v1 rejects the mismatch at admission)
  * probe/port cross-references

Unknown fields are errors (mirrors --validate=strict server-side
validation): a typo like `tolerations` under the container instead of the
pod spec must fail here the way the apiserver would fail it.
"""

from __future__ import annotations

from typing import Any, Dict, List

__all__ = ["validate_manifest", "SchemaError"]


class SchemaError(ValueError):
    def __init__(self, path: str, msg: str):
        self.path = path
        super().__init__(f"{path}: {msg}")


# --- schema vocabulary -----------------------------------------------------
# A schema node is one of:
#   {"type": "object", "fields": {name: node}, "required": [...],
#    "open": bool}        open=True permits unknown keys (map-like objects)
#   {"type": "array", "items": node, "min_items": int}
#   {"type": "string"} / {"type": "int"} / {"type": "bool"}
#   {"type": "string", "enum": [...]}
#   {"type": "int-or-string"}        (IntOrString, e.g. targetPort)
#   {"type": "map-string"}           (map[string]string: labels, selectors)
#   {"type": "quantity"}             (resource.Quantity: "200m", "64Mi", 1)
#   {"type": "any"}

def _check(node: Dict[str, Any], value: Any, path: str,
           errors: List[SchemaError]) -> None:
    t = node["type"]
    if t == "any":
        return
    if t == "object":
        if not isinstance(value, dict):
            errors.append(SchemaError(path, f"expected object, got {type(value).__name__}"))
            return
        fields = node.get("fields", {})
        for req in node.get("required", []):
            if req not in value:
                errors.append(SchemaError(path, f"missing required field {req!r}"))
        for k, v in value.items():
            if k in fields:
                _check(fields[k], v, f"{path}.{k}", errors)
            elif not node.get("open", False):
                errors.append(SchemaError(path, f"unknown field {k!r}"))
        return
    if t == "array":
        if not isinstance(value, list):
            errors.append(SchemaError(path, f"expected array, got {type(value).__name__}"))
            return
        if len(value) < node.get("min_items", 0):
            errors.append(SchemaError(path, f"needs >= {node['min_items']} items"))
        for i, item in enumerate(value):
            _check(node["items"], item, f"{path}[{i}]", errors)
        return
    if t == "string":
        if not isinstance(value, str):
            errors.append(SchemaError(path, f"expected string, got {value!r}"))
            return
        enum = node.get("enum")
        if enum and value not in enum:
            errors.append(SchemaError(path, f"{value!r} not in {enum}"))
        return
    if t == "int":
        if not isinstance(value, int) or isinstance(value, bool):
            errors.append(SchemaError(path, f"expected integer, got {value!r}"))
        return
    if t == "bool":
        if not isinstance(value, bool):
            errors.append(SchemaError(path, f"expected boolean, got {value!r}"))
        return
    if t == "int-or-string":
        if not isinstance(value, (int, str)) or isinstance(value, bool):
            errors.append(SchemaError(path, f"expected int-or-string, got {value!r}"))
        return
    if t == "map-string":
        if not isinstance(value, dict) or not all(
                isinstance(k, str) and isinstance(v, str)
                for k, v in value.items()):
            errors.append(SchemaError(path, "expected map[string]string"))
        return
    if t == "quantity":
        if isinstance(value, bool) or not isinstance(value, (int, float, str)):
            errors.append(SchemaError(path, f"expected quantity, got {value!r}"))
        return
    raise AssertionError(f"bad schema node type {t!r}")


# --- pinned schemas (Kubernetes v1.29 OpenAPI subset) ----------------------

_METADATA = {
    "type": "object", "required": ["name"],
    "fields": {
        "name": {"type": "string"},
        "namespace": {"type": "string"},
        "labels": {"type": "map-string"},
        "annotations": {"type": "map-string"},
    },
}

_LABEL_SELECTOR = {
    "type": "object",
    "fields": {
        "matchLabels": {"type": "map-string"},
        "matchExpressions": {"type": "array", "items": {"type": "any"}},
    },
}

_PROBE = {
    "type": "object",
    "fields": {
        "httpGet": {"type": "object", "required": ["path", "port"], "fields": {
            "path": {"type": "string"},
            "port": {"type": "int-or-string"},
            "scheme": {"type": "string", "enum": ["HTTP", "HTTPS"]},
        }},
        "exec": {"type": "object", "fields": {
            "command": {"type": "array", "items": {"type": "string"}}}},
        "initialDelaySeconds": {"type": "int"},
        "periodSeconds": {"type": "int"},
        "timeoutSeconds": {"type": "int"},
        "failureThreshold": {"type": "int"},
        "successThreshold": {"type": "int"},
    },
}

_RESOURCES = {
    "type": "object",
    "fields": {
        "requests": {"type": "object", "open": True, "fields": {}},
        "limits": {"type": "object", "open": True, "fields": {}},
    },
}

_CONTAINER = {
    "type": "object", "required": ["name", "image"],
    "fields": {
        "name": {"type": "string"},
        "image": {"type": "string"},
        "imagePullPolicy": {"type": "string",
                            "enum": ["Always", "IfNotPresent", "Never"]},
        "command": {"type": "array", "items": {"type": "string"}},
        "args": {"type": "array", "items": {"type": "string"}},
        "env": {"type": "array", "items": {
            "type": "object", "required": ["name"], "fields": {
                "name": {"type": "string"},
                "value": {"type": "string"},
                "valueFrom": {"type": "any"},
            }}},
        "ports": {"type": "array", "items": {
            "type": "object", "required": ["containerPort"], "fields": {
                "name": {"type": "string"},
                "containerPort": {"type": "int"},
                "protocol": {"type": "string", "enum": ["TCP", "UDP", "SCTP"]},
            }}},
        "securityContext": {"type": "object", "fields": {
            "runAsUser": {"type": "int"},
            "privileged": {"type": "bool"},
            "capabilities": {"type": "object", "fields": {
                "add": {"type": "array", "items": {"type": "string"}},
                "drop": {"type": "array", "items": {"type": "string"}},
            }},
        }},
        "resources": _RESOURCES,
        "readinessProbe": _PROBE,
        "livenessProbe": _PROBE,
        "volumeMounts": {"type": "array", "items": {
            "type": "object", "required": ["name", "mountPath"], "fields": {
                "name": {"type": "string"},
                "mountPath": {"type": "string"},
                "readOnly": {"type": "bool"},
            }}},
    },
}

_POD_SPEC = {
    "type": "object", "required": ["containers"],
    "fields": {
        "containers": {"type": "array", "items": _CONTAINER, "min_items": 1},
        "nodeSelector": {"type": "map-string"},
        "tolerations": {"type": "array", "items": {
            "type": "object", "fields": {
                "key": {"type": "string"},
                "operator": {"type": "string", "enum": ["Exists", "Equal"]},
                "value": {"type": "string"},
                "effect": {"type": "string",
                           "enum": ["NoSchedule", "PreferNoSchedule",
                                    "NoExecute"]},
            }}},
        "volumes": {"type": "array", "items": {
            "type": "object", "required": ["name"], "fields": {
                "name": {"type": "string"},
                "hostPath": {"type": "object", "required": ["path"], "fields": {
                    "path": {"type": "string"},
                    "type": {"type": "string"},
                }},
                "configMap": {"type": "any"},
                "emptyDir": {"type": "any"},
            }}},
        "serviceAccountName": {"type": "string"},
        "hostNetwork": {"type": "bool"},
        "priorityClassName": {"type": "string"},
    },
}

_POD_TEMPLATE = {
    "type": "object", "required": ["spec"],
    "fields": {
        "metadata": {"type": "object", "fields": {
            "labels": {"type": "map-string"},
            "annotations": {"type": "map-string"},
        }},
        "spec": _POD_SPEC,
    },
}

_DAEMONSET = {
    "type": "object", "required": ["apiVersion", "kind", "metadata", "spec"],
    "fields": {
        "apiVersion": {"type": "string", "enum": ["apps/v1"]},
        "kind": {"type": "string", "enum": ["DaemonSet"]},
        "metadata": _METADATA,
        "spec": {"type": "object", "required": ["selector", "template"],
                 "fields": {
                     "selector": _LABEL_SELECTOR,
                     "template": _POD_TEMPLATE,
                     "updateStrategy": {"type": "object", "fields": {
                         "type": {"type": "string",
                                  "enum": ["RollingUpdate", "OnDelete"]},
                         "rollingUpdate": {"type": "any"},
                     }},
                     "minReadySeconds": {"type": "int"},
                 }},
    },
}

_DEPLOYMENT = {
    "type": "object", "required": ["apiVersion", "kind", "metadata", "spec"],
    "fields": {
        "apiVersion": {"type": "string", "enum": ["apps/v1"]},
        "kind": {"type": "string", "enum": ["Deployment"]},
        "metadata": _METADATA,
        "spec": {"type": "object", "required": ["selector", "template"],
                 "fields": {
                     "replicas": {"type": "int"},
                     "selector": _LABEL_SELECTOR,
                     "template": _POD_TEMPLATE,
                     "strategy": {"type": "any"},
                     "minReadySeconds": {"type": "int"},
                 }},
    },
}

_SERVICE = {
    "type": "object", "required": ["apiVersion", "kind", "metadata", "spec"],
    "fields": {
        "apiVersion": {"type": "string", "enum": ["v1"]},
        "kind": {"type": "string", "enum": ["Service"]},
        "metadata": _METADATA,
        "spec": {"type": "object", "required": ["ports"], "fields": {
            "selector": {"type": "map-string"},
            "type": {"type": "string",
                     "enum": ["ClusterIP", "NodePort", "LoadBalancer",
                              "ExternalName"]},
            "clusterIP": {"type": "string"},
            "ports": {"type": "array", "min_items": 1, "items": {
                "type": "object", "required": ["port"], "fields": {
                    "name": {"type": "string"},
                    "port": {"type": "int"},
                    "targetPort": {"type": "int-or-string"},
                    "protocol": {"type": "string",
                                 "enum": ["TCP", "UDP", "SCTP"]},
                }}},
        }},
    },
}

# autoscaling/v2 (GA since k8s 1.23; v2beta1 — the reference's version,
# cuda-test-hpa.yaml:1 — was removed in 1.26)
_HPA_METRIC = {
    "type": "object", "required": ["type"],
    "fields": {
        "type": {"type": "string",
                 "enum": ["Object", "Pods", "Resource", "External",
                          "ContainerResource"]},
        "object": {"type": "object",
                   "required": ["describedObject", "metric", "target"],
                   "fields": {
                       "describedObject": {"type": "object",
                                           "required": ["kind", "name"],
                                           "fields": {
                                               "apiVersion": {"type": "string"},
                                               "kind": {"type": "string"},
                                               "name": {"type": "string"},
                                           }},
                       "metric": {"type": "object", "required": ["name"],
                                  "fields": {"name": {"type": "string"},
                                             "selector": {"type": "any"}}},
                       "target": {"type": "object", "required": ["type"],
                                  "fields": {
                                      "type": {"type": "string",
                                               "enum": ["Value",
                                                        "AverageValue",
                                                        "Utilization"]},
                                      "value": {"type": "quantity"},
                                      "averageValue": {"type": "quantity"},
                                      "averageUtilization": {"type": "int"},
                                  }},
                   }},
        "pods": {"type": "any"},
        "resource": {"type": "any"},
        "external": {"type": "any"},
        "containerResource": {"type": "any"},
    },
}

_HPA_POLICY = {
    "type": "object", "required": ["type", "value", "periodSeconds"],
    "fields": {
        "type": {"type": "string", "enum": ["Pods", "Percent"]},
        "value": {"type": "int"},
        "periodSeconds": {"type": "int"},
    },
}

_HPA_RULES = {
    "type": "object",
    "fields": {
        "stabilizationWindowSeconds": {"type": "int"},
        "selectPolicy": {"type": "string",
                         "enum": ["Max", "Min", "Disabled"]},
        "policies": {"type": "array", "items": _HPA_POLICY},
    },
}

_HPA = {
    "type": "object", "required": ["apiVersion", "kind", "metadata", "spec"],
    "fields": {
        "apiVersion": {"type": "string", "enum": ["autoscaling/v2"]},
        "kind": {"type": "string", "enum": ["HorizontalPodAutoscaler"]},
        "metadata": _METADATA,
        "spec": {"type": "object",
                 "required": ["scaleTargetRef", "maxReplicas"],
                 "fields": {
                     "scaleTargetRef": {"type": "object",
                                        "required": ["kind", "name"],
                                        "fields": {
                                            "apiVersion": {"type": "string"},
                                            "kind": {"type": "string"},
                                            "name": {"type": "string"},
                                        }},
                     "minReplicas": {"type": "int"},
                     "maxReplicas": {"type": "int"},
                     "metrics": {"type": "array", "items": _HPA_METRIC},
                     "behavior": {"type": "object", "fields": {
                         "scaleUp": _HPA_RULES,
                         "scaleDown": _HPA_RULES,
                     }},
                 }},
    },
}

# prometheus-operator monitoring.coreos.com/v1 PrometheusRule CRD subset
_PROMRULE = {
    "type": "object", "required": ["apiVersion", "kind", "metadata", "spec"],
    "fields": {
        "apiVersion": {"type": "string", "enum": ["monitoring.coreos.com/v1"]},
        "kind": {"type": "string", "enum": ["PrometheusRule"]},
        "metadata": _METADATA,
        "spec": {"type": "object", "required": ["groups"], "fields": {
            "groups": {"type": "array", "min_items": 1, "items": {
                "type": "object", "required": ["name", "rules"], "fields": {
                    "name": {"type": "string"},
                    "interval": {"type": "string"},
                    "rules": {"type": "array", "min_items": 1, "items": {
                        "type": "object", "required": ["expr"], "fields": {
                            "record": {"type": "string"},
                            "alert": {"type": "string"},
                            "expr": {"type": "int-or-string"},
                            "for": {"type": "string"},
                            "labels": {"type": "map-string"},
                            "annotations": {"type": "map-string"},
                        }}},
                }}},
        }},
    },
}

_CONFIGMAP = {
    "type": "object", "required": ["apiVersion", "kind", "metadata"],
    "fields": {
        "apiVersion": {"type": "string", "enum": ["v1"]},
        "kind": {"type": "string", "enum": ["ConfigMap"]},
        "metadata": _METADATA,
        "data": {"type": "object", "open": True, "fields": {}},
        "binaryData": {"type": "object", "open": True, "fields": {}},
    },
}

SCHEMAS = {
    "DaemonSet": _DAEMONSET,
    "Deployment": _DEPLOYMENT,
    "Service": _SERVICE,
    "HorizontalPodAutoscaler": _HPA,
    "PrometheusRule": _PROMRULE,
    "ConfigMap": _CONFIGMAP,
}


# --- cross-field invariants the apiserver enforces at admission ------------

def _cross_checks(doc: dict, errors: List[SchemaError]) -> None:
    kind = doc.get("kind")
    if kind in ("DaemonSet", "Deployment"):
        sel = (doc.get("spec", {}).get("selector", {}) or {}).get(
            "matchLabels", {}) or {}
        tmpl = (doc.get("spec", {}).get("template", {}).get(
            "metadata", {}) or {}).get("labels", {}) or {}
        for k, v in sel.items():
            if tmpl.get(k) != v:
                errors.append(SchemaError(
                    "spec.selector",
                    f"selector {k}={v} not matched by template labels "
                    f"{tmpl} (apps/v1 rejects this)"))
        # probe ports must reference a declared containerPort (by number
        # or name) — kubelet would fail the probe forever otherwise
        for ci, c in enumerate(doc.get("spec", {}).get("template", {})
                               .get("spec", {}).get("containers", [])):
            declared = set()
            for p in c.get("ports", []):
                declared.add(p.get("containerPort"))
                if "name" in p:
                    declared.add(p["name"])
            for probe_name in ("readinessProbe", "livenessProbe"):
                port = (c.get(probe_name, {}).get("httpGet", {}) or {}).get(
                    "port")
                if port is not None and declared and port not in declared:
                    errors.append(SchemaError(
                        f"spec.template.spec.containers[{ci}].{probe_name}",
                        f"probe port {port!r} not among declared ports "
                        f"{sorted(declared, key=str)}"))
    if kind == "HorizontalPodAutoscaler":
        spec = doc.get("spec", {})
        mn = spec.get("minReplicas", 1)
        mx = spec.get("maxReplicas", 0)
        if isinstance(mn, int) and isinstance(mx, int) and mx < mn:
            errors.append(SchemaError(
                "spec.maxReplicas", f"maxReplicas {mx} < minReplicas {mn}"))
        for mi, m in enumerate(spec.get("metrics", [])):
            t = m.get("type")
            key = {"Object": "object", "Pods": "pods",
                   "Resource": "resource", "External": "external",
                   "ContainerResource": "containerResource"}.get(t)
            if key and key not in m:
                errors.append(SchemaError(
                    f"spec.metrics[{mi}]",
                    f"type {t} requires a {key!r} stanza"))


def validate_manifest(doc: dict) -> List[SchemaError]:
    """Validate one manifest document. Returns the error list (empty =
    valid). Unknown kinds are an error — every shipped manifest must be
    covered by a pinned schema."""
    errors: List[SchemaError] = []
    kind = doc.get("kind")
    schema = SCHEMAS.get(kind)
    if schema is None:
        return [SchemaError("kind", f"no pinned schema for kind {kind!r}")]
    _check(schema, doc, kind, errors)
    _cross_checks(doc, errors)
    return errors


def _main(argv):
    """CLI: python -m mi355x_gpu_hpa.k8s_schema deploy/*.yaml — the offline
    stand-in for kubectl's server-side validation. Exit 1 on any error."""
    import sys

    import yaml

    rc = 0
    for path in argv:
        try:
            docs = [d for d in yaml.safe_load_all(open(path)) if d]
        except Exception as e:  # noqa: BLE001
            print(f"{path}: unparseable YAML: {e}")
            rc = 1
            continue
        for doc in docs:
            kind = doc.get("kind", "?")
            if kind == "Kustomization" or "kind" not in doc:
                continue  # not a k8s object (kustomize / Helm values)
            errors = validate_manifest(doc)
            if errors:
                rc = 1
                for e in errors:
                    print(f"{path} [{kind}]: {e}")
            else:
                print(f"{path} [{kind}]: OK")
    return rc


if __name__ == "__main__":
    import sys

    sys.exit(_main(sys.argv[1:]))
