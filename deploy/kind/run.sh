#!/usr/bin/env bash
# run.sh — executable form of the CPU-only integration harness (config 1).
#
# Drives the ENTIRE L2-L5 cluster path on a GPU-less kind cluster and
# FAILS (non-zero exit) if any seam breaks: stub exporter up -> Prometheus
# scrape -> recording rule evaluates -> prometheus-adapter serves
# cuda_test_gpu_avg on custom.metrics.k8s.io -> HPA scales the deployment
# when the stub's busy% steps over the threshold. These are the
# reference's manual probes (reference README.md:42-122) as assertions.
#
# Requirements: docker, kind, kubectl, helm >= 3, network access to the
# prometheus-community charts. Takes ~6-10 min (mostly image pulls).
#
# Usage: deploy/kind/run.sh [--keep]   (--keep leaves the cluster running)

set -euo pipefail
cd "$(dirname "$0")"

CLUSTER=gpu-hpa-test
KEEP=0
[[ "${1:-}" == "--keep" ]] && KEEP=1

log() { printf '\n== %s\n' "$*"; }
fail() { printf 'FAIL: %s\n' "$*" >&2; exit 1; }

retry() { # retry <seconds> <desc> <cmd...>
    local deadline=$((SECONDS + $1)) desc=$2
    shift 2
    until "$@" >/dev/null 2>&1; do
        ((SECONDS < deadline)) || fail "timeout waiting for: $desc"
        sleep 5
    done
}

cleanup() {
    [[ $KEEP == 1 ]] && { log "keeping cluster $CLUSTER"; return; }
    kind delete cluster --name "$CLUSTER" >/dev/null 2>&1 || true
}
trap cleanup EXIT

log "0. kind cluster"
kind get clusters 2>/dev/null | grep -qx "$CLUSTER" ||
    kind create cluster --name "$CLUSTER" --wait 120s
kubectl config use-context "kind-$CLUSTER" >/dev/null

log "1. stub exporter (DaemonSet stand-in, 1 fake GPU)"
kubectl apply -f stub-exporter.yaml
retry 180 "stub exporter ready" \
    kubectl wait --for=condition=ready pod -l app.kubernetes.io/name=stub-exporter --timeout=5s

log "   probe: /metrics serves dcgm_gpu_temp (reference README.md:42-47)"
kubectl port-forward svc/stub-exporter 19400:9400 >/dev/null 2>&1 &
PF1=$!
sleep 2
curl -fsS localhost:19400/metrics | grep -q dcgm_gpu_temp ||
    fail "stub exporter /metrics missing dcgm_gpu_temp"
kill $PF1 2>/dev/null || true

log "2. kube-prometheus-stack (scrape config + ksm label allowlist)"
helm repo add prometheus-community \
    https://prometheus-community.github.io/helm-charts >/dev/null 2>&1 || true
helm repo update >/dev/null
helm upgrade --install kube-prometheus-stack \
    prometheus-community/kube-prometheus-stack \
    --values ../kube-prometheus-stack-values.yaml --wait --timeout 10m

log "3. recording rule + CPU cuda-test deployment"
kubectl apply -f ../cuda-test-prometheusrule.yaml
kubectl apply -f cuda-test-cpu-deployment.yaml
retry 120 "cuda-test pods ready" \
    kubectl wait --for=condition=ready pod -l app=cuda-test --timeout=5s

log "   drive busy to 40% and wait for cuda_test_gpu_avg in Prometheus"
kubectl port-forward svc/stub-exporter 19400:9400 >/dev/null 2>&1 &
PF1=$!
sleep 2
curl -fsS -X POST --data "40" localhost:19400/busy >/dev/null
kubectl port-forward svc/kube-prometheus-stack-prometheus 19090:9090 \
    >/dev/null 2>&1 &
PF2=$!
sleep 2
rule_has_value() {
    curl -fsS 'localhost:19090/api/v1/query?query=cuda_test_gpu_avg' |
        grep -q '"value"'
}
retry 120 "recording rule emits cuda_test_gpu_avg (README.md:80-88)" \
    rule_has_value

log "4. prometheus-adapter"
helm upgrade --install prometheus-adapter \
    prometheus-community/prometheus-adapter \
    --set prometheus.url="http://kube-prometheus-stack-prometheus.default.svc" \
    --wait --timeout 5m

log "   probe: adapter registration (README.md:98-102)"
adapter_serves_metric() {
    kubectl get --raw /apis/custom.metrics.k8s.io/v1beta1 2>/dev/null |
        grep -q cuda_test_gpu_avg
}
retry 180 "custom.metrics.k8s.io serves cuda_test_gpu_avg" adapter_serves_metric

log "5. HPA"
kubectl apply -f ../cuda-test-hpa.yaml

log "6. closed loop: 40% busy over target 5 must scale 1 -> >1 replicas"
hpa_scaled_up() {
    local n
    n=$(kubectl get deploy cuda-test -o jsonpath='{.spec.replicas}')
    [[ "${n:-1}" -gt 1 ]]
}
retry 300 "HPA scale-up (README.md:112-122)" hpa_scaled_up
kubectl get hpa cuda-test

kill $PF1 $PF2 2>/dev/null || true
log "PASS: all reference probes green end-to-end on a real cluster"
