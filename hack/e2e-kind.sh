#!/usr/bin/env bash
# Tier-3 end-to-end harness: kind cluster + real Argo Workflows controller +
# this controller, driving examples/inline-hello.yaml through a full
# submit→run→status cycle (the reference's README.md:54-160 flow, scripted).
#
# Requirements: kind, kubectl, docker. Usage:
#   hack/e2e-kind.sh            # create cluster, install, run one cycle, verify
#   KEEP=1 hack/e2e-kind.sh     # leave the cluster up afterwards
set -euo pipefail

cd "$(dirname "$0")/.."
CLUSTER=${CLUSTER:-active-monitor-e2e}
TIMEOUT=${TIMEOUT:-300}

cleanup() {
  if [[ "${KEEP:-0}" != "1" ]]; then
    kind delete cluster --name "$CLUSTER" >/dev/null 2>&1 || true
  fi
}
trap cleanup EXIT

echo "==> creating kind cluster ($CLUSTER)"
kind create cluster --name "$CLUSTER" --config hack/kind.cluster.yaml --wait 120s

echo "==> installing Argo Workflows (instanceID: activemonitor-workflows)"
kubectl apply -f deploy/deploy-argo.yaml
kubectl -n health rollout status deploy/workflow-controller --timeout=180s

echo "==> installing the HealthCheck CRD + controller RBAC"
kubectl apply -f config/crd/bases/activemonitor.keikoproj.io_healthchecks.yaml
kubectl apply -f config/crd/bases/argoproj.io_workflows.yaml 2>/dev/null || true

if [[ "${IN_CLUSTER:-0}" == "1" ]]; then
  echo "==> deploying the controller in-cluster"
  kubectl apply -f deploy/deploy-active-monitor.yaml
  kubectl -n health rollout status deploy/activemonitor-controller --timeout=180s
else
  echo "==> running the controller locally against the kind apiserver"
  python -m active_monitor_amd.cmd.main \
    --backend http \
    --max-workers 4 \
    --metrics-bind-address 0 --health-probe-bind-address 0 \
    >/tmp/am-e2e-controller.log 2>&1 &
  AM_PID=$!
  trap 'kill $AM_PID 2>/dev/null || true; cleanup' EXIT
fi

echo "==> applying examples/inline-hello.yaml"
kubectl apply -f examples/inline-hello.yaml

echo "==> waiting (${TIMEOUT}s) for a completed cycle (totalHealthCheckRuns >= 1)"
deadline=$((SECONDS + TIMEOUT))
while true; do
  runs=$(kubectl -n health get hc inline-hello \
    -o jsonpath='{.status.totalHealthCheckRuns}' 2>/dev/null || echo "")
  status=$(kubectl -n health get hc inline-hello \
    -o jsonpath='{.status.status}' 2>/dev/null || echo "")
  if [[ -n "$runs" && "$runs" -ge 1 ]]; then
    echo "==> cycle complete: status=$status runs=$runs"
    kubectl -n health get hc
    kubectl -n health get workflows.argoproj.io \
      -l workflows.argoproj.io/controller-instanceid=activemonitor-workflows
    break
  fi
  if (( SECONDS >= deadline )); then
    echo "!! timed out waiting for a completed cycle" >&2
    kubectl -n health get hc -o yaml || true
    kubectl -n health get workflows.argoproj.io -o wide || true
    [[ -f /tmp/am-e2e-controller.log ]] && tail -50 /tmp/am-e2e-controller.log
    exit 1
  fi
  sleep 3
done

echo "==> e2e PASSED"
