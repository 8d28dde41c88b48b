#!/bin/sh
# k3samd cluster validation — the scripted version of the reference's
# "Making sure it all works" walkthrough. Run from the repo root with a
# working kubeconfig after `helm install k3samd ...`.
set -eu

ns="${K3SAMD_NAMESPACE:-k3samd}"
fail=0
say() { printf '%s\n' "$*"; }

say "== control plane =="
kubectl get pods -n "$ns" -o wide
not_running=$(kubectl get pods -n "$ns" --no-headers 2>/dev/null |
  awk '$3 != "Running" {print $1}')
if [ -n "$not_running" ]; then
  say "FAIL: pods not Running: $not_running"; fail=1
else
  say "OK: all k3samd pods Running"
fi

say "== node labels =="
kubectl get nodes -l amd.com/gpu.present=true \
  -o custom-columns=NODE:.metadata.name,ARCH:.metadata.labels.amd\\.com/gpu\\.arch,COUNT:.metadata.labels.amd\\.com/gpu\\.count

say "== allocatable =="
alloc=$(kubectl get nodes \
  -o jsonpath='{range .items[*]}{.metadata.name}{" "}{.status.allocatable.amd\.com/gpu}{"\n"}{end}')
say "$alloc"
total=$(printf '%s\n' "$alloc" | awk '{s += $2} END {print s+0}')
if [ "$total" -lt 1 ]; then
  say "FAIL: no amd.com/gpu allocatable anywhere"; fail=1
else
  say "OK: $total amd.com/gpu allocatable cluster-wide"
fi

say "== smoke pod =="
kubectl delete pod mi-stream --ignore-not-found >/dev/null
kubectl apply -f deploy/manifests/mi-stream.yaml
if kubectl wait --for=jsonpath='{.status.phase}'=Succeeded pod/mi-stream \
     --timeout=180s; then
  kubectl logs mi-stream
  if kubectl logs mi-stream | grep -q '"payload": "mi-stream"'; then
    say "OK: STREAM smoke pod produced its report"
  else
    say "FAIL: smoke pod logs missing the JSON report"; fail=1
  fi
else
  say "FAIL: smoke pod did not succeed"; kubectl describe pod mi-stream; fail=1
fi
kubectl delete pod mi-stream --ignore-not-found >/dev/null

exit "$fail"
