#!/bin/sh
# k3samd node installer — automates README §1 + deploy/containerd-runtime.md
# on a K3S GPU node: build/install binaries and register the `amd` runtime.
#
#   ./deploy/scripts/install-node.sh [--no-build] [--prefix /usr/local/bin]
set -eu

PREFIX=/usr/local/bin
BUILD=1
for arg in "$@"; do
  case "$arg" in
    --no-build) BUILD=0 ;;
    --prefix=*) PREFIX="${arg#--prefix=}" ;;
    *) echo "usage: $0 [--no-build] [--prefix=/usr/local/bin]"; exit 2 ;;
  esac
done

repo_root="$(cd "$(dirname "$0")/../.." && pwd)"

if [ "$BUILD" = 1 ]; then
  make -C "$repo_root/native" -j"$(nproc)"
fi

for bin in k3samd-oci-runtime k3samd-device-plugin k3samd-node-labeller \
           k3samd-cdi-gen mi355x-smi mi-stream mi-allreduce; do
  if [ -x "$repo_root/native/bin/$bin" ]; then
    install -m 0755 "$repo_root/native/bin/$bin" "$PREFIX/$bin"
    echo "installed $PREFIX/$bin"
  fi
done

# sanity: KFD present?
if [ ! -e /dev/kfd ]; then
  echo "WARNING: /dev/kfd not found — amdgpu driver not loaded on this node"
fi
"$PREFIX/mi355x-smi" || true

# containerd runtime registration for K3S (idempotent)
TMPL_DIR=/var/lib/rancher/k3s/agent/etc/containerd
TMPL="$TMPL_DIR/config.toml.tmpl"
if [ -d /var/lib/rancher/k3s ]; then
  mkdir -p "$TMPL_DIR"
  if [ -f "$TMPL" ] && grep -q 'runtimes.amd' "$TMPL"; then
    echo "containerd template already registers the amd runtime"
  else
    [ -f "$TMPL" ] || printf '{{ template "base" . }}\n' > "$TMPL"
    cat >> "$TMPL" <<EOF

[plugins."io.containerd.grpc.v1.cri".containerd.runtimes.amd]
  runtime_type = "io.containerd.runc.v2"
[plugins."io.containerd.grpc.v1.cri".containerd.runtimes.amd.options]
  BinaryName = "$PREFIX/k3samd-oci-runtime"
EOF
    echo "wrote $TMPL — restart k3s (systemctl restart k3s || k3s-agent)"
  fi
else
  echo "k3s not detected; see deploy/containerd-runtime.md for manual steps"
fi
