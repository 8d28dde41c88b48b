"""k3samd — MI355X-native K3S GPU enablement stack.

A from-scratch, AMD-native (CDNA4 / gfx950) re-implementation of the
capabilities of the K3S-NVidia reference stack (/root/reference):

* ``native/`` — C++ kubelet device plugin (hand-rolled gRPC over HTTP/2),
  OCI prestart hook, KFD topology enumeration, node labeller, mi355x-smi.
* ``k3samd.ops`` — hand-written CDNA4 HIP kernels (STREAM HBM3E suite,
  MFMA matrix-core smoke) used by the in-pod GPU payloads and bench.py.
* ``k3samd.parallel`` — RCCL-over-xGMI collective smoke paths
  (torch.distributed; gloo on CPU for tests).
* ``k3samd.utils`` — profiling tooling (rocprofv3 rocpd database
  summarizer); topology truth lives in the C++ library (native/topology),
  exposed to Python via the binaries' ``--json`` outputs.
* ``deploy/`` — Helm chart + manifests with the same values surface as the
  reference (values.yaml:1-18).
"""

__version__ = "0.1.0"
