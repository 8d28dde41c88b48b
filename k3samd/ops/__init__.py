"""CDNA4 HIP kernel bindings.

The native extension (``k3samd/_C*.so``) is compiled in-tree for gfx950 by
``k3samd.build``. On a GPU machine a missing extension is a hard error —
there is deliberately no silent eager/PyTorch fallback for the compute path
(the kernels ARE the product, the MI355X analog of the reference's in-pod
``nvidia-smi`` payload, /root/reference/nvidia-smi.yaml:13).
"""

from __future__ import annotations

_C = None
_import_error: Exception | None = None

try:
    import torch  # noqa: F401  — the extension links against torch's libs
    from k3samd import _C  # type: ignore[attr-defined]
except ImportError as e:  # extension not built (CPU-only dev box is fine)
    _import_error = e


def _require_native():
    if _C is None:
        raise RuntimeError(
            "k3samd native extension is not built. Run "
            "`python -m k3samd.build` (or __graft_entry__.build()). "
            f"Original import error: {_import_error}"
        )
    return _C


def native_available() -> bool:
    return _C is not None


def stream_triad(a, b, c, s, nontemporal=False):
    """a = b + s*c over fp32 tensors, 16B/lane vectorized CDNA4 kernel."""
    _require_native().stream_triad(a, b, c, float(s), nontemporal)


def stream_copy(a, b, nontemporal=False):
    _require_native().stream_copy(a, b, nontemporal)


def stream_scale(a, c, s, nontemporal=False):
    _require_native().stream_scale(a, c, float(s), nontemporal)


def stream_add(a, b, c, nontemporal=False):
    _require_native().stream_add(a, b, c, nontemporal)


def mfma_throughput(out, iters, shape=16):
    """Issue `iters` MFMA groups per wave (shape 16 = v_mfma_f32_16x16x32,
    shape 32 = v_mfma_f32_32x32x16); returns FLOPs issued."""
    return _require_native().mfma_throughput(out, int(iters), int(shape))


# Validated on MI355X silicon (first gpurun, 2026-09-13): layout 0 — lane l
# holds A[l&15][(l>>4)*8 + j] (k-contiguous 8-element groups) — matches the
# hardware fragment mapping of v_mfma_f32_16x16x32_bf16 (max err 1.9e-6 vs
# torch fp32 matmul); layout 1 does not.
MFMA_LAYOUT = 0


def mfma_gemm16(A, B, layout=MFMA_LAYOUT):
    """Single-tile D[16,16] = A[16,32] @ B[32,16] via one bf16 MFMA."""
    return _require_native().mfma_gemm16(A, B, int(layout))


def mx_gemm16(A, B, fmt=0):
    """Single-tile D[16,16] = dequant(A[16,128]) @ dequant(B[128,16]) via
    one block-scaled v_mfma_scale_f32_16x16x128_f8f6f4 with unit scales.

    fmt 0 = fp8 e4m3 (A uint8 [16,128]; B uint8 [16,128] packed
    column-major: row c = column c of B). fmt 4 = fp4 e2m1 (packed two
    elements per byte, low nibble = even k: A [16,64], B [16,64]).
    """
    return _require_native().mx_gemm16(A, B, int(fmt))
