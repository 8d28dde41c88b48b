"""GPU numerics tests: CDNA4 kernels vs plain PyTorch fp32 references."""

import pytest
import torch

from k3samd import ops

pytestmark = pytest.mark.gpu

requires_gpu = pytest.mark.skipif(not torch.cuda.is_available(),
                                  reason="no GPU")


@requires_gpu
def test_native_loaded():
    # On a GPU box the native extension must be present — no silent fallback.
    assert ops.native_available(), "native extension not built on GPU box"


@requires_gpu
@pytest.mark.parametrize("nt", [False, True])
def test_triad_numerics(nt):
    dev = torch.device("cuda", 0)
    n = 1 << 20
    b = torch.rand(n, device=dev)
    c = torch.rand(n, device=dev)
    a = torch.empty_like(b)
    ops.stream_triad(a, b, c, 3.25, nontemporal=nt)
    torch.cuda.synchronize()
    ref = b + 3.25 * c
    assert torch.allclose(a, ref, rtol=1e-6, atol=1e-6)


@requires_gpu
@pytest.mark.parametrize("nt", [False, True])
def test_copy_scale_add_numerics(nt):
    dev = torch.device("cuda", 0)
    n = 1 << 18
    b = torch.rand(n, device=dev)
    c = torch.rand(n, device=dev)
    a = torch.empty_like(b)

    ops.stream_copy(a, b, nontemporal=nt)
    torch.cuda.synchronize()
    assert torch.equal(a, b)

    ops.stream_scale(a, c, 0.5, nontemporal=nt)
    torch.cuda.synchronize()
    assert torch.allclose(a, 0.5 * c, rtol=1e-6, atol=1e-6)

    ops.stream_add(a, b, c, nontemporal=nt)
    torch.cuda.synchronize()
    assert torch.allclose(a, b + c, rtol=1e-6, atol=1e-6)


@requires_gpu
def test_triad_tail_and_offsets():
    """Non-power-of-two sizes (still /4) hit the guard path correctly."""
    dev = torch.device("cuda", 0)
    for n in (4, 256, 1 << 16, (1 << 16) + 4, 3 * 7 * 64 * 4):
        b = torch.rand(n, device=dev)
        c = torch.rand(n, device=dev)
        a = torch.empty_like(b)
        ops.stream_triad(a, b, c, 1.5)
        torch.cuda.synchronize()
        assert torch.allclose(a, b + 1.5 * c, rtol=1e-6, atol=1e-6), n


@requires_gpu
def test_mfma_gemm16_vs_torch():
    """MFMA single tile vs torch fp32 matmul; asymmetric inputs on purpose."""
    dev = torch.device("cuda", 0)
    assert ops._require_native().has_mfma()
    torch.manual_seed(0)
    A = torch.randn(16, 32, device=dev, dtype=torch.bfloat16)
    B = torch.randn(32, 16, device=dev, dtype=torch.bfloat16)
    ref = A.float() @ B.float()
    errs = {}
    for layout in (0, 1):
        D = ops.mfma_gemm16(A, B, layout)
        torch.cuda.synchronize()
        errs[layout] = float((D - ref).abs().max())
    # exactly one layout should match the hardware mapping
    assert min(errs.values()) < 1e-3, f"no layout matched: {errs}"


@requires_gpu
@pytest.mark.parametrize("shape", [16, 32, 8, 4])
def test_mfma_throughput_runs(shape):
    dev = torch.device("cuda", 0)
    out = torch.zeros(2048, device=dev)
    flops = ops.mfma_throughput(out, 100, shape=shape)
    torch.cuda.synchronize()
    assert flops > 0
    assert torch.isfinite(out).all()
