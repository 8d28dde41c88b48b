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


def test_mx_gemm16_fp8_numerics():
    """Block-scaled MX fp8 MFMA vs a torch fp32 reference on the
    dequantized operands. Inputs are exactly representable in e4m3 (they
    ARE e4m3 casts), scales are 1.0, accumulation is fp32 — so the only
    tolerance needed is fp32 summation order over K=128."""
    torch.manual_seed(7)
    Af = torch.randn(16, 128)
    Bf = torch.randn(128, 16)
    A8 = Af.to(torch.float8_e4m3fn)
    B8 = Bf.to(torch.float8_e4m3fn)
    ref = A8.to(torch.float32) @ B8.to(torch.float32)
    A_bytes = A8.view(torch.uint8).cuda()
    # column-major pack: row c of the argument = column c of B
    B_bytes = B8.t().contiguous().view(torch.uint8).cuda()
    D = ops.mx_gemm16(A_bytes, B_bytes, fmt=0).cpu()
    err = (D - ref).abs().max().item()
    assert err < 5e-4, f"max err {err}\nD={D[0,:4]}\nref={ref[0,:4]}"


E2M1 = torch.tensor([0.0, 0.5, 1.0, 1.5, 2.0, 3.0, 4.0, 6.0])


def _fp4_random(shape, gen):
    idx = torch.randint(0, 8, shape, generator=gen)
    sign = torch.randint(0, 2, shape, generator=gen)
    vals = E2M1[idx] * (1 - 2 * sign.float())
    codes = (sign * 8 + idx).to(torch.uint8)
    return vals, codes


def _pack_nibbles_lo_even(codes):
    """[R, K] 4-bit codes -> [R, K//2] bytes, low nibble = even k."""
    lo = codes[:, 0::2]
    hi = codes[:, 1::2]
    return (lo | (hi << 4)).contiguous()


def test_mx_gemm16_fp4_numerics():
    """fp4-e2m1 through the same scaled MFMA: every product is exact in
    fp32 (e2m1 x e2m1 has <= 5 mantissa bits), so only summation-order
    error remains."""
    gen = torch.Generator().manual_seed(11)
    Av, Ac = _fp4_random((16, 128), gen)
    Bv, Bc = _fp4_random((128, 16), gen)
    ref = Av @ Bv
    A_bytes = _pack_nibbles_lo_even(Ac).cuda()
    B_bytes = _pack_nibbles_lo_even(Bc.t().contiguous()).cuda()
    D = ops.mx_gemm16(A_bytes, B_bytes, fmt=4).cpu()
    err = (D - ref).abs().max().item()
    assert err < 1e-3, f"max err {err}\nD={D[0,:4]}\nref={ref[0,:4]}"


def test_mx_gemm16_fp8_identity_rows():
    """Basis check that pins the fragment layout itself: A = 'identity'
    (A[i][k]=1 iff k==i), so D row i must equal B row i exactly."""
    Af = torch.zeros(16, 128)
    for i in range(16):
        Af[i, i] = 1.0
    A8 = Af.to(torch.float8_e4m3fn)
    Bf = torch.randn(128, 16)
    B8 = Bf.to(torch.float8_e4m3fn)
    ref = A8.to(torch.float32) @ B8.to(torch.float32)
    D = ops.mx_gemm16(A8.view(torch.uint8).cuda(),
                      B8.t().contiguous().view(torch.uint8).cuda(),
                      fmt=0).cpu()
    assert torch.equal(D, ref) or (D - ref).abs().max().item() < 1e-6, \
        (D[:3, :3], ref[:3, :3])
