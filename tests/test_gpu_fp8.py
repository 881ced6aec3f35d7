"""fp8 (MX-scaled e4m3, unit scales) GEMM on a real MI355X."""

import pytest

torch = pytest.importorskip("torch")

pytestmark = [
    pytest.mark.gpu,
    pytest.mark.skipif(not torch.cuda.is_available(), reason="needs an MI355X"),
]


@pytest.fixture(scope="module")
def attest():
    from k8s_cc_manager_amd.ops import attest as a

    assert a.probe_available()
    return a


def _fp8(t):
    return t.to(torch.float8_e4m3fn)


def test_fp8_gemm_integer_exact(attest):
    """Small integers are exact in e4m3; fp32 accumulation on both
    sides -> bitwise agreement with the torch fp32 reference."""
    m = n = 256
    k = 512
    torch.manual_seed(7)
    a = _fp8(torch.randint(-2, 2, (m, k), device="cuda").float())
    bt = _fp8(torch.randint(-2, 2, (n, k), device="cuda").float())
    ref = a.float() @ bt.float().t()
    for trial in range(5):
        c = torch.full((m, n), float("nan"), device="cuda", dtype=torch.float32)
        attest.mfma_gemm_fp8(0, a.data_ptr(), bt.data_ptr(), c.data_ptr(), m, n, k)
        torch.cuda.synchronize()
        assert torch.equal(c, ref), f"trial {trial}: max diff {(c-ref).abs().max()}"


def test_fp8_deep_kernel_race_screen(attest):
    """The deep-pipelined fp8 kernel (M,N%256, K%256): bitwise integer
    race screen across repeated runs (sync-structure discipline)."""
    m = n = 512
    k = 768  # K%256 == 0 -> deep kernel; odd nk exercises tail drains
    a = _fp8(torch.randint(-2, 2, (m, k), device="cuda").float())
    bt = _fp8(torch.randint(-2, 2, (n, k), device="cuda").float())
    ref = a.float() @ bt.float().t()
    for trial in range(10):
        c = torch.empty(m, n, device="cuda", dtype=torch.float32)
        attest.mfma_gemm_fp8(0, a.data_ptr(), bt.data_ptr(), c.data_ptr(), m, n, k)
        torch.cuda.synchronize()
        assert torch.equal(c, ref), f"trial {trial}"


@pytest.mark.parametrize("which", [0, 2, 3, 4, 5, 6])
def test_fp8_variants_bitwise(attest, which):
    """Every fp8 structure variant (BK=128 / BK=256 / BK=64-hiocc /
    BK=128 3-blk / BK=128 single-buffered 4-blk / 256x128-tile) must
    agree bitwise with the fp32 reference on integer data."""
    m = n = 512
    k = 1024
    torch.manual_seed(which)
    a = _fp8(torch.randint(-2, 2, (m, k), device="cuda").float())
    bt = _fp8(torch.randint(-2, 2, (n, k), device="cuda").float())
    ref = a.float() @ bt.float().t()
    for trial in range(5):
        c = torch.empty(m, n, device="cuda", dtype=torch.float32)
        attest.mfma_gemm_fp8_variant(
            0, a.data_ptr(), bt.data_ptr(), c.data_ptr(), m, n, k, which
        )
        torch.cuda.synchronize()
        assert torch.equal(c, ref), f"which={which} trial={trial}"


def test_fp8_step3_kernel_still_used_for_small_shapes(attest):
    """K%256 != 0 routes to the 128-tile step-3 kernel."""
    m, n, k = 128, 128, 384
    a = _fp8(torch.randint(-2, 2, (m, k), device="cuda").float())
    bt = _fp8(torch.randint(-2, 2, (n, k), device="cuda").float())
    ref = a.float() @ bt.float().t()
    c = torch.empty(m, n, device="cuda", dtype=torch.float32)
    attest.mfma_gemm_fp8(0, a.data_ptr(), bt.data_ptr(), c.data_ptr(), m, n, k)
    torch.cuda.synchronize()
    assert torch.equal(c, ref)


@pytest.mark.parametrize("m,n,k", [(256, 512, 256), (512, 256, 1024)])
def test_fp8_gemm_random_matches_reference(attest, m, n, k):
    torch.manual_seed(m + n + k)
    a = _fp8(torch.randn(m, k, device="cuda"))
    bt = _fp8(torch.randn(n, k, device="cuda"))
    ref = a.float() @ bt.float().t()
    c = torch.empty(m, n, device="cuda", dtype=torch.float32)
    attest.mfma_gemm_fp8(0, a.data_ptr(), bt.data_ptr(), c.data_ptr(), m, n, k)
    torch.cuda.synchronize()
    err = (c - ref).abs().max().item()
    scale = ref.abs().max().item()
    assert err <= 1e-3 * max(scale, 1.0), f"err={err} scale={scale}"
