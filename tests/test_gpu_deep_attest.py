"""Deep attestation (rocprof counter readback) on a real MI355X."""

import pytest

torch = pytest.importorskip("torch")

pytestmark = [
    pytest.mark.gpu,
    pytest.mark.skipif(not torch.cuda.is_available(), reason="needs an MI355X"),
]


def test_deep_attest_counts_mfma_cycles():
    from k8s_cc_manager_amd.ops.deep_attest import (
        deep_attest_device,
        rocprof_available,
    )

    if not rocprof_available():
        pytest.skip("rocprofv3 not installed")
    cycles = deep_attest_device(0, gemm_dim=512, timeout=240)
    # two 512^3 GEMM dispatches -> ~1.3e8 MFMA-cycles of work spread
    # over the sampled SEs; anything clearly nonzero proves the matrix
    # pipes ran (counter granularity is per-SE samples)
    assert cycles > 1e4, f"implausibly few MFMA busy cycles: {cycles}"
