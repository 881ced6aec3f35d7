"""GPU-box device-layer tests: real amdsmi enumeration, PCI scan, KFD."""

import pytest

torch = pytest.importorskip("torch")

pytestmark = [
    pytest.mark.gpu,
    pytest.mark.skipif(not torch.cuda.is_available(), reason="needs an MI355X"),
]


def test_native_pci_scan_finds_amd_gpu():
    from k8s_cc_manager_amd.device import native

    found = native.pci_scan()
    gpus = [e for e in found if e["driver"] == "amdgpu"]
    assert gpus, f"no amdgpu devices in PCI scan: {found}"


def test_kfd_alive():
    from k8s_cc_manager_amd.device import native

    ver = native.kfd_version()
    assert ver is not None and ver[0] >= 1, f"KFD not answering: {ver}"


def test_kfd_topology_has_gfx950():
    from k8s_cc_manager_amd.device import native

    topo = native.kfd_topology()
    gpus = [e for e in topo if e.get("is_gpu")]
    assert gpus
    # gfx_target_version 90500 == gfx950
    assert any(e.get("gfx_target_version", 0) // 100 == 905 for e in gpus), topo


def test_amdsmi_backend_enumeration(tmp_path):
    from k8s_cc_manager_amd.device.amdsmi_backend import AmdSmiBackend

    be = AmdSmiBackend(state_dir=str(tmp_path))
    devices, count = be.find_devices()
    assert count >= 1
    dev = devices[0]
    assert dev.is_gpu()
    assert dev.cc_query_supported
    assert ":" in dev.bdf


def test_amdsmi_transition_shadow_register(tmp_path):
    """Full staged transition on the real backend without FLR
    (CC_MANAGER_ALLOW_RESET unset): mode latches at reset(), boot-wait
    exercises amdsmi + the HIP liveness kernel."""
    from k8s_cc_manager_amd.core.transition import TransitionEngine
    from k8s_cc_manager_amd.device.amdsmi_backend import AmdSmiBackend

    be = AmdSmiBackend(state_dir=str(tmp_path))
    devices, _ = be.find_devices()
    engine = TransitionEngine()
    report = engine.apply_cc_mode(devices, devices, "devtools")
    assert report.ok, report.error
    assert all(d.query_cc_mode() == "devtools" for d in devices)
    # state persisted (crash-resume)
    be2 = AmdSmiBackend(state_dir=str(tmp_path))
    assert all(d.query_cc_mode() == "devtools" for d in be2.find_devices()[0])
