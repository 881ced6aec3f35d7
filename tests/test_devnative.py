"""Native PCI/KFD library against a synthetic sysfs tree (CPU) and the
real one (GPU-marked tests in test_gpu_device.py)."""

import struct

import pytest

from k8s_cc_manager_amd.device import native

pytestmark = pytest.mark.skipif(
    not native.available(), reason="_devnative not built"
)


@pytest.fixture
def fake_sysfs(tmp_path):
    """Two AMD accelerators, one NVIDIA GPU (must be ignored), one AMD
    NIC-class device (ignored)."""
    root = tmp_path / "pci"
    devs = {
        "0000:0a:00.0": ("0x1002", "0x75a3", "0x038000"),  # MI355X-like
        "0000:1b:00.0": ("0x1002", "0x75a3", "0x030200"),  # VGA-class AMD
        "0000:2c:00.0": ("0x10de", "0x2330", "0x030200"),  # NVIDIA: ignore
        "0000:3d:00.0": ("0x1002", "0x1479", "0x020000"),  # AMD NIC: ignore
    }
    for bdf, (vendor, device, cls) in devs.items():
        d = root / bdf
        d.mkdir(parents=True)
        (d / "vendor").write_text(vendor + "\n")
        (d / "device").write_text(device + "\n")
        (d / "class").write_text(cls + "\n")
        (d / "numa_node").write_text("1\n")
        (d / "reset").write_text("")
        # 64-byte config header: little-endian vendor, device
        cfg = struct.pack("<HH", int(vendor, 16), int(device, 16)) + b"\x00" * 60
        (d / "config").write_bytes(cfg)
    return root


def test_pci_scan_filters_amd_gpus(fake_sysfs):
    found = native.pci_scan(root=str(fake_sysfs))
    bdfs = sorted(e["bdf"] for e in found)
    assert bdfs == ["0000:0a:00.0", "0000:1b:00.0"]
    accel = next(e for e in found if e["bdf"] == "0000:0a:00.0")
    assert accel["vendor"] == 0x1002
    assert accel["class"] == 0x038000
    assert accel["numa_node"] == 1
    assert accel["has_reset"] is True


def test_pci_config_read(fake_sysfs):
    raw = native.pci_config_read("0000:0a:00.0", 0, 4, root=str(fake_sysfs))
    vendor, device = struct.unpack("<HH", raw)
    assert vendor == 0x1002
    assert device == 0x75A3


def test_pci_reset_writes_node(fake_sysfs):
    native.pci_reset("0000:0a:00.0", root=str(fake_sysfs))
    assert (fake_sysfs / "0000:0a:00.0" / "reset").read_text() == "1"


def test_kfd_topology_parse(tmp_path):
    nodes = tmp_path / "nodes"
    # node 0: CPU (simd_count 0); node 1: GPU
    (nodes / "0").mkdir(parents=True)
    (nodes / "0" / "properties").write_text(
        "cpu_cores_count 128\nsimd_count 0\nio_links_count 1\n"
    )
    (nodes / "1").mkdir(parents=True)
    (nodes / "1" / "properties").write_text(
        "cpu_cores_count 0\nsimd_count 1024\nsimd_per_cu 4\n"
        "gfx_target_version 90500\nio_links_count 8\n"
        "vendor_id 4098\ndevice_id 30115\n"
    )
    topo = native.kfd_topology(root=str(nodes))
    by_node = {e["node"]: e for e in topo}
    assert by_node[0]["is_gpu"] is False
    assert by_node[1]["is_gpu"] is True
    assert by_node[1]["cu_count"] == 256
    assert by_node[1]["gfx_target_version"] == 90500
    assert by_node[1]["io_links_count"] == 8


def test_kfd_version_missing_dev(tmp_path):
    assert native.kfd_version(dev_path=str(tmp_path / "nokfd")) is None
