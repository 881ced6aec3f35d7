from pathlib import Path

from k8s_cc_manager_amd.core.hostprobe import SNP_PARAM, TDX_PARAM, is_host_cc_enabled


def _mk(root: Path, rel: str, content: str):
    p = root / rel.lstrip("/")
    p.parent.mkdir(parents=True, exist_ok=True)
    p.write_text(content)


def test_no_params_means_no_cc(tmp_path):
    assert not is_host_cc_enabled(root=str(tmp_path))


def test_snp_enabled(tmp_path):
    _mk(tmp_path, SNP_PARAM, "Y\n")
    assert is_host_cc_enabled(root=str(tmp_path))


def test_snp_disabled(tmp_path):
    _mk(tmp_path, SNP_PARAM, "N\n")
    assert not is_host_cc_enabled(root=str(tmp_path))


def test_tdx_enabled(tmp_path):
    _mk(tmp_path, TDX_PARAM, "1\n")
    assert is_host_cc_enabled(root=str(tmp_path))
