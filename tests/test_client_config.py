"""K8s client construction paths: in-cluster SA credentials and the
in-cluster -> kubeconfig fallback order (reference main.py:128-140)."""

import yaml

import k8s_cc_manager_amd.k8s.client as client_mod
from k8s_cc_manager_amd.k8s.client import K8sClient, load_client


def test_in_cluster_reads_sa_credentials(tmp_path, monkeypatch):
    sa = tmp_path / "serviceaccount"
    sa.mkdir()
    (sa / "token").write_text("sekrit-token\n")
    (sa / "ca.crt").write_text("---ca---")
    monkeypatch.setattr(client_mod, "SA_DIR", str(sa))
    monkeypatch.setenv("KUBERNETES_SERVICE_HOST", "10.0.0.1")
    monkeypatch.setenv("KUBERNETES_SERVICE_PORT", "6443")
    c = K8sClient.in_cluster()
    assert c.base_url == "https://10.0.0.1:6443"
    assert c._session.headers["Authorization"] == "Bearer sekrit-token"
    assert c._verify == str(sa / "ca.crt")


def test_in_cluster_requires_env(monkeypatch):
    monkeypatch.delenv("KUBERNETES_SERVICE_HOST", raising=False)
    try:
        K8sClient.in_cluster()
        assert False, "should have raised"
    except RuntimeError as e:
        assert "in-cluster" in str(e)


def test_load_client_falls_back_to_kubeconfig(tmp_path, monkeypatch, fake_cluster):
    cluster, url = fake_cluster
    cluster.add_node("n1", labels={"k": "v"})
    kc = {
        "apiVersion": "v1",
        "kind": "Config",
        "current-context": "t",
        "contexts": [{"name": "t", "context": {"cluster": "c", "user": "u"}}],
        "clusters": [{"name": "c", "cluster": {"server": url}}],
        "users": [{"name": "u", "user": {}}],
    }
    path = tmp_path / "kc"
    path.write_text(yaml.safe_dump(kc))
    monkeypatch.delenv("KUBERNETES_SERVICE_HOST", raising=False)
    c = load_client(str(path))
    assert c.get_node("n1")["metadata"]["labels"]["k"] == "v"
