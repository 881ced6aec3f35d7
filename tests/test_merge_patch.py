"""Direct unit tests of the fake API server's strategic-merge-patch
semantics (everything else rides on these: label deletes via null,
nested merges, spec.unschedulable lifecycle)."""

from k8s_cc_manager_amd.k8s.client import K8sClient
from k8s_cc_manager_amd.k8s.fakecluster import _merge_patch


def test_merge_patch_null_deletes():
    target = {"metadata": {"labels": {"a": "1", "b": "2"}}}
    _merge_patch(target, {"metadata": {"labels": {"a": None, "c": "3"}}})
    assert target["metadata"]["labels"] == {"b": "2", "c": "3"}


def test_merge_patch_replaces_non_dict_with_dict():
    target = {"spec": "bogus"}
    _merge_patch(target, {"spec": {"unschedulable": True}})
    assert target["spec"] == {"unschedulable": True}


def test_merge_patch_scalar_overwrite_and_deep_merge():
    target = {"a": {"b": {"c": 1, "keep": "x"}}, "top": 1}
    _merge_patch(target, {"a": {"b": {"c": 2}}, "top": 2})
    assert target == {"a": {"b": {"c": 2, "keep": "x"}}, "top": 2}


def test_label_delete_roundtrip_over_http(fake_cluster):
    """Deleting a label with a null value through the real HTTP path."""
    cluster, url = fake_cluster
    cluster.add_node("n0", labels={"keep": "1", "drop": "2"})
    k8s = K8sClient(url)
    k8s.patch_node_labels("n0", {"drop": None, "new": "3"})
    labels = cluster.node_labels("n0")
    assert labels == {"keep": "1", "new": "3"}


def test_uncordon_removes_unschedulable_key(fake_cluster):
    """Cordon sets spec.unschedulable; uncordon must clear it the way
    kubectl does (the field disappears rather than reading false)."""
    cluster, url = fake_cluster
    cluster.add_node("n0")
    k8s = K8sClient(url)
    k8s.patch_node("n0", unschedulable=True)
    assert cluster.node_unschedulable("n0")
    k8s.patch_node("n0", unschedulable=False)
    assert not cluster.node_unschedulable("n0")
    node = cluster.get_node_copy("n0")
    assert "unschedulable" not in node["spec"]


def test_merge_patch_idempotent_property():
    """Applying the same patch twice is a no-op the second time."""
    import copy

    from hypothesis import given, settings
    from hypothesis import strategies as st

    scalars = st.one_of(st.none(), st.integers(-5, 5), st.text(max_size=4))
    patches = st.recursive(
        st.dictionaries(st.text(max_size=3), scalars, max_size=3),
        lambda children: st.dictionaries(
            st.text(max_size=3), st.one_of(scalars, children), max_size=3
        ),
        max_leaves=8,
    )

    @settings(max_examples=200, deadline=None)
    @given(patches, patches)
    def check(base, patch):
        target = copy.deepcopy(base)
        _merge_patch(target, patch)
        once = copy.deepcopy(target)
        _merge_patch(target, patch)
        assert target == once

    check()
