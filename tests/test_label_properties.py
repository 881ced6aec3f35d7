"""Property-based tests of the eviction label algebra (hypothesis).

The algebra encodes the subtlest durable state in the system
(SURVEY.md §4 calls it the highest-value unit target); these properties
hold for EVERY label value Kubernetes can store, not just the examples.
"""

from hypothesis import given, settings
from hypothesis import strategies as st

from k8s_cc_manager_amd.k8s.eviction import (
    PAUSED_VALUE,
    pause_value,
    unpause_value,
)

# Valid Kubernetes label values: empty, or [A-Za-z0-9]([A-Za-z0-9._-]*
# [A-Za-z0-9])? — must START and END alphanumeric (values like '_' are
# rejected by the API server, so the algebra need not round-trip them;
# hypothesis found that '_'-only values are lossy — in the reference's
# algebra too, via its trailing .strip('_')).
_alnum = "abcdefghijklmnopqrstuvwxyzABCDEFGHIJKLMNOPQRSTUVWXYZ0123456789"
_mid = _alnum + "._-"


@st.composite
def _k8s_label_values(draw):
    if draw(st.booleans()) and draw(st.integers(0, 4)) == 0:
        return ""
    first = draw(st.sampled_from(_alnum))
    middle = draw(st.text(alphabet=st.sampled_from(_mid), max_size=18))
    if middle:
        last = draw(st.sampled_from(_alnum))
        return first + middle + last
    return first


label_values = _k8s_label_values()


@settings(max_examples=300)
@given(label_values)
def test_pause_is_idempotent(value):
    once = pause_value(value)
    assert pause_value(once) == once


@settings(max_examples=300)
@given(label_values)
def test_unpause_inverts_pause_semantically(value):
    restored = unpause_value(pause_value(value))
    if value == "":
        assert restored == ""
    elif value == "false":
        assert restored == "false"
    elif value == "true":
        assert restored == "true"
    elif PAUSED_VALUE in value:
        # already-paused input: restore yields its unpaused form
        assert restored == unpause_value(value)
    else:
        assert restored == value


@settings(max_examples=300)
@given(label_values)
def test_paused_form_never_schedules(value):
    """Whatever the input, its paused form must be one the operator
    treats as not-deployed ('', 'false', or containing the marker)."""
    paused = pause_value(value)
    assert paused in ("", "false") or PAUSED_VALUE in paused


@settings(max_examples=300)
@given(label_values)
def test_unpause_never_yields_paused(value):
    """Restore never leaves the pause marker behind."""
    assert PAUSED_VALUE not in unpause_value(pause_value(value))


@settings(max_examples=300)
@given(label_values)
def test_crash_mid_cycle_recoverable(value):
    """pause -> crash -> pause -> unpause still restores the original
    meaning (a restarted manager re-pauses before restoring)."""
    twice = pause_value(pause_value(value))
    restored = unpause_value(twice)
    assert restored == unpause_value(pause_value(value))
