"""Unit tests for the eviction label algebra — the highest-value unit
target per SURVEY.md §4 (it encodes all the subtle state;
reference semantics: gpu_operator_eviction.py:43-95)."""

import pytest

from k8s_cc_manager_amd.k8s.eviction import (
    PAUSED_VALUE,
    pause_value,
    unpause_value,
)
from k8s_cc_manager_amd.labels import ready_value_for_state


@pytest.mark.parametrize(
    "value,expected",
    [
        (None, ""),
        ("", ""),
        ("false", "false"),
        ("true", PAUSED_VALUE),
        (PAUSED_VALUE, PAUSED_VALUE),  # idempotent
        ("custom_" + PAUSED_VALUE, "custom_" + PAUSED_VALUE),
        ("custom", "custom_" + PAUSED_VALUE),
    ],
)
def test_pause_value(value, expected):
    assert pause_value(value) == expected


@pytest.mark.parametrize(
    "value,expected",
    [
        (None, ""),
        ("", ""),
        ("false", "false"),
        (PAUSED_VALUE, "true"),
        ("custom_" + PAUSED_VALUE, "custom"),
        ("true", "true"),
        ("custom", "custom"),
    ],
)
def test_unpause_value(value, expected):
    assert unpause_value(value) == expected


@pytest.mark.parametrize(
    "value", [None, "", "false", "true", "custom", "a_b_c"]
)
def test_roundtrip(value):
    """unpause(pause(x)) restores the user-visible meaning of x."""
    paused = pause_value(value)
    restored = unpause_value(paused)
    if value in (None, ""):
        assert restored == ""
    elif value == "false":
        assert restored == "false"
    else:
        assert restored == value

    # pausing twice never stacks suffixes
    assert pause_value(paused) == paused


def test_crash_recoverability():
    """The original value is derivable from the paused label alone
    (crash mid-eviction must not lose custom values)."""
    for original in ("true", "custom", "a_b"):
        assert unpause_value(pause_value(original)) == original


@pytest.mark.parametrize(
    "state,ready",
    [
        ("on", "true"),
        ("ppcie", "true"),
        ("off", "false"),
        ("devtools", ""),
        ("failed", ""),
        ("", ""),
    ],
)
def test_ready_derivation(state, ready):
    assert ready_value_for_state(state) == ready
