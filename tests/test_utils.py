"""plan_rebalance spec table + recovery validation + delay jitter.

The planner cases are a port of the reference's table-driven suite
(test/utils.test.js:13-286), which is the de-facto spec for dead-backend
replacement, the max cap, starvation control and issue regressions
(bug #30).
"""

import pytest

from cueball_amd import utils


def plan(spares, dead, target, maximum, singleton=False):
    return utils.plan_rebalance(spares, dead, target, maximum, singleton)


def test_simple_addition():
    p = plan({"b1": []}, {}, 4, 10)
    assert p["remove"] == []
    assert p["add"] == ["b1", "b1", "b1", "b1"]


def test_addition_over_2_options():
    p = plan({"b1": [], "b2": []}, {}, 5, 10)
    assert p["remove"] == []
    assert p["add"] == ["b1", "b1", "b1", "b2", "b2"]


def test_add_with_existing():
    p = plan({"b1": ["c1"], "b2": ["c2"]}, {}, 4, 10)
    assert p["remove"] == []
    assert p["add"] == ["b1", "b2"]


def test_add_none():
    p = plan({"b1": ["c1", "c3"], "b2": ["c2", "c4"]}, {}, 4, 10)
    assert p["remove"] == []
    assert p["add"] == []


def test_add_and_remove():
    p = plan({"b1": ["c1", "c2", "c3"], "b2": ["c4"]}, {}, 4, 10)
    assert len(p["remove"]) == 1
    assert p["remove"][0] in ["c1", "c2", "c3"]
    assert p["add"] == ["b2"]


def test_add_from_unbalanced():
    p = plan({"b1": ["c1", "c2", "c3"], "b2": ["c4"]}, {}, 6, 10)
    assert p["remove"] == []
    assert p["add"] == ["b2", "b2"]


def test_shrink():
    p = plan({"b1": ["c1", "c2", "c3"], "b2": ["c4", "c5", "c6"]}, {}, 4, 10)
    assert p["remove"] == ["c4", "c1"]
    assert p["add"] == []


def test_lots_of_nodes():
    p = plan({"b1": ["c1", "c2", "c3", "c4"], "b2": [], "b3": [], "b4": [],
              "b5": [], "b6": [], "b7": []}, {}, 5, 10)
    assert p["remove"] == ["c1", "c2", "c3"]
    assert p["add"] == ["b2", "b3", "b4", "b5"]


def test_more_nodes():
    p = plan({"b3": [], "b1": [], "b2": [], "b4": [],
              "b5": ["c1", "c2", "c3", "c4"], "b6": [], "b7": []}, {}, 6, 10)
    assert p["remove"] == ["c1", "c2", "c3"]
    assert p["add"] == ["b3", "b1", "b2", "b4", "b6"]


def test_excess_spread_out():
    p = plan({"b3": ["c1"], "b1": ["c2"], "b2": ["c3"], "b4": ["c4"],
              "b5": ["c5"], "b6": ["c6"], "b7": []}, {}, 3, 10)
    assert p["remove"] == ["c6", "c5", "c4"]
    assert p["add"] == []


def test_odd_number():
    p = plan({"b3": ["c1"], "b1": [], "b2": []}, {}, 4, 10)
    assert p["remove"] == []
    assert p["add"] == ["b3", "b1", "b2"]


def test_reordering():
    p = plan({"b2": [], "b1": ["c1"], "b3": ["c2"]}, {}, 2, 10)
    assert p["remove"] == ["c2"]
    assert p["add"] == ["b2"]


def test_dead_replacement():
    p = plan({"b1": [], "b2": [], "b3": []}, {"b1": True}, 2, 10)
    assert p["remove"] == []
    assert p["add"] == ["b1", "b2", "b3"]


def test_dead_replacement_and_shrink():
    p = plan({"b1": ["c1", "c3"], "b2": ["c2"], "b3": []}, {"b1": True},
             3, 10)
    assert p["remove"] == ["c1"]
    assert p["add"] == ["b2", "b3"]


def test_dead_again():
    p = plan({"b1": ["c1"], "b2": ["c2"]}, {"b1": True}, 1, 2)
    assert p["remove"] == []
    assert p["add"] == []


def test_nested_dead():
    p = plan({"b1": [], "b2": ["c2"], "b3": [], "b4": []},
             {"b1": True, "b3": True}, 2, 10)
    assert p["remove"] == []
    assert p["add"] == ["b1", "b3", "b4"]


def test_nested_dead_with_cap():
    p = plan({"b1": [], "b2": ["c2"], "b3": [], "b4": []},
             {"b1": True, "b3": True}, 2, 3)
    assert p["remove"] == []
    assert p["add"] == ["b1", "b4"]


def test_dead_backend_starvation_single():
    p = plan({"b1": ["c1"]}, {"b1": True}, 2, 10)
    assert p["remove"] == []
    assert p["add"] == []


def test_dead_backend_starvation_pair():
    p = plan({"b1": ["c1"], "b2": []}, {"b1": True}, 3, 10)
    assert p["remove"] == []
    assert p["add"] == ["b2", "b2", "b2"]


def test_bug_30_all_dead_cap():
    spares = {
        "16uN6JsJFild9cHyl2+LSyRHmNc=": ["c1"],
        "c7QG0UOYCpm6m/hYUX0jBenbM70=": ["c2"],
        "ashWtupYHh1QH33UP/T2+6hvi8c=": [],
        "4QMg6SChOmtF8s6lfK32lLoKUFs=": [],
    }
    dead = {
        "c7QG0UOYCpm6m/hYUX0jBenbM70=": True,
        "16uN6JsJFild9cHyl2+LSyRHmNc=": True,
        "4QMg6SChOmtF8s6lfK32lLoKUFs=": True,
        "ashWtupYHh1QH33UP/T2+6hvi8c=": True,
    }
    p = plan(spares, dead, 3, 4)
    assert p["remove"] == []
    assert p["add"] == ["ashWtupYHh1QH33UP/T2+6hvi8c=",
                        "4QMg6SChOmtF8s6lfK32lLoKUFs="]


def test_singleton_mode():
    # Sets: at most one connection per distinct backend
    p = plan({"b1": [], "b2": [], "b3": []}, {}, 5, 10, singleton=True)
    assert p["add"] == ["b1", "b2", "b3"]


def test_singleton_dead_replacement():
    p = plan({"b1": [], "b2": []}, {"b1": True}, 2, 10, singleton=True)
    assert p["add"] == ["b1", "b2"]


# -- recovery validation (lib/utils.js:125-186) ---------------------------

def test_recovery_ok():
    utils.assert_recovery({"retries": 2, "timeout": 100, "delay": 50})


def test_recovery_unknown_key():
    with pytest.raises(ValueError):
        utils.assert_recovery({"retries": 2, "timeout": 100, "delay": 50,
                               "bogus": 1})


def test_recovery_requires_max_delay_for_many_retries():
    with pytest.raises(ValueError):
        utils.assert_recovery({"retries": 40, "timeout": 100, "delay": 50})
    utils.assert_recovery({"retries": 40, "timeout": 100, "delay": 50,
                           "maxDelay": 5000, "maxTimeout": 5000})


def test_recovery_exponential_overflow_guard():
    # delay * 2^retries >= 1 day requires maxDelay
    with pytest.raises(ValueError):
        utils.assert_recovery({"retries": 31, "timeout": 100, "delay": 1000,
                               "maxTimeout": 1000})


def test_recovery_max_less_than_base():
    with pytest.raises(ValueError):
        utils.assert_recovery({"retries": 2, "timeout": 100, "delay": 50,
                               "maxDelay": 10})


def test_recovery_set_requires_default():
    with pytest.raises(ValueError):
        utils.assert_recovery_set({})
    utils.assert_recovery_set({
        "default": {"retries": 1, "timeout": 100, "delay": 50}})


# -- delay jitter (lib/utils.js:446-461) ----------------------------------

def test_gen_delay_spread():
    for _ in range(200):
        d = utils.gen_delay(1000, 0.2)
        assert 900 <= d <= 1100


def test_gen_delay_from_recovery_dict():
    d = utils.gen_delay({"delay": 100, "delaySpread": 0.0})
    assert d == 100


def test_shuffle_preserves_elements():
    xs = list(range(50))
    out = utils.shuffle(list(xs))
    assert sorted(out) == xs
