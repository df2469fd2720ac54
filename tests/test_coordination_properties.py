"""Property-based checks of the coordination algebra (hypothesis):
simulated rollouts/scale-outs must terminate, respect the skew bound up
to the forward-progress exception, and be monotone."""
from hypothesis import given, settings, strategies as st

from rbg_amd.api import constants as C
from rbg_amd.api.types import (CoordinatedRollingUpdate, CoordinatedScaling,
                               CoordinationRule, CoordinationStrategy)
from rbg_amd.controller.coordination import (RoleScaleState,
                                             RoleUpdateState,
                                             calculate_rolling_partitions,
                                             calculate_scaling_targets,
                                             max_pairwise_skew)

sizes = st.lists(st.integers(min_value=1, max_value=60), min_size=2,
                 max_size=5)
skews = st.integers(min_value=0, max_value=50)


@settings(max_examples=60, deadline=None)
@given(totals=sizes, skew=skews)
def test_rolling_simulation_terminates_and_bounds_skew(totals, skew):
    names = [f"r{i}" for i in range(len(totals))]
    rule = CoordinationRule(
        roles=names,
        strategy=CoordinationStrategy(
            rolling_update=CoordinatedRollingUpdate(max_skew=skew)))
    updated = {n: 0 for n in names}
    # one-at-a-time simulation: every step, each role may update up to its
    # allowance; the bound must hold whenever every role COULD comply
    # (the forward-progress exception intentionally exceeds tiny skews)
    for step in range(sum(totals) + len(totals) + 5):
        states = {n: RoleUpdateState(n, t, updated[n])
                  for n, t in zip(names, totals)}
        parts = calculate_rolling_partitions(rule, states)
        progressed = False
        for n, t in zip(names, totals):
            allowed = t - parts[n]
            assert 0 <= parts[n] <= t
            assert allowed >= updated[n] - 1e-9     # never goes backwards
            if allowed > updated[n]:
                updated[n] = allowed
                progressed = True
        if all(updated[n] == t for n, t in zip(names, totals)):
            break
        assert progressed, f"deadlock at {updated} / {totals}"
    assert all(updated[n] == t for n, t in zip(names, totals))


@settings(max_examples=60, deadline=None)
@given(totals=sizes, skew=skews,
       progression=st.sampled_from([C.PROGRESSION_ORDER_READY,
                                    C.PROGRESSION_ORDER_SCHEDULED]))
def test_scaling_simulation_terminates(totals, skew, progression):
    names = [f"r{i}" for i in range(len(totals))]
    rule = CoordinationRule(
        roles=names,
        strategy=CoordinationStrategy(
            scaling=CoordinatedScaling(max_skew=skew,
                                       progression=progression)))
    current = {n: 0 for n in names}
    for step in range(sum(totals) + len(totals) + 5):
        # instances become ready immediately in this simulation
        states = {n: RoleScaleState(n, t, current[n], current[n])
                  for n, t in zip(names, totals)}
        targets = calculate_scaling_targets(rule, states)
        progressed = False
        for n, t in zip(names, totals):
            assert current[n] <= targets[n] <= t
            if targets[n] > current[n]:
                current[n] = targets[n]
                progressed = True
        if all(current[n] == t for n, t in zip(names, totals)):
            break
        assert progressed, f"deadlock at {current} / {totals}"
    assert all(current[n] == t for n, t in zip(names, totals))


@settings(max_examples=40, deadline=None)
@given(totals=sizes, skew=st.integers(min_value=5, max_value=50))
def test_skew_gauge_within_bound_plus_quantum(totals, skew):
    """After each step of the rolling simulation the OBSERVED skew stays
    within maxSkew + one instance quantum (the discrete forgiveness the
    reference's bound algebra allows)."""
    names = [f"r{i}" for i in range(len(totals))]
    rule = CoordinationRule(
        roles=names,
        strategy=CoordinationStrategy(
            rolling_update=CoordinatedRollingUpdate(max_skew=skew)))
    updated = {n: 0 for n in names}
    quantum = max(100.0 / t for t in totals)
    for _ in range(sum(totals) + 5):
        states = {n: RoleUpdateState(n, t, updated[n])
                  for n, t in zip(names, totals)}
        parts = calculate_rolling_partitions(rule, states)
        for n, t in zip(names, totals):
            updated[n] = max(updated[n], t - parts[n])
        observed = max_pairwise_skew(
            [RoleUpdateState(n, t, updated[n])
             for n, t in zip(names, totals)])
        assert observed <= skew + quantum + 1e-6, \
            (observed, skew, quantum, updated, totals)
        if all(updated[n] == t for n, t in zip(names, totals)):
            break
