"""Probabilistic profiling tests (reference: main.go:541-548)."""

from parca_agent_amd.probabilistic import (
    ProbabilisticController,
    should_profile,
)


def test_threshold_extremes():
    assert should_profile("m1", 0, 100)
    assert not should_profile("m1", 0, 0)


def test_fraction_approximates_threshold():
    for threshold in (25, 50, 75):
        on = sum(should_profile("machine-x", i, threshold)
                 for i in range(2000))
        assert abs(on / 2000 - threshold / 100) < 0.05


def test_decision_stable_within_interval():
    assert should_profile("m", 7, 50) == should_profile("m", 7, 50)


def test_fleet_decorrelated():
    """Different machines make independent decisions per interval."""
    decisions = [should_profile(f"machine-{m}", 3, 50) for m in range(200)]
    on = sum(decisions)
    assert 60 < on < 140


class FakeService:
    def __init__(self):
        self.running = False
        self.transitions = []

    def start(self):
        self.running = True
        self.transitions.append("start")

    def stop(self):
        self.running = False
        self.transitions.append("stop")


def test_controller_switching():
    now = [0.0]
    svc = FakeService()
    c = ProbabilisticController(svc, threshold=50, interval_seconds=60,
                                machine_id="m", clock=lambda: now[0])
    # Find an interval where the decision flips so both paths run.
    c._apply()
    first = svc.running
    flipped = False
    for i in range(1, 50):
        now[0] = i * 60.0
        c._apply()
        if svc.running != first:
            flipped = True
            break
    assert flipped
    assert svc.transitions
