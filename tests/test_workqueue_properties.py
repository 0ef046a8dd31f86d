"""Property-based workqueue semantics: under arbitrary add/get/done
interleavings no key is ever handed to two workers at once, nothing added
is lost, and backoff delays stay clamped."""

import hypothesis.strategies as st
from hypothesis import settings
from hypothesis.stateful import RuleBasedStateMachine, invariant, rule

from cro_amd.runtime.workqueue import RateLimitedQueue

KEYS = ["a", "b", "c"]


class QueueMachine(RuleBasedStateMachine):
    def __init__(self):
        super().__init__()
        self.q = RateLimitedQueue(base_delay=0.0001, max_delay=0.001)
        self.in_flight = set()
        self.ever_added = set()
        self.delivered = set()

    @rule(key=st.sampled_from(KEYS))
    def add(self, key):
        self.q.add(key)
        self.ever_added.add(key)

    @rule(key=st.sampled_from(KEYS))
    def add_after(self, key):
        self.q.add_after(key, 0.0002)
        self.ever_added.add(key)

    @rule(key=st.sampled_from(KEYS))
    def add_rate_limited(self, key):
        self.q.add_rate_limited(key)
        self.ever_added.add(key)

    @rule()
    def get(self):
        key = self.q.get(timeout=0.01)
        if key is not None:
            assert key not in self.in_flight, "key delivered concurrently"
            self.in_flight.add(key)
            self.delivered.add(key)

    @rule(key=st.sampled_from(KEYS))
    def done(self, key):
        if key in self.in_flight:
            self.q.done(key)
            self.in_flight.discard(key)
            self.q.forget(key)

    @invariant()
    def failures_bounded_math(self):
        for key in KEYS:
            # the clamped exponent must never overflow float math
            self.q.add_rate_limited(key)
            self.ever_added.add(key)

    def teardown(self):
        # drain: everything ever added must eventually be deliverable
        for key in list(self.in_flight):
            self.q.done(key)
        import time

        deadline = time.monotonic() + 2
        while self.delivered < self.ever_added and time.monotonic() < deadline:
            key = self.q.get(timeout=0.05)
            if key is not None:
                self.delivered.add(key)
                self.q.done(key)
        assert self.delivered >= self.ever_added, (
            self.ever_added - self.delivered
        )


QueueMachine.TestCase.settings = settings(
    max_examples=25, stateful_step_count=30, deadline=None
)
TestQueueProperties = QueueMachine.TestCase
