"""Property-based apiserver-semantics tests (hypothesis stateful): arbitrary
operation interleavings must preserve the invariants every reconciler
relies on — monotonic resourceVersions, spec/status isolation, finalizer
lifecycle, and watch-stream completeness."""

import hypothesis.strategies as st
from hypothesis import settings
from hypothesis.stateful import Bundle, RuleBasedStateMachine, invariant, rule

from cro_amd.api.v1alpha1.types import ComposableResource, ComposableResourceSpec
from cro_amd.runtime.errors import AlreadyExistsError, ConflictError, NotFoundError
from cro_amd.runtime.store import InMemoryStore

NAMES = [f"obj-{i}" for i in range(4)]


def make_obj(name, model="m0"):
    r = ComposableResource(
        spec=ComposableResourceSpec(type="gpu", model=model, target_node="n0")
    )
    r.metadata.name = name
    return r


class StoreMachine(RuleBasedStateMachine):
    def __init__(self):
        super().__init__()
        self.store = InMemoryStore()
        self.events = self.store.watch(["ComposableResource"])
        self.max_rv_seen = 0
        # our model of what should exist: name -> has_finalizer, deleting
        self.model = {}

    def _rv(self, obj):
        rv = int(obj.metadata.resourceVersion)
        assert rv > 0
        return rv

    @rule(name=st.sampled_from(NAMES), finalizer=st.booleans())
    def create(self, name, finalizer):
        obj = make_obj(name)
        if finalizer:
            obj.metadata.finalizers = ["t/f"]
        try:
            created = self.store.create(obj)
        except AlreadyExistsError:
            assert name in self.model
            return
        assert name not in self.model
        self.model[name] = {"finalizer": finalizer, "deleting": False}
        self._rv(created)

    @rule(name=st.sampled_from(NAMES), model=st.sampled_from(["m0", "m1"]))
    def update_spec(self, name, model):
        try:
            cur = self.store.get("ComposableResource", name)
        except NotFoundError:
            assert name not in self.model
            return
        old_status = cur.status.model_copy(deep=True)
        cur.spec.model = model
        cur.status.state = "Bogus"  # must be ignored by spec update
        try:
            updated = self.store.update(cur)
        except NotFoundError:
            return
        assert updated.status == old_status  # spec update never touches status
        assert updated.spec.model == model

    @rule(name=st.sampled_from(NAMES), state=st.sampled_from(["Attaching", "Online", ""]))
    def update_status(self, name, state):
        try:
            cur = self.store.get("ComposableResource", name)
        except NotFoundError:
            return
        old_model = cur.spec.model
        cur.status.state = state
        cur.spec.model = "sneaky"  # must be ignored by status update
        try:
            updated = self.store.update_status(cur)
        except NotFoundError:
            return
        assert updated.spec.model == old_model
        assert updated.status.state == state

    @rule(name=st.sampled_from(NAMES))
    def stale_write_conflicts(self, name):
        try:
            a = self.store.get("ComposableResource", name)
        except NotFoundError:
            return
        b = a.model_copy(deep=True)
        a.status.state = "X"
        first = self.store.update_status(a)
        if first.metadata.resourceVersion != b.metadata.resourceVersion:
            b.status.state = "Y"
            try:
                self.store.update_status(b)
                raise AssertionError("stale write must conflict")
            except ConflictError:
                pass

    @rule(name=st.sampled_from(NAMES))
    def delete(self, name):
        try:
            self.store.delete("ComposableResource", name)
        except NotFoundError:
            assert name not in self.model
            return
        entry = self.model.get(name)
        if entry is None:
            return
        if entry["finalizer"]:
            entry["deleting"] = True
            got = self.store.get("ComposableResource", name)
            assert got.metadata.deletionTimestamp is not None
        else:
            self.model.pop(name, None)
            try:
                self.store.get("ComposableResource", name)
                raise AssertionError("object should be gone")
            except NotFoundError:
                pass

    @rule(name=st.sampled_from(NAMES))
    def clear_finalizer(self, name):
        try:
            cur = self.store.get("ComposableResource", name)
        except NotFoundError:
            return
        cur.metadata.finalizers = []
        try:
            self.store.update(cur)
        except (NotFoundError, ConflictError):
            return
        entry = self.model.get(name)
        if entry is not None:
            entry["finalizer"] = False
            if entry["deleting"]:
                self.model.pop(name, None)
                try:
                    self.store.get("ComposableResource", name)
                    raise AssertionError("cleared finalizer on deleting object must remove it")
                except NotFoundError:
                    pass

    @invariant()
    def rv_monotonic_in_watch(self):
        import queue

        while True:
            try:
                ev = self.events.get_nowait()
            except queue.Empty:
                return
            rv = int(ev.object.metadata.resourceVersion)
            if ev.type in ("ADDED", "MODIFIED"):
                assert rv > self.max_rv_seen, (ev.type, rv, self.max_rv_seen)
                self.max_rv_seen = rv

    @invariant()
    def model_matches_store(self):
        stored = {o.metadata.name for o in self.store.list("ComposableResource")}
        assert stored == set(self.model), (stored, set(self.model))

    @invariant()
    def event_log_replays_to_current_state(self):
        """Applying events_since(token) over any token must reconstruct
        the live object set — the correctness condition for resumable
        watches. Token 0 is valid while the log hasn't compacted."""
        events = self.store.events_since(0, ["ComposableResource"])
        if events is None:
            return  # compacted; resume would fall back to re-list
        alive = {}
        for ev in events:
            if ev.type == "DELETED":
                alive.pop(ev.object.metadata.name, None)
            else:
                alive[ev.object.metadata.name] = ev.object
        stored = {o.metadata.name for o in self.store.list("ComposableResource")}
        assert set(alive) == stored, (set(alive), stored)
        # seqs strictly increase
        seqs = [ev.seq for ev in events]
        assert seqs == sorted(seqs) and len(set(seqs)) == len(seqs)


StoreMachine.TestCase.settings = settings(max_examples=30, stateful_step_count=40, deadline=None)
TestStoreProperties = StoreMachine.TestCase
