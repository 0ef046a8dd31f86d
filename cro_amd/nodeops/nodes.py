"""Node-level helpers: existence, capacity admission, debounced restarts.

Parity with internal/utils/nodes.go:

* :func:`check_node_existed` / :func:`get_all_nodes` (nodes.go:119-144);
* :func:`check_node_capacity_sufficient` — admission of ``other_spec``
  CPU/memory/ephemeral/pods requirements against node capacity
  (nodes.go:78-117), sized in deployments for 8 × MI355X / 288 GB HBM3E
  per node;
* :class:`Debouncer` — the 10-second restart debounce the reference applies
  to daemonset rollouts and DRA-plugin kills (nodes.go:56-67,
  gpus.go:1140), generalized so any restart hook can be wrapped.
"""

from __future__ import annotations

import threading
import time
from typing import Callable, Dict, List

from ..api.v1alpha1.types import Node, NodeSpecRequirements
from ..runtime.client import Client
from ..runtime.errors import NotFoundError


def check_node_existed(client: Client, name: str) -> None:
    client.get(Node, name)  # raises NotFoundError


def node_exists(client: Client, name: str) -> bool:
    try:
        client.get(Node, name)
        return True
    except NotFoundError:
        return False


def get_all_nodes(client: Client) -> List[Node]:
    # read-only snapshot: allocation only inspects names/capacity
    return client.list(Node, copy=False)


def check_node_capacity_sufficient(
    client: Client, node_name: str, other_spec: NodeSpecRequirements
) -> bool:
    node = client.get(Node, node_name)
    alloc = node.status.capacity
    return (
        alloc.milli_cpu >= other_spec.milli_cpu
        and alloc.memory >= other_spec.memory
        and alloc.ephemeral_storage >= other_spec.ephemeral_storage
        and alloc.allowed_pod_number >= other_spec.allowed_pod_number
    )


RESTARTED_AT_ANNOTATION = "kubectl.kubernetes.io/restartedAt"
RESTART_DEBOUNCE_SECONDS = 10.0


def restart_daemonset(client: Client, namespace: str, name: str) -> bool:
    """Rolling restart via the ``restartedAt`` template annotation with the
    stability + 10-second debounce guards (nodes.go:35-76 parity): skipped
    when the daemonset has nothing scheduled or a rollout is already in
    flight (ready < desired, current < desired, unavailable, misscheduled),
    or when it restarted within the debounce window.  Returns True when a
    restart was issued.  Raises NotFoundError when the daemonset does not
    exist and ValueError on an unparseable restartedAt stamp (reference
    errors there too).
    """
    from ..api.v1alpha1.types import DaemonSet

    full_name = f"{namespace}/{name}"
    ds = client.get(DaemonSet, full_name)
    st = ds.status
    if st.desired_number_scheduled == 0:
        return False  # nothing scheduled — restart would be a no-op
    if (
        st.number_ready < st.desired_number_scheduled
        or st.current_number_scheduled < st.desired_number_scheduled
        or st.number_unavailable > 0
        or st.number_misscheduled > 0
    ):
        return False  # rollout in flight / unstable — do not stack restarts
    last = ds.spec.template_annotations.get(RESTARTED_AT_ANNOTATION, "")
    if last:
        try:
            last_ts = time.mktime(time.strptime(last, "%Y-%m-%dT%H:%M:%SZ"))
        except ValueError as exc:
            raise ValueError(
                f"failed to parse restartedAt annotation for DaemonSet "
                f"{full_name}: {exc}"
            )
        if time.mktime(time.gmtime()) - last_ts < RESTART_DEBOUNCE_SECONDS:
            return False  # restarted moments ago — debounce
    ds.spec.template_annotations[RESTARTED_AT_ANNOTATION] = time.strftime(
        "%Y-%m-%dT%H:%M:%SZ", time.gmtime()
    )
    client.update(ds)
    return True


class Debouncer:
    """Per-key debounce: a wrapped call is skipped when the same key fired
    within ``interval`` seconds (nodes.go:56-67 restartedAt guard)."""

    def __init__(self, interval: float = 10.0):
        self.interval = interval
        self._last: Dict[str, float] = {}
        self._lock = threading.Lock()

    def __call__(self, key: str, fn: Callable[[], None]) -> bool:
        """Run fn unless debounced; returns True if it ran."""
        now = time.monotonic()
        with self._lock:
            last = self._last.get(key, 0.0)
            if now - last < self.interval:
                return False
            self._last[key] = now
        fn()
        return True

    def wrap(self, key_fn: Callable[[str], str], fn: Callable[[str], None]):
        def wrapped(node: str) -> None:
            self(key_fn(node), lambda: fn(node))

        return wrapped
