"""Distributed deployment shape: an API-server process (uvicorn over the
in-memory store, admission server-side) driven by a SEPARATE operator
manager running all controllers through RemoteClient watch streams —
the cluster topology, exercised over real sockets."""

import socket
import threading
import time

import pytest

from cro_amd.api.v1alpha1.types import ComposabilityRequest, ComposableResource, Node
from cro_amd.controllers import build_manager
from cro_amd.fabric.adapter import Adapter
from cro_amd.fabric.mock import MockFabric
from cro_amd.nodeops.amdgpu import MockNodeOps
from cro_amd.runtime.errors import AdmissionDenied, NotFoundError
from cro_amd.runtime.manager import Manager
from cro_amd.runtime.remote import RemoteClient
from cro_amd.server.api import build_app
from tests.conftest import make_request


def free_port() -> int:
    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    port = s.getsockname()[1]
    s.close()
    return port


@pytest.fixture
def api_server():
    """uvicorn serving the store + admission — no controllers."""
    import uvicorn

    server_mgr = build_manager(Adapter("DRA", MockFabric()), None)  # store+admission only
    port = free_port()
    app = build_app(server_mgr.client)
    server = uvicorn.Server(
        uvicorn.Config(app, host="127.0.0.1", port=port, log_level="error")
    )
    thread = threading.Thread(target=server.run, daemon=True)
    thread.start()
    deadline = time.monotonic() + 15
    import httpx

    while time.monotonic() < deadline:
        try:
            if httpx.get(f"http://127.0.0.1:{port}/healthz", timeout=1).status_code == 200:
                break
        except Exception:
            time.sleep(0.05)
    else:
        pytest.fail("api server did not come up")
    yield f"http://127.0.0.1:{port}", server_mgr
    server.should_exit = True
    thread.join(timeout=5)


@pytest.fixture
def remote_operator(api_server):
    """A full operator manager whose only link to the world is HTTP."""
    url, server_mgr = api_server
    remote = RemoteClient(url)
    fabric = MockFabric(models={"mi355x": 8})
    mgr = build_manager(
        Adapter("DRA", fabric), None, client=remote, enable_webhook=False
    )
    ops = MockNodeOps(client=mgr.client)
    mgr.resource_reconciler.node_ops = ops

    orig_add = fabric.add_resource

    def add_resource(resource):
        did, cdi = orig_add(resource)
        ops.fabric_composed(resource.spec.target_node, did)
        return did, cdi

    fabric.add_resource = add_resource

    node = Node()
    node.metadata.name = "node0"
    remote.create(node)
    ops.set_driver("node0", True)
    mgr.start()
    yield mgr, remote, fabric, server_mgr
    mgr.stop()
    remote.close()


def wait_for(predicate, timeout=20.0):
    deadline = time.monotonic() + timeout
    while time.monotonic() < deadline:
        if predicate():
            return True
        time.sleep(0.02)
    return predicate()


def test_remote_crud_and_errors(api_server):
    url, _ = api_server
    remote = RemoteClient(url)
    try:
        created = remote.create(make_request("r1", target_node="node0"))
        assert created.metadata.uid
        assert remote.get(ComposabilityRequest, "r1").metadata.name == "r1"
        with pytest.raises(NotFoundError):
            remote.get(ComposabilityRequest, "ghost")
        # admission enforced server-side
        with pytest.raises(AdmissionDenied):
            remote.create(make_request("dup", target_node="node0"))
        created.spec.resource.size = 2
        updated = remote.update(created)
        assert updated.spec.resource.size == 2
        updated.status.state = "NodeAllocating"
        assert remote.update_status(updated).status.state == "NodeAllocating"
        remote.delete(ComposabilityRequest, "r1")
        assert remote.try_get(ComposabilityRequest, "r1") is None
    finally:
        remote.close()


def test_remote_operator_full_lifecycle(remote_operator):
    mgr, remote, fabric, server_mgr = remote_operator
    remote.create(make_request("r1", size=2, target_node="node0"))
    assert wait_for(
        lambda: (req := remote.try_get(ComposabilityRequest, "r1")) is not None
        and req.status.state == "Running"
    ), (remote.try_get(ComposabilityRequest, "r1") or object()).__dict__
    req = remote.get(ComposabilityRequest, "r1")
    assert len(req.status.resources) == 2
    assert all(v.state == "Online" for v in req.status.resources.values())
    assert len(fabric.attached_to("node0")) == 2

    remote.delete(ComposabilityRequest, "r1")
    assert wait_for(lambda: remote.try_get(ComposabilityRequest, "r1") is None)
    assert wait_for(lambda: fabric.attached_to("node0") == [])
    assert server_mgr.client.list(ComposableResource) == []


def test_watch_disconnect_releases_watcher(api_server):
    """Server-side watchers must be unsubscribed when the stream client
    disconnects — otherwise every event fans out to dead queues forever."""
    url, server_mgr = api_server
    before = len(server_mgr.store._watchers)
    r1 = RemoteClient(url)
    q = r1.watch(["ComposabilityRequest"])
    # wait for the stream to register server-side
    assert wait_for(lambda: len(server_mgr.store._watchers) > before, timeout=10)
    r1.close()
    assert wait_for(
        lambda: len(server_mgr.store._watchers) == before, timeout=10
    ), f"{len(server_mgr.store._watchers)} watchers still registered"


def test_api_server_outage_mid_lifecycle_recovers():
    """The API server dies mid-attach and comes back (same store): the
    remote operator's watch streams reconnect (re-list replay) and write
    retries back off — the lifecycle converges."""
    import uvicorn

    from cro_amd.fabric.mock import MockFabricConfig

    server_mgr = build_manager(Adapter("DRA", MockFabric()), None)
    port = free_port()
    app = build_app(server_mgr.client)

    def start_server():
        server = uvicorn.Server(
            uvicorn.Config(app, host="127.0.0.1", port=port, log_level="error")
        )
        thread = threading.Thread(target=server.run, daemon=True)
        thread.start()
        import httpx

        deadline = time.monotonic() + 15
        while time.monotonic() < deadline:
            try:
                if httpx.get(f"http://127.0.0.1:{port}/healthz", timeout=1).status_code == 200:
                    return server, thread
            except Exception:
                time.sleep(0.05)
        raise AssertionError("server did not come up")

    server, thread = start_server()

    remote = RemoteClient(f"http://127.0.0.1:{port}")
    # slow async fabric so the outage lands mid-attach
    fabric = MockFabric(
        models={"mi355x": 8},
        config=MockFabricConfig(asynchronous=True, attach_latency=1.0),
    )
    mgr = build_manager(Adapter("DRA", fabric), None, client=remote, enable_webhook=False)
    ops = MockNodeOps(client=mgr.client)
    mgr.resource_reconciler.node_ops = ops

    orig_add = fabric.add_resource

    def add_resource(resource):
        did, cdi = orig_add(resource)
        ops.fabric_composed(resource.spec.target_node, did)
        return did, cdi

    fabric.add_resource = add_resource

    node = Node()
    node.metadata.name = "node0"
    remote.create(node)
    ops.set_driver("node0", True)
    mgr.start()
    try:
        remote.create(make_request("r1", size=1, target_node="node0"))
        time.sleep(0.3)  # mid-attach (fabric still composing)

        server.should_exit = True
        thread.join(timeout=10)
        time.sleep(1.0)  # outage window: workers hit connection errors

        server, thread = start_server()  # same app → same store

        assert wait_for(
            lambda: (req := remote.try_get(ComposabilityRequest, "r1")) is not None
            and req.status.state == "Running",
            timeout=30,
        ), (remote.try_get(ComposabilityRequest, "r1") or object()).__dict__

        remote.delete(ComposabilityRequest, "r1")
        assert wait_for(lambda: remote.try_get(ComposabilityRequest, "r1") is None, timeout=20)
        assert fabric.attached_to("node0") == []
    finally:
        mgr.stop()
        remote.close()
        server.should_exit = True
        thread.join(timeout=5)


def test_croctl_watch_over_http(api_server, capsys):
    """croctl watch consumes the ndjson list+watch stream over a real
    socket (the ASGI test client cannot terminate an infinite stream, so
    this lives here with the uvicorn fixture)."""
    from cro_amd.cmd.croctl import main as croctl

    url, server_mgr = api_server
    server_mgr.client.create(make_request("w1", target_node="node0"))
    server_mgr.client.create(make_request("w2", model="mi300x", target_node="node0"))
    rc = croctl(["--server", url, "watch", "composabilityrequests", "--count", "2"])
    out = capsys.readouterr().out
    assert rc == 0
    lines = [ln for ln in out.splitlines() if ln.strip()]
    assert len(lines) == 2
    assert all(ln.startswith("ADDED") for ln in lines)
    assert {"w1", "w2"} <= {ln.split()[1] for ln in lines}


def test_remote_client_credentials(monkeypatch):
    """Bearer token + CA plumbing for real kube-apiserver use (the URL
    scheme already matches cluster-scoped CRD paths)."""
    from cro_amd.runtime.remote import RemoteClient

    c = RemoteClient("http://api", token="sa-token")
    assert c._http.headers["authorization"] == "Bearer sa-token"

    monkeypatch.setenv("CRO_API_TOKEN", "env-token")
    c2 = RemoteClient("http://api")
    assert c2._http.headers["authorization"] == "Bearer env-token"

    monkeypatch.delenv("CRO_API_TOKEN")
    c3 = RemoteClient("http://api")
    assert "authorization" not in c3._http.headers


def test_watch_resume_token(api_server):
    """kube-apiserver watch semantics: list → take ListMeta.resourceVersion
    → watch from it (only missed events replay); an aged-out token gets ONE
    ERROR event carrying a metav1.Status 410 Expired and the stream ENDS
    (the client re-lists — the real 410-Gone contract)."""
    import json

    import httpx

    url, server_mgr = api_server
    server_mgr.client.create(make_request("a", target_node="node0"))
    server_mgr.client.create(make_request("b", model="mi300x", target_node="node0"))

    base = f"{url}/apis/cro.hpsys.ibm.ie.com/v1alpha1/composabilityrequests"
    listed = httpx.get(base, timeout=10).json()
    assert sorted(i["metadata"]["name"] for i in listed["items"]) == ["a", "b"]
    rv = int(listed["metadata"]["resourceVersion"])
    assert rv > 0

    # miss one event while disconnected
    server_mgr.client.create(
        make_request("c", model="mi308x", target_node="node0")
    )
    with httpx.stream(
        "GET", base, params={"watch": "true", "resourceVersion": str(rv)}, timeout=10
    ) as resp:
        first = None
        for line in resp.iter_lines():
            if line.strip():
                first = json.loads(line)
                break
    assert first["type"] == "ADDED"
    assert first["object"]["metadata"]["name"] == "c"  # ONLY the missed event
    # the resume token IS the object resourceVersion — no framing field
    assert "rv" not in first
    assert int(first["object"]["metadata"]["resourceVersion"]) > rv

    # aged-out token → ONE 410 Expired Status, stream ends (no replay)
    store = server_mgr.store
    while store._event_log:
        store._event_log.popleft()
    with httpx.stream(
        "GET", base, params={"watch": "true", "resourceVersion": "1"}, timeout=10
    ) as resp:
        got = [json.loads(l) for l in resp.iter_lines() if l.strip()]
    assert len(got) == 1
    err = got[0]
    assert err["type"] == "ERROR"
    assert err["object"]["kind"] == "Status"
    assert err["object"]["code"] == 410
    assert err["object"]["reason"] == "Expired"


def test_watch_sendinitialevents(api_server):
    """The 1.27+ WatchList protocol: sendInitialEvents=true streams ADDED
    per current object then a BOOKMARK annotated k8s.io/initial-events-end
    at the snapshot rv."""
    import json

    import httpx

    url, server_mgr = api_server
    server_mgr.client.create(make_request("w1", target_node="node0"))
    server_mgr.client.create(make_request("w2", model="mi300x", target_node="node0"))
    base = f"{url}/apis/cro.hpsys.ibm.ie.com/v1alpha1/composabilityrequests"
    with httpx.stream(
        "GET", base,
        params={"watch": "true", "sendInitialEvents": "true",
                "allowWatchBookmarks": "true"},
        timeout=10,
    ) as resp:
        got = []
        for line in resp.iter_lines():
            if line.strip():
                got.append(json.loads(line))
            if len(got) == 3:
                break
    assert [g["type"] for g in got] == ["ADDED", "ADDED", "BOOKMARK"]
    bm = got[2]["object"]
    assert bm["metadata"]["annotations"] == {"k8s.io/initial-events-end": "true"}
    assert int(bm["metadata"]["resourceVersion"]) > 0


def test_store_events_since_semantics():
    from cro_amd.runtime.store import InMemoryStore
    from cro_amd.runtime.client import Client

    store = InMemoryStore()
    client = Client(store)
    assert store.events_since(0) == []  # nothing happened yet
    client.create(make_request("x", target_node="n"))
    evs = store.events_since(0)
    assert len(evs) == 1 and evs[0].object.metadata.name == "x"
    assert store.events_since(evs[0].seq) == []
    assert store.events_since(0, kinds=["ComposableResource"]) == []
    # compaction gap → None
    store._event_log.popleft()
    client.create(make_request("y", model="m2", target_node="n"))
    assert store.events_since(0) is None


def test_events_since_foreign_token_forces_relist():
    """A resume token larger than the store's own sequence (server
    restarted with fresh state) is not comparable — must expire, not
    silently return 'caught up'."""
    from cro_amd.runtime.client import Client
    from cro_amd.runtime.store import InMemoryStore

    store = InMemoryStore()
    Client(store).create(make_request("x", target_node="n"))
    assert store.events_since(999) is None
    assert store.events_since(store.current_seq()) == []


def test_api_server_state_loss_syncer_repairs(monkeypatch):
    """Catastrophic apiserver restart with a FRESH store (etcd loss): the
    operator's watch streams carry now-foreign resume tokens — the server
    answers Expired + empty re-list — and the fabric still holds the
    composed device with no CR anywhere. The operator's upstream syncer
    must repair the orphan through the new server (detach CR → physical
    detach).

    Short CRO_WATCH_TIMEOUT: in-process uvicorn shutdown can leave the old
    streams half-open (reading keepalives from the dead store); the k8s
    watch-timeout forces the reconnect that discovers the new server."""
    import uvicorn

    monkeypatch.setenv("CRO_WATCH_TIMEOUT", "1.5")
    port = free_port()

    def start_server(app):
        server = uvicorn.Server(
            uvicorn.Config(app, host="127.0.0.1", port=port, log_level="error")
        )
        thread = threading.Thread(target=server.run, daemon=True)
        thread.start()
        import httpx

        deadline = time.monotonic() + 15
        while time.monotonic() < deadline:
            try:
                if httpx.get(f"http://127.0.0.1:{port}/healthz", timeout=1).status_code == 200:
                    return server, thread
            except Exception:
                time.sleep(0.05)
        raise AssertionError("server did not come up")

    server_mgr = build_manager(Adapter("DRA", MockFabric()), None)
    server, thread = start_server(build_app(server_mgr.client))

    remote = RemoteClient(f"http://127.0.0.1:{port}")
    fabric = MockFabric(models={"mi355x": 8})
    mgr = build_manager(
        Adapter("DRA", fabric), None, client=remote, enable_webhook=False,
        syncer_period=0.3, syncer_grace=0.5,
    )
    ops = MockNodeOps(client=mgr.client)
    mgr.resource_reconciler.node_ops = ops
    mgr.syncer.node_ops = ops

    orig_add = fabric.add_resource

    def add_resource(resource):
        did, cdi = orig_add(resource)
        ops.fabric_composed(resource.spec.target_node, did)
        return did, cdi

    fabric.add_resource = add_resource

    node = Node()
    node.metadata.name = "node0"
    remote.create(node)
    ops.set_driver("node0", True)
    mgr.start()
    try:
        remote.create(make_request("r1", size=1, target_node="node0"))
        assert wait_for(
            lambda: (r := remote.try_get(ComposabilityRequest, "r1")) is not None
            and r.status.state == "Running",
            timeout=20,
        )
        assert len(fabric.attached_to("node0")) == 1

        # replace the server with a FRESH empty store on the same port
        server.should_exit = True
        thread.join(timeout=10)
        fresh_mgr = build_manager(Adapter("DRA", MockFabric()), None)
        server, thread = start_server(build_app(fresh_mgr.client))
        node2 = Node()
        node2.metadata.name = "node0"
        remote.create(node2)  # re-register (entrypoint would do this)

        # no CR exists anywhere, yet the fabric holds the device — the
        # syncer must detect the orphan and walk a full detach through
        # the new server
        assert wait_for(lambda: fabric.attached_to("node0") == [], timeout=30), (
            fabric.attached_to("node0"),
            [r.metadata.name for r in fresh_mgr.client.list(ComposableResource)],
        )
        assert wait_for(
            lambda: fresh_mgr.client.list(ComposableResource) == [], timeout=20
        )
    finally:
        mgr.stop()
        remote.close()
        server.should_exit = True
        thread.join(timeout=10)


def test_watch_timeout_recycling_under_churn(api_server, monkeypatch):
    """With a 1 s watch timeout, streams recycle many times while
    lifecycles churn; rv-token resumes must make the recycling invisible
    (no missed events, no stalls)."""
    monkeypatch.setenv("CRO_WATCH_TIMEOUT", "1")
    url, server_mgr = api_server
    remote = RemoteClient(url)
    fabric = MockFabric(models={"mi355x": 8})
    mgr = build_manager(Adapter("DRA", fabric), None, client=remote, enable_webhook=False)
    ops = MockNodeOps(client=mgr.client)
    mgr.resource_reconciler.node_ops = ops

    orig_add = fabric.add_resource

    def add(resource):
        did, cdi = orig_add(resource)
        ops.fabric_composed(resource.spec.target_node, did)
        return did, cdi

    fabric.add_resource = add
    node = Node()
    node.metadata.name = "node0"
    remote.create(node)
    ops.set_driver("node0", True)
    mgr.start()
    try:
        t0 = time.monotonic()
        cycles = 0
        while time.monotonic() - t0 < 6.0:  # spans ≥5 stream generations
            name = f"wt-{cycles}"
            remote.create(make_request(name, size=1, target_node="node0"))
            assert wait_for(
                lambda: (r := remote.try_get(ComposabilityRequest, name)) is not None
                and r.status.state == "Running",
                timeout=15,
            ), name
            remote.delete(ComposabilityRequest, name)
            assert wait_for(
                lambda: remote.try_get(ComposabilityRequest, name) is None, timeout=15
            ), name
            cycles += 1
        assert cycles >= 5  # made real progress across recycles
        assert fabric.attached_to("node0") == []
    finally:
        mgr.stop()
        remote.close()


def test_openshift_machine_chain_over_remote(api_server):
    """FTI's node→Machine→BMH annotation chain resolves through
    RemoteClient (the cluster shape where the FTI backend runs off-node
    and reads these objects over the API)."""
    from cro_amd.api.v1alpha1.types import BareMetalHost, Machine
    from cro_amd.fabric.fti.machines import resolve_machine_id_openshift

    url, _ = api_server
    remote = RemoteClient(url)
    try:
        node = Node()
        node.metadata.name = "worker-0"
        node.metadata.annotations["machine.openshift.io/machine"] = "ns/m-0"
        remote.create(node)
        m = Machine()
        m.metadata.name = "ns/m-0"
        m.metadata.annotations["metal3.io/BareMetalHost"] = "ns/bmh-0"
        remote.create(m)
        bmh = BareMetalHost()
        bmh.metadata.name = "ns/bmh-0"
        bmh.metadata.annotations["cluster-manager.cdi.io/machine"] = "uuid-42"
        remote.create(bmh)
        assert resolve_machine_id_openshift(remote, "worker-0") == "uuid-42"
    finally:
        remote.close()


def test_cached_client_informer_reads(api_server):
    """cache=True: get/list served from the watch-fed informer cache
    (client-go SharedInformer shape) — reads cost zero RTTs once synced,
    write responses land read-your-writes, deletes converge via the
    watch."""
    url, server_mgr = api_server
    server_mgr.client.create(make_request("c1", target_node="node0"))

    rc = RemoteClient(url, cache=True)
    try:
        # first read starts the informer (HTTP fallback until synced)
        items = rc.list(ComposabilityRequest)
        assert [i.metadata.name for i in items] == ["c1"]
        inf = rc._informers["ComposabilityRequest"]
        assert inf.synced.wait(10)
        # synced: reads now come from the cache
        assert [i.metadata.name for i in rc.list(ComposabilityRequest)] == ["c1"]

        # server-side create propagates into the cache via the watch
        server_mgr.client.create(make_request("c2", model="m2", target_node="node0"))
        deadline = time.monotonic() + 10
        while time.monotonic() < deadline:
            if rc.try_get(ComposabilityRequest, "c2") is not None:
                break
            time.sleep(0.02)
        assert rc.get(ComposabilityRequest, "c2").spec.resource.model == "m2"

        # read-your-writes: an update is visible immediately (no watch wait)
        cur = rc.get(ComposabilityRequest, "c1")
        cur.spec.resource.size = 3
        updated = rc.update(cur)
        assert rc.get(ComposabilityRequest, "c1").spec.resource.size == 3
        assert int(rc.get(ComposabilityRequest, "c1").metadata.resourceVersion) >= int(
            updated.metadata.resourceVersion)

        # cached reads are COPIES: mutating one does not poison the cache
        a = rc.get(ComposabilityRequest, "c1")
        a.spec.resource.size = 99
        assert rc.get(ComposabilityRequest, "c1").spec.resource.size == 3

        # delete converges through the watch stream
        server_mgr.client.delete(ComposabilityRequest, "c2")
        deadline = time.monotonic() + 10
        while time.monotonic() < deadline:
            if rc.try_get(ComposabilityRequest, "c2") is None:
                break
            time.sleep(0.02)
        assert rc.try_get(ComposabilityRequest, "c2") is None
    finally:
        rc.close()


def test_cached_client_runs_controllers(api_server):
    """A full operator over the CACHED client drives a request to Running
    — the controller-runtime topology (controllers read from cache)."""
    url, server_mgr = api_server
    remote = RemoteClient(url, cache=True)
    fabric = MockFabric(models={"mi355x": 8})
    mgr = build_manager(
        Adapter("DRA", fabric), None, client=remote, enable_webhook=False
    )
    ops = MockNodeOps(client=mgr.client)
    mgr.resource_reconciler.node_ops = ops

    orig_add = fabric.add_resource

    def add_resource(resource):
        did, cdi = orig_add(resource)
        ops.fabric_composed(resource.spec.target_node, did)
        return did, cdi

    fabric.add_resource = add_resource
    node = Node()
    node.metadata.name = "cachenode"
    node.status.capacity.milli_cpu = 64000
    node.status.capacity.memory = 1 << 40
    node.status.capacity.allowed_pod_number = 128
    server_mgr.client.create(node)
    mgr.start()
    try:
        server_mgr.client.create(make_request("cr1", size=2, target_node="cachenode"))
        assert mgr.wait_for(
            lambda: (r := server_mgr.client.try_get(ComposabilityRequest, "cr1")) is not None
            and r.status.state == "Running",
            timeout=30,
        ), (lambda r: r.status.state if r else "gone")(
            server_mgr.client.try_get(ComposabilityRequest, "cr1"))
        server_mgr.client.delete(ComposabilityRequest, "cr1")
        assert mgr.wait_for(
            lambda: server_mgr.client.try_get(ComposabilityRequest, "cr1") is None,
            timeout=30,
        )
    finally:
        mgr.stop()
        remote.close()


def test_cached_client_converges_under_random_ops(api_server):
    """Property: after ANY sequence of server-side create/update/delete
    (including rapid name reuse — the tombstone race surface), the
    informer cache converges to exactly the server's object set."""
    import random

    url, server_mgr = api_server
    rc = RemoteClient(url, cache=True)
    rng = random.Random(7)
    try:
        rc.list(ComposabilityRequest)  # start + sync the informer
        assert rc._informers["ComposabilityRequest"].synced.wait(10)

        names = [f"p{i}" for i in range(6)]
        live = {}
        for step in range(300):
            name = rng.choice(names)
            op = rng.random()
            if name not in live:
                server_mgr.client.create(
                    make_request(name, model=f"m-{name}", target_node="node0"))
                live[name] = 1
            elif op < 0.5:
                cur = server_mgr.client.get(ComposabilityRequest, name)
                cur.spec.resource.size = rng.randint(1, 8)
                server_mgr.client.update(cur)
            else:
                server_mgr.client.delete(ComposabilityRequest, name)
                del live[name]
            if rng.random() < 0.2:
                # interleave CLIENT-SIDE writes (the offer/tombstone path)
                cached = rc.try_get(ComposabilityRequest, name)
                if cached is not None:
                    cached.spec.resource.size = rng.randint(1, 8)
                    try:
                        rc.update(cached)
                    except Exception:
                        pass  # conflict/deleted under us — expected

        # convergence: cache == server (object set and resourceVersions)
        deadline = time.monotonic() + 10
        while time.monotonic() < deadline:
            server_objs = {
                o.metadata.name: o.metadata.resourceVersion
                for o in server_mgr.client.list(ComposabilityRequest)
            }
            cache_objs = {
                o.metadata.name: o.metadata.resourceVersion
                for o in rc.list(ComposabilityRequest)
            }
            if server_objs == cache_objs:
                break
            time.sleep(0.05)
        assert server_objs == cache_objs, (server_objs, cache_objs)
    finally:
        rc.close()


def test_cached_client_relist_prunes_deleted():
    """An object deleted while the informer is disconnected must be pruned
    by the reconnect re-list (client-go sync semantics) — otherwise the
    cache serves phantoms forever. Forced via apiserver replacement on the
    same port with a store missing one object (state loss → foreign resume
    rv → 410 Expired → re-list → prune + synthetic DELETED).  A short
    CRO_WATCH_TIMEOUT lets the first server's stream end so it can shut
    down (open streams otherwise pin uvicorn's graceful shutdown)."""
    import os as _os

    import uvicorn

    _os.environ["CRO_WATCH_TIMEOUT"] = "2"
    port = free_port()

    def start_server(mgr):
        server = uvicorn.Server(uvicorn.Config(
            build_app(mgr.client), host="127.0.0.1", port=port,
            log_level="error"))
        thread = threading.Thread(target=server.run, daemon=True)
        thread.start()
        import httpx

        deadline = time.monotonic() + 15
        while time.monotonic() < deadline:
            try:
                if httpx.get(f"http://127.0.0.1:{port}/healthz", timeout=1).status_code == 200:
                    return server, thread
            except Exception:
                time.sleep(0.05)
        raise AssertionError("server did not come up")

    mgr1 = build_manager(Adapter("DRA", MockFabric()), None)
    mgr1.client.create(make_request("keep", target_node="node0"))
    mgr1.client.create(make_request("vanish", model="m2", target_node="node0"))
    server1, thread1 = start_server(mgr1)

    rc = RemoteClient(f"http://127.0.0.1:{port}", cache=True)
    try:
        rc.list(ComposabilityRequest)
        inf = rc._informers["ComposabilityRequest"]
        assert inf.synced.wait(10)
        assert wait_for(
            lambda: {o.metadata.name for o in rc.list(ComposabilityRequest)}
            == {"keep", "vanish"}, timeout=10)
        q = rc.watch(["ComposabilityRequest"])
        while not q.empty():
            q.get_nowait()  # drain the subscribe replay

        server1.should_exit = True
        thread1.join(timeout=5)

        mgr2 = build_manager(Adapter("DRA", MockFabric()), None)
        mgr2.client.create(make_request("keep", target_node="node0"))
        server2, thread2 = start_server(mgr2)
        try:
            # the informer reconnects, re-lists, prunes "vanish"
            assert wait_for(
                lambda: {o.metadata.name for o in rc.list(ComposabilityRequest)}
                == {"keep"}, timeout=20), [
                    o.metadata.name for o in rc.list(ComposabilityRequest)]
            assert rc.try_get(ComposabilityRequest, "vanish") is None
            # the subscriber saw the synthetic DELETED
            deadline = time.monotonic() + 10
            deleted = []
            while time.monotonic() < deadline and not deleted:
                try:
                    ev = q.get(timeout=1)
                except Exception:
                    continue
                if ev.type == "DELETED" and ev.object.metadata.name == "vanish":
                    deleted.append(ev)
            assert deleted, "no synthetic DELETED for the pruned object"
        finally:
            server2.should_exit = True
            thread2.join(timeout=5)
    finally:
        _os.environ.pop("CRO_WATCH_TIMEOUT", None)
        rc.close()


def test_cached_client_syncer_repairs_drift(api_server):
    """The upstream syncer over the CACHED client: out-of-band fabric
    drift (a composed device with no CR) is detected against cached lists
    and repaired with a detach CR after the grace period — the cache must
    not hide the CR set from the diff."""
    from cro_amd.api.v1alpha1.types import ComposableResource

    url, server_mgr = api_server
    remote = RemoteClient(url, cache=True)
    fabric = MockFabric(models={"mi355x": 8})
    mgr = build_manager(
        Adapter("DRA", fabric), None, client=remote, enable_webhook=False,
        syncer_period=0.3, syncer_grace=0.5,
    )
    ops = MockNodeOps(client=mgr.client)
    mgr.resource_reconciler.node_ops = ops
    mgr.syncer.node_ops = ops

    orig_add = fabric.add_resource

    def add_resource(resource):
        did, cdi = orig_add(resource)
        ops.fabric_composed(resource.spec.target_node, did)
        return did, cdi

    fabric.add_resource = add_resource
    node = Node()
    node.metadata.name = "drift-node"
    node.status.capacity.milli_cpu = 64000
    node.status.capacity.memory = 1 << 40
    node.status.capacity.allowed_pod_number = 128
    server_mgr.client.create(node)
    mgr.start()
    try:
        # out-of-band: the fabric composes a device behind the operator's back
        did = next(iter(fabric._pool))
        fabric.force_attach(did, "drift-node")
        ops.fabric_composed("drift-node", did)
        # the syncer (reading CRs through the cache) must repair: a
        # ready-to-detach CR appears, drains, and the fabric releases it
        assert wait_for(lambda: fabric.attached_to("drift-node") == [], timeout=30), \
            fabric.attached_to("drift-node")
        assert wait_for(
            lambda: server_mgr.client.list(ComposableResource) == [], timeout=30)
    finally:
        mgr.stop()
        remote.close()
