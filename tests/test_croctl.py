"""croctl CLI tests against an in-process API server."""

import pytest
import yaml
from fastapi.testclient import TestClient

from cro_amd.bench_harness import build_local_stack
from cro_amd.cmd.croctl import main
from cro_amd.server.api import build_app


@pytest.fixture
def cli(capsys):
    stack = build_local_stack(node_name="node0", use_gpu=False)
    stack.mgr.start()
    http = TestClient(build_app(stack.mgr.client))

    def run(*args):
        rc = main(list(args), client=http)
        out = capsys.readouterr()
        return rc, out.out, out.err

    yield run, stack
    stack.mgr.stop()


def write_request(tmp_path, name="r1", size=1):
    path = tmp_path / f"{name}.yaml"
    path.write_text(
        yaml.safe_dump(
            {
                "apiVersion": "cro.hpsys.ibm.ie.com/v1alpha1",
                "kind": "ComposabilityRequest",
                "metadata": {"name": name},
                "spec": {
                    "resource": {
                        "type": "gpu", "model": "mi355x", "size": size,
                        "target_node": "node0",
                    }
                },
            }
        )
    )
    return str(path)


def test_apply_get_scale_delete(cli, tmp_path):
    run, stack = cli
    rc, out, _ = run("apply", "-f", write_request(tmp_path))
    assert rc == 0 and "created" in out

    assert stack.mgr.wait_for(
        lambda: run("get", "composabilityrequests", "r1")[1].count("Running") > 0,
        timeout=10,
    )
    rc, out, _ = run("get", "composabilityrequests")
    assert rc == 0
    assert "NAME" in out and "r1" in out and "mi355x" in out

    rc, out, _ = run("scale", "composabilityrequests", "r1", "--size", "3")
    assert rc == 0
    assert stack.mgr.wait_for(
        lambda: "DEVICES" in run("get", "composabilityrequests", "r1")[1]
        and "3" in run("get", "composabilityrequests", "r1")[1].split("\n")[1],
        timeout=10,
    )

    rc, out, _ = run("describe", "composabilityrequests", "r1")
    assert rc == 0
    described = yaml.safe_load(out)
    assert described["spec"]["resource"]["size"] == 3

    rc, out, _ = run("delete", "composabilityrequests", "r1")
    assert rc == 0
    assert stack.mgr.wait_for(
        lambda: run("get", "composabilityrequests", "r1")[0] == 1, timeout=10
    )


def test_apply_update_existing(cli, tmp_path):
    run, stack = cli
    run("apply", "-f", write_request(tmp_path, size=1))
    rc, out, _ = run("apply", "-f", write_request(tmp_path, size=2))
    assert rc == 0 and "configured" in out


def test_get_composableresources_table(cli, tmp_path):
    run, stack = cli
    run("apply", "-f", write_request(tmp_path))
    assert stack.mgr.wait_for(
        lambda: "Online" in run("get", "composableresources")[1], timeout=10
    )
    rc, out, _ = run("get", "composableresources")
    assert "node0" in out and "GPU-" in out


def test_error_paths(cli):
    run, _ = cli
    rc, _, err = run("get", "composabilityrequests", "nope")
    assert rc == 1 and "404" in err
    rc, _, err = run("delete", "composabilityrequests", "nope")
    assert rc == 1


def test_events_after_lifecycle(cli, tmp_path):
    run, stack = cli
    run("apply", "-f", write_request(tmp_path, name="e1"))
    assert stack.mgr.wait_for(
        lambda: "Running" in run("get", "composabilityrequests", "e1")[1],
        timeout=10,
    )
    stack.mgr.recorder.flush()  # buffered recorder: drain before asserting
    rc, out, _ = run("events")
    assert rc == 0 and "NodesAllocated" in out and "Online" in out
    rc, out, _ = run("events", "--for", "ComposabilityRequest/e1")
    assert rc == 0 and "Running" in out


def test_apply_multidoc_and_directory(cli, tmp_path):
    """kubectl parity: -f with multi-document YAML and with a directory."""
    multi = tmp_path / "pair.yaml"
    multi.write_text(
        "\n---\n".join(
            yaml.safe_dump(
                {
                    "apiVersion": "cro.hpsys.ibm.ie.com/v1alpha1",
                    "kind": "ComposabilityRequest",
                    "metadata": {"name": n},
                    "spec": {"resource": {
                        "type": "gpu", "model": m, "size": 1,
                        "target_node": "node0",
                    }},
                }
            )
            for n, m in (("md1", "mi355x"), ("md2", "mi300x"))
        )
    )
    run, stack = cli
    rc, out, _ = run("apply", "-f", str(multi))
    assert rc == 0
    assert "md1 created" in out and "md2 created" in out

    d = tmp_path / "bundle"
    d.mkdir()
    (d / "one.yaml").write_text(
        yaml.safe_dump({
            "apiVersion": "cro.hpsys.ibm.ie.com/v1alpha1",
            "kind": "ComposabilityRequest",
            "metadata": {"name": "md3"},
            "spec": {"resource": {"type": "gpu", "model": "mi308x",
                                  "size": 1, "target_node": "node0"}},
        })
    )
    rc, out, _ = run("apply", "-f", str(d))
    assert rc == 0 and "md3 created" in out


def test_describe_includes_events(cli, tmp_path):
    """kubectl-describe parity: the object's event trail is appended."""
    run, stack = cli
    run("apply", "-f", write_request(tmp_path, name="d1"))
    assert stack.mgr.wait_for(
        lambda: "Running" in run("get", "composabilityrequests", "d1")[1],
        timeout=10,
    )
    stack.mgr.recorder.flush()
    rc, out, _ = run("describe", "composabilityrequests", "d1")
    assert rc == 0
    assert "Events:" in out and "NodesAllocated" in out and "Running" in out


def test_get_with_label_selector(cli, tmp_path):
    run, stack = cli
    run("apply", "-f", write_request(tmp_path, name="l1"))
    assert stack.mgr.wait_for(
        lambda: "Online" in run("get", "composableresources")[1], timeout=10
    )
    rc, out, _ = run(
        "get", "composableresources", "-l", "app.kubernetes.io/managed-by=l1"
    )
    assert rc == 0 and "gpu-" in out
    rc, out, _ = run(
        "get", "composableresources", "-l", "app.kubernetes.io/managed-by=ghost"
    )
    assert rc == 0 and "gpu-" not in out


def test_patch_command(cli):
    run, stack = cli
    from tests.conftest import make_request

    stack.mgr.client.create(make_request("pz", target_node="node0"))
    rc, out, _ = run("patch", "composabilityrequests", "pz",
                     "-p", '{"spec": {"resource": {"size": 4}}}')
    assert rc == 0 and "patched" in out
    from cro_amd.api.v1alpha1.types import ComposabilityRequest

    assert stack.mgr.client.get(
        ComposabilityRequest, "pz").spec.resource.size == 4
