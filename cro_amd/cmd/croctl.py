"""croctl — kubectl-style CLI for the standalone cro-amd API.

    croctl get composabilityrequests [NAME]
    croctl get composableresources
    croctl describe composabilityrequests NAME
    croctl apply -f request.yaml
    croctl delete composabilityrequests NAME
    croctl scale composabilityrequests NAME --size N
    croctl events [--for KIND/NAME]          # lifecycle audit trail

Server selection: --server or CRO_SERVER (default http://127.0.0.1:8080).
"""

from __future__ import annotations

import argparse
import json
import os
import sys

import httpx
import yaml

BASE = "/apis/cro.hpsys.ibm.ie.com/v1alpha1"

_COLUMNS = {
    "composabilityrequests": (
        ("NAME", lambda o: o["metadata"]["name"]),
        ("TYPE", lambda o: o["spec"]["resource"]["type"]),
        ("MODEL", lambda o: o["spec"]["resource"]["model"]),
        ("SIZE", lambda o: str(o["spec"]["resource"]["size"])),
        ("STATE", lambda o: o["status"]["state"] or "<none>"),
        ("DEVICES", lambda o: str(len(o["status"].get("resources", {})))),
    ),
    "composableresources": (
        ("NAME", lambda o: o["metadata"]["name"]),
        ("TYPE", lambda o: o["spec"]["type"]),
        ("NODE", lambda o: o["spec"]["target_node"]),
        ("STATE", lambda o: o["status"]["state"] or "<none>"),
        ("DEVICE", lambda o: o["status"].get("device_id", "")),
        ("ERROR", lambda o: (o["status"].get("error", "") or "")[:40]),
    ),
}
_COLUMNS["nodes"] = (
    ("NAME", lambda o: o["metadata"]["name"]),
    ("PROVIDER-ID", lambda o: o["status"].get("provider_id", "")),
    ("CPU(m)", lambda o: str(o["status"]["capacity"]["milli_cpu"])),
    ("PODS", lambda o: str(o["status"]["capacity"]["allowed_pod_number"])),
)
_COLUMNS["resourceslices"] = (
    ("NAME", lambda o: o["metadata"]["name"]),
    ("NODE", lambda o: o["spec"]["node_name"]),
    ("DRIVER", lambda o: o["spec"]["driver"]),
    ("DEVICES", lambda o: str(len(o["spec"]["devices"]))),
)
_COLUMNS["leases"] = (
    ("NAME", lambda o: o["metadata"]["name"]),
    ("HOLDER", lambda o: o["spec"].get("holderIdentity", "") or "<released>"),
    ("RENEWED", lambda o: (o["spec"].get("renewTime") or "")[11:19]),
    ("TRANSITIONS", lambda o: str(o["spec"].get("leaseTransitions", 0))),
)
_COLUMNS["events"] = (
    ("LAST-SEEN", lambda o: (o.get("last_seen", "") or "")[11:19]),
    ("TYPE", lambda o: o.get("type", "")),
    ("REASON", lambda o: o.get("reason", "")),
    ("OBJECT", lambda o: f'{o.get("involved_kind", "")}/{o.get("involved_name", "")}'),
    ("COUNT", lambda o: str(o.get("count", 1))),
    ("MESSAGE", lambda o: (o.get("message", "") or "")[:60]),
)
_DEFAULT_COLUMNS = (
    ("NAME", lambda o: o["metadata"]["name"]),
    ("KIND", lambda o: o.get("kind", "")),
)


def _print_table(plural: str, items: list) -> None:
    columns = _COLUMNS.get(plural, _DEFAULT_COLUMNS)
    rows = [[fn(o) for _, fn in columns] for o in items]
    headers = [h for h, _ in columns]
    widths = [
        max(len(headers[i]), *(len(r[i]) for r in rows)) if rows else len(headers[i])
        for i in range(len(headers))
    ]
    print("  ".join(h.ljust(w) for h, w in zip(headers, widths)))
    for r in rows:
        print("  ".join(c.ljust(w) for c, w in zip(r, widths)))


def main(argv=None, client: httpx.Client = None) -> int:
    from .. import __version__

    p = argparse.ArgumentParser(prog="croctl")
    p.add_argument("--version", action="version", version=f"croctl {__version__}")
    p.add_argument("--server", default=os.environ.get("CRO_SERVER", "http://127.0.0.1:8080"))
    p.add_argument("-o", "--output", choices=["table", "yaml", "json"], default="table")
    p.add_argument("--insecure-skip-tls-verify", action="store_true",
                   help="accept any server certificate (self-signed demos)")
    p.add_argument("--certificate-authority", default="",
                   metavar="CA_PEM", help="CA bundle for the server certificate")
    p.add_argument("--token", default=os.environ.get("CRO_API_TOKEN", ""),
                   help="bearer token for the API (default: $CRO_API_TOKEN)")
    sub = p.add_subparsers(dest="command", required=True)

    g = sub.add_parser("get")
    g.add_argument("plural")
    g.add_argument("name", nargs="?")
    g.add_argument("-l", "--selector", default="",
                   metavar="K=V[,K2=V2]", help="label selector")

    d = sub.add_parser("describe")
    d.add_argument("plural")
    d.add_argument("name")

    a = sub.add_parser("apply")
    a.add_argument("-f", "--filename", required=True)

    rm = sub.add_parser("delete")
    rm.add_argument("plural")
    rm.add_argument("name")

    pa = sub.add_parser("patch")
    pa.add_argument("plural")
    pa.add_argument("name")
    pa.add_argument("-p", "--patch", required=True,
                    metavar="JSON", help="RFC 7386 merge patch body")

    sc = sub.add_parser("scale")
    sc.add_argument("plural")
    sc.add_argument("name")
    sc.add_argument("--size", type=int, required=True)

    ev = sub.add_parser("events")
    ev.add_argument("--for", dest="for_object", default="",
                    metavar="KIND/NAME", help="filter by involved object")

    wa = sub.add_parser("watch")
    wa.add_argument("plural")
    wa.add_argument("--count", type=int, default=0,
                    help="exit after N events (0 = forever)")

    args = p.parse_args(argv)
    verify = True
    if args.insecure_skip_tls_verify:
        verify = False
    elif args.certificate_authority:
        verify = args.certificate_authority
    headers = {"Authorization": f"Bearer {args.token}"} if args.token else None
    http = client or httpx.Client(
        base_url=args.server, timeout=30, verify=verify, headers=headers
    )

    def fail(resp) -> int:
        print(f"error: {resp.status_code}: {resp.text}", file=sys.stderr)
        return 1

    if args.command == "get":
        if args.name:
            resp = http.get(f"{BASE}/{args.plural}/{args.name}")
            if resp.status_code != 200:
                return fail(resp)
            items = [resp.json()]
        else:
            params = {"labelSelector": args.selector} if args.selector else {}
            resp = http.get(f"{BASE}/{args.plural}", params=params)
            if resp.status_code != 200:
                return fail(resp)
            items = resp.json()["items"]
        if args.output == "json":
            print(json.dumps(items, indent=2))
        elif args.output == "yaml":
            print(yaml.safe_dump(items, sort_keys=False))
        else:
            _print_table(args.plural, items)
        return 0

    if args.command == "watch":
        # WatchList protocol (k8s 1.27+): sendInitialEvents streams ADDED
        # per current object, an initial-events-end BOOKMARK, then live
        seen = 0
        with http.stream(
            "GET",
            f"{BASE}/{args.plural}?watch=true&sendInitialEvents=true"
            "&allowWatchBookmarks=true",
        ) as resp:
            if resp.status_code != 200:
                print(f"error: {resp.status_code}", file=sys.stderr)
                return 1
            for line in resp.iter_lines():
                if not line.strip():
                    continue  # keepalive
                ev = json.loads(line)
                if ev["type"] == "BOOKMARK":
                    continue
                obj = ev["object"]
                state = (obj.get("status") or {}).get("state", "")
                print(f'{ev["type"]:<9} {obj["metadata"]["name"]} {state}')
                seen += 1
                if args.count and seen >= args.count:
                    return 0
        return 0

    if args.command == "events":
        resp = http.get(f"{BASE}/events")
        if resp.status_code != 200:
            return fail(resp)
        items = sorted(resp.json()["items"], key=lambda o: o.get("last_seen", ""))
        if args.for_object:
            kind, _, name = args.for_object.partition("/")
            items = [
                o for o in items
                if o.get("involved_kind", "").lower() == kind.lower()
                and (not name or o.get("involved_name", "") == name)
            ]
        if args.output == "json":
            print(json.dumps(items, indent=2))
        elif args.output == "yaml":
            print(yaml.safe_dump(items, sort_keys=False))
        else:
            _print_table("events", items)
        return 0

    if args.command == "describe":
        resp = http.get(f"{BASE}/{args.plural}/{args.name}")
        if resp.status_code != 200:
            return fail(resp)
        obj = resp.json()
        print(yaml.safe_dump(obj, sort_keys=False))
        # kubectl-describe parity: the object's recent events
        ev_resp = http.get(f"{BASE}/events")
        if ev_resp.status_code == 200:
            kind = obj.get("kind", "")
            mine = sorted(
                (
                    e for e in ev_resp.json()["items"]
                    if e.get("involved_kind") == kind
                    and e.get("involved_name") == args.name
                ),
                key=lambda e: e.get("last_seen", ""),
            )
            if mine:
                print("Events:")
                for e in mine[-15:]:
                    print(
                        f"  {(e.get('last_seen') or '')[11:19]}  "
                        f"{e.get('type', ''):<8} {e.get('reason', ''):<18} "
                        f"(x{e.get('count', 1)})  {e.get('message', '')[:70]}"
                    )
        return 0

    def update_with_conflict_retry(plural: str, name: str, mutate) -> httpx.Response:
        """GET-mutate-PUT with optimistic-concurrency retries: concurrent
        reconciles bump resourceVersion between our read and write."""
        resp = None
        for _ in range(8):
            current = http.get(f"{BASE}/{plural}/{name}")
            if current.status_code != 200:
                return current
            obj = current.json()
            mutate(obj)
            resp = http.put(f"{BASE}/{plural}/{name}", json=obj)
            if resp.status_code != 409:
                return resp
        return resp

    if args.command == "apply":
        # kubectl shape: -f FILE (multi-document YAML) or -f DIRECTORY
        paths = [args.filename]
        if os.path.isdir(args.filename):
            paths = sorted(
                os.path.join(args.filename, p)
                for p in os.listdir(args.filename)
                if p.endswith((".yaml", ".yml", ".json"))
            )
        docs = []
        for path in paths:
            with open(path) as f:
                docs.extend(d for d in yaml.safe_load_all(f) if d)
        if not docs:
            print("error: no objects to apply", file=sys.stderr)
            return 1
        for obj in docs:
            plural = obj["kind"].lower() + "s"
            name = obj["metadata"]["name"]
            if http.get(f"{BASE}/{plural}/{name}").status_code == 200:
                resp = update_with_conflict_retry(
                    plural, name, lambda cur, o=obj: cur.update({"spec": o["spec"]})
                )
                verb = "configured"
            else:
                resp = http.post(f"{BASE}/{plural}", json=obj)
                verb = "created"
            if resp.status_code not in (200, 201):
                return fail(resp)
            print(f"{plural}/{name} {verb}")
        return 0

    if args.command == "patch":
        resp = http.patch(
            f"{BASE}/{args.plural}/{args.name}",
            content=args.patch,
            headers={"Content-Type": "application/merge-patch+json"},
        )
        if resp.status_code != 200:
            return fail(resp)
        print(f"{args.plural}/{args.name} patched")
        return 0

    if args.command == "delete":
        resp = http.delete(f"{BASE}/{args.plural}/{args.name}")
        if resp.status_code not in (200, 202):
            return fail(resp)
        print(f"{args.plural}/{args.name} deleted")
        return 0

    if args.command == "scale":
        def set_size(obj):
            obj["spec"]["resource"]["size"] = args.size

        resp = update_with_conflict_retry(args.plural, args.name, set_size)
        if resp is None or resp.status_code != 200:
            return fail(resp)
        print(f"{args.plural}/{args.name} scaled to {args.size}")
        return 0

    return 2


if __name__ == "__main__":
    sys.exit(main())
