"""lwsctl — kubectl-style CLI for the lws_amd API server.

    python -m lws_amd.client.ctl apply -f examples/lws-llama70b.yaml
    python -m lws_amd.client.ctl get lws
    python -m lws_amd.client.ctl get lws my-lws -o yaml
    python -m lws_amd.client.ctl scale lws my-lws --replicas 4
    python -m lws_amd.client.ctl delete lws my-lws
"""
from __future__ import annotations

import argparse
import json
import sys

import yaml

from ..api import serde
from .clientset import RESOURCES, Clientset

ALIASES = {
    "lws": "leaderworkersets",
    "leaderworkerset": "leaderworkersets",
    "ds": "disaggregatedsets",
    "disaggregatedset": "disaggregatedsets",
    "dsrs": "disaggregatedsetrolescalers",
    "pod": "pods", "po": "pods",
    "sts": "statefulsets", "statefulset": "statefulsets",
    "svc": "services", "service": "services",
}
KIND_TO_RESOURCE = {kind: res for res, (kind, _) in RESOURCES.items()}


def _resource(name: str) -> str:
    r = ALIASES.get(name, name)
    if r not in RESOURCES:
        sys.exit(f"error: unknown resource {name!r}")
    return r


def _client(args) -> Clientset:
    return Clientset.for_server(args.server)


def cmd_apply(args) -> int:
    cs = _client(args)
    with open(args.filename) as f:
        docs = [d for d in yaml.safe_load_all(f) if d]
    for doc in docs:
        kind = doc.get("kind", "")
        resource = KIND_TO_RESOURCE.get(kind)
        if resource is None:
            sys.exit(f"error: unsupported kind {kind!r}")
        _, model = RESOURCES[resource]
        obj = serde.from_dict(model, doc)
        ns = obj.metadata.namespace or args.namespace
        rc = getattr(cs, {
            "leaderworkersets": "leader_worker_sets",
            "disaggregatedsets": "disaggregated_sets",
            "disaggregatedsetrolescalers": "role_scalers"}[resource])(ns)
        out = rc.apply(obj)
        print(f"{kind.lower()}/{out.metadata.name} applied")
    return 0


def _table(objs, resource):
    rows = []
    if resource == "leaderworkersets":
        rows.append(("NAME", "READY", "DESIRED", "UP-TO-DATE"))
        for o in objs:
            rows.append((o.metadata.name, o.status.ready_replicas,
                         o.spec.replicas, o.status.updated_replicas))
    elif resource == "disaggregatedsets":
        rows.append(("NAME", "ROLES", "AVAILABLE"))
        for o in objs:
            cond = next((c.status for c in o.status.conditions
                         if c.type == "Available"), "Unknown")
            rows.append((o.metadata.name,
                         ",".join(r.name for r in o.spec.roles), cond))
    elif resource == "pods":
        rows.append(("NAME", "PHASE", "NODE"))
        for o in objs:
            rows.append((o.metadata.name, o.status.phase, o.node_name))
    else:
        rows.append(("NAME",))
        for o in objs:
            rows.append((o.metadata.name,))
    widths = [max(len(str(r[i])) for r in rows) for i in range(len(rows[0]))]
    for r in rows:
        print("  ".join(str(v).ljust(w) for v, w in zip(r, widths)))


def cmd_get(args) -> int:
    resource = _resource(args.resource)
    cs = _client(args)
    rc_name = {"leaderworkersets": "leader_worker_sets",
               "disaggregatedsets": "disaggregated_sets",
               "disaggregatedsetrolescalers": "role_scalers",
               "pods": "pods", "statefulsets": "statefulsets",
               "services": "services"}[resource]
    rc = getattr(cs, rc_name)(args.namespace)
    if args.name:
        obj = rc.get(args.name)
        if obj is None:
            sys.exit(f"error: {args.name} not found")
        if args.output == "yaml":
            print(yaml.safe_dump(serde.to_dict(obj), sort_keys=False))
        elif args.output == "json":
            print(json.dumps(serde.to_dict(obj), indent=2))
        else:
            _table([obj], resource)
    else:
        _table(rc.list(), resource)
    return 0


def cmd_delete(args) -> int:
    resource = _resource(args.resource)
    cs = _client(args)
    rc = getattr(cs, {"leaderworkersets": "leader_worker_sets",
                      "disaggregatedsets": "disaggregated_sets",
                      "disaggregatedsetrolescalers": "role_scalers"}[resource]
                 )(args.namespace)
    rc.delete(args.name)
    print(f"{args.resource}/{args.name} deleted")
    return 0


def cmd_events(args) -> int:
    import time as _time
    cs = _client(args)
    evs = cs.events(args.namespace).list()
    evs.sort(key=lambda e: e.last_timestamp)
    print(f"{'TYPE':8s} {'REASON':22s} {'OBJECT':32s} {'COUNT':5s} "
          f"{'AGE':6s} MESSAGE")
    now = _time.time()
    for e in evs:
        obj = f"{e.involved_object.kind}/{e.involved_object.name}"
        age = f"{int(now - e.last_timestamp)}s"
        print(f"{e.type:8s} {e.reason:22s} {obj:32s} {e.count:<5d} "
              f"{age:6s} {e.message}")
    return 0


def cmd_describe(args) -> int:
    resource = _resource(args.resource)
    cs = _client(args)
    rc_name = {"leaderworkersets": "leader_worker_sets",
               "disaggregatedsets": "disaggregated_sets",
               "disaggregatedsetrolescalers": "role_scalers",
               "pods": "pods", "statefulsets": "statefulsets",
               "services": "services"}[resource]
    obj = getattr(cs, rc_name)(args.namespace).get(args.name)
    if obj is None:
        sys.exit(f"error: {args.name} not found")
    d = serde.to_dict(obj)
    print(yaml.safe_dump({"metadata": d.get("metadata"),
                          "spec": d.get("spec"),
                          "status": d.get("status")}, sort_keys=False))
    evs = [e for e in cs.events(args.namespace).list()
           if e.involved_object.name == args.name]
    if evs:
        print("Events:")
        for e in sorted(evs, key=lambda e: e.last_timestamp):
            print(f"  {e.type} {e.reason} (x{e.count}): {e.message}")
    return 0


def cmd_wait(args) -> int:
    """kubectl wait --for=condition=Available analogue (the DS e2e
    waiters poll conditions the same way)."""
    import time as _time
    resource = _resource(args.resource)
    cs = _client(args)
    rc_name = {"leaderworkersets": "leader_worker_sets",
               "disaggregatedsets": "disaggregated_sets"}[resource]
    rc = getattr(cs, rc_name)(args.namespace)
    want = args.for_condition.split("=")
    cond_type = want[0]
    cond_status = want[1] if len(want) > 1 else "True"
    deadline = _time.time() + args.timeout
    while _time.time() < deadline:
        obj = rc.get(args.name)
        if obj is not None:
            for c in obj.status.conditions or []:
                if c.type == cond_type and c.status == cond_status:
                    print(f"{args.resource}/{args.name} condition met: "
                          f"{cond_type}={cond_status}")
                    return 0
        _time.sleep(0.2)
    sys.exit(f"error: timed out waiting for {cond_type}={cond_status} "
             f"on {args.resource}/{args.name}")


def cmd_scale(args) -> int:
    resource = _resource(args.resource)
    cs = _client(args)
    rc = getattr(cs, {"leaderworkersets": "leader_worker_sets",
                      "disaggregatedsetrolescalers": "role_scalers"}[resource]
                 )(args.namespace)
    out = rc.scale(args.name, args.replicas)
    print(f"{args.resource}/{args.name} scaled to {args.replicas}")
    return 0


def main(argv=None) -> int:
    p = argparse.ArgumentParser(prog="lwsctl")
    p.add_argument("--server", default="http://127.0.0.1:8080")
    p.add_argument("-n", "--namespace", default="default")
    sub = p.add_subparsers(dest="cmd", required=True)

    sp = sub.add_parser("apply")
    sp.add_argument("-f", "--filename", required=True)
    sp.set_defaults(fn=cmd_apply)

    sp = sub.add_parser("get")
    sp.add_argument("resource")
    sp.add_argument("name", nargs="?")
    sp.add_argument("-o", "--output", default="table")
    sp.set_defaults(fn=cmd_get)

    sp = sub.add_parser("delete")
    sp.add_argument("resource")
    sp.add_argument("name")
    sp.set_defaults(fn=cmd_delete)

    sp = sub.add_parser("events")
    sp.set_defaults(fn=cmd_events)

    sp = sub.add_parser("describe")
    sp.add_argument("resource")
    sp.add_argument("name")
    sp.set_defaults(fn=cmd_describe)

    sp = sub.add_parser("wait")
    sp.add_argument("resource")
    sp.add_argument("name")
    sp.add_argument("--for", dest="for_condition", default="Available",
                    help="condition, e.g. Available or Available=True")
    sp.add_argument("--timeout", type=float, default=120.0)
    sp.set_defaults(fn=cmd_wait)

    sp = sub.add_parser("scale")
    sp.add_argument("resource")
    sp.add_argument("name")
    sp.add_argument("--replicas", type=int, required=True)
    sp.set_defaults(fn=cmd_scale)

    args = p.parse_args(argv)
    return args.fn(args)


if __name__ == "__main__":
    sys.exit(main())
