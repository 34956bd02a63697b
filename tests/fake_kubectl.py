#!/usr/bin/env python3
"""A recording kubectl stand-in for K8sDriver behavioral tests (no
cluster in CI). State lives under $FAKE_KUBE_DIR:
  applied.jsonl   every manifest piped to `apply -f -`
  deleted.jsonl   every delete
Behavior knobs: a Deployment named *-stuck-* fails rollout status."""
import json
import os
import sys

STATE = os.environ["FAKE_KUBE_DIR"]


def record(fname, obj):
    with open(os.path.join(STATE, fname), "a") as f:
        f.write(json.dumps(obj) + "\n")


def load(fname):
    path = os.path.join(STATE, fname)
    if not os.path.exists(path):
        return []
    return [json.loads(l) for l in open(path) if l.strip()]


def main():
    args = sys.argv[1:]
    ns = "default"
    if "-n" in args:
        i = args.index("-n")
        ns = args[i + 1]
        args = args[:i] + args[i + 2:]
    verb = args[0] if args else ""

    if verb == "apply":
        manifest = json.loads(sys.stdin.read())
        record("applied.jsonl", {"ns": ns, "manifest": manifest})
        name = manifest.get("metadata", {}).get("name", "")
        print(f"{manifest.get('kind', 'object').lower()}/{name} configured")
        return 0

    if verb == "rollout":  # rollout status deployment/<name> --timeout=Ns
        target = args[2]
        if "-stuck-" in target:
            print("error: deadline exceeded", file=sys.stderr)
            return 1
        print(f'{target} successfully rolled out')
        return 0

    if verb == "get" and args[1] == "pods":
        sel = args[args.index("-l") + 1] if "-l" in args else ""
        svc = sel.split("=", 1)[1] if "=" in sel else ""
        items = []
        for rec in load("applied.jsonl"):
            m = rec["manifest"]
            if m.get("kind") == "Deployment" and \
                    m["metadata"].get("name") == svc and rec["ns"] == ns:
                for i in range(m.get("spec", {}).get("replicas", 1)):
                    items.append({
                        "metadata": {"name": f"{svc}-{i}"},
                        "status": {"podIP": f"10.0.0.{i + 1}"},
                    })
        print(json.dumps({"items": items}))
        return 0

    if verb == "get" and args[1] == "kubetorchworkloads":
        # latest apply wins per (ns, name)
        latest = {}
        for rec in load("applied.jsonl"):
            m = rec["manifest"]
            if m.get("kind") == "KubetorchWorkload":
                latest[(rec["ns"], m["metadata"]["name"])] = m
        print(json.dumps({"items": list(latest.values())}))
        return 0

    if verb == "get" and args[1] == "secrets":
        items = [rec["manifest"] for rec in load("applied.jsonl")
                 if rec["manifest"].get("kind") == "Secret" and rec["ns"] == ns]
        print(json.dumps({"items": items}))
        return 0

    if verb == "get" and args[1] == "events":
        print(json.dumps({"items": []}))
        return 0

    if verb == "delete":
        record("deleted.jsonl", {"ns": ns, "args": args[1:]})
        return 0

    print(f"fake kubectl: unhandled {args}", file=sys.stderr)
    return 2


if __name__ == "__main__":
    sys.exit(main())
