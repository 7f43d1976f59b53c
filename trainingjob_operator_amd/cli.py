"""aitjctl — an ops CLI for AITrainingJobs.

The reference's only interface was raw kubectl (README.md:12-19:
`kubectl get aitj`); this adds first-class get/describe/resize/delete
with operator-aware output (phase, per-role replica counters, restart
counts, conditions, the EdlPolicy=Auto target annotation).

    python -m trainingjob_operator_amd.cli get -n default
    python -m trainingjob_operator_amd.cli describe my-job
    python -m trainingjob_operator_amd.cli resize my-job 8 --role trainer
    python -m trainingjob_operator_amd.cli delete my-job

`resize` is the Manual-elastic entry point: it patches spec.replicas
inside [minReplicas, maxReplicas] and the controller performs the
epoch-bumped world restart.
"""
from __future__ import annotations

import argparse
import sys
import time
from typing import List, Optional

from .api.types import AITrainingJob
from .controller.pods import TARGET_ANNOTATION
from .utils.k8stime import parse_time


def _age(created: Optional[str], now: Optional[float] = None) -> str:
    t = parse_time(created)
    if t is None:
        return "?"
    s = int((now or time.time()) - t)
    if s < 120:
        return f"{s}s"
    if s < 7200:
        return f"{s // 60}m"
    if s < 172800:
        return f"{s // 3600}h"
    return f"{s // 86400}d"


def _fmt_table(rows: List[List[str]]) -> str:
    if not rows:
        return ""
    widths = [max(len(r[i]) for r in rows) for i in range(len(rows[0]))]
    return "\n".join("  ".join(c.ljust(w) for c, w in zip(r, widths)).rstrip()
                     for r in rows)


def cmd_get(api, namespace: str, out=None) -> int:
    out = out or sys.stdout
    rows = [["NAME", "PHASE", "REPLICAS", "RESTARTS", "AGE"]]
    for jd in api.list_jobs(namespace or None):
        job = AITrainingJob.from_dict(jd)
        reps = []
        restarts = 0
        for rt, rs in sorted(job.spec.replica_specs.items()):
            st = job.status.replica_statuses.get(rt)
            active = st.active if st else 0
            reps.append(f"{rt}:{active}/{rs.replicas or 0}")
            restarts += job.status.restart_counts.get(rt, 0)
        rows.append([job.name, job.status.phase or "None",
                     ",".join(reps) or "-", str(restarts),
                     _age(job.metadata.get("creationTimestamp"))])
    print(_fmt_table(rows), file=out)
    return 0


def cmd_describe(api, namespace: str, name: str, out=None) -> int:
    out = out or sys.stdout
    job = AITrainingJob.from_dict(api.get_job(namespace, name))
    print(f"Name:      {job.name}", file=out)
    print(f"Namespace: {job.namespace}", file=out)
    print(f"Phase:     {job.status.phase or 'None'}", file=out)
    for label, v in (("Start", job.status.start_time),
                     ("Running", job.status.start_running_time),
                     ("End", job.status.end_time)):
        if v:
            print(f"{label + ':':<11}{v}", file=out)
    print("Replicas:", file=out)
    for rt, rs in sorted(job.spec.replica_specs.items()):
        st = job.status.replica_statuses.get(rt)
        line = (f"  {rt}: {rs.replicas or 0} desired"
                f" (min={rs.min_replicas} max={rs.max_replicas}"
                f" edl={rs.edl_policy or 'Never'})")
        tgt = job.annotations.get(f"{TARGET_ANNOTATION}-{rt.lower()}")
        if tgt is not None:
            line += f" auto-target={tgt}"
        print(line, file=out)
        if st:
            print(f"    pending={st.pending} active={st.active} "
                  f"succeeded={st.succeeded} failed={st.failed} "
                  f"restarts={job.status.restart_counts.get(rt, 0)}",
                  file=out)
    if job.status.conditions:
        print("Conditions:", file=out)
        rows = [["  TYPE", "STATUS", "REASON", "MESSAGE"]]
        for c in job.status.conditions:
            rows.append(["  " + c.type, c.status, c.reason or "",
                         (c.message or "")[:60]])
        print(_fmt_table(rows), file=out)
    try:
        events = api.list_events(namespace, involved_name=name)
    except Exception:                      # older servers: best-effort
        events = []
    if events:
        print("Events:", file=out)
        rows = [["  AGE", "TYPE", "REASON", "MESSAGE"]]
        for e in events[-10:]:
            rows.append([
                "  " + _age(e.get("lastTimestamp")
                            or e.get("firstTimestamp")),
                e.get("type", ""), e.get("reason", ""),
                (e.get("message") or "")[:60]])
        print(_fmt_table(rows), file=out)
    return 0


def cmd_resize(api, namespace: str, name: str, replicas: int,
               role: str, out=None) -> int:
    out = out or sys.stdout
    jd = api.get_job(namespace, name)
    job = AITrainingJob.from_dict(jd)
    rs = job.spec.replica_specs.get(role)
    if rs is None:
        print(f"error: role {role!r} not in "
              f"{sorted(job.spec.replica_specs)}", file=sys.stderr)
        return 1
    lo = rs.min_replicas if rs.min_replicas is not None else replicas
    hi = rs.max_replicas if rs.max_replicas is not None else replicas
    if not lo <= replicas <= hi:
        print(f"error: {replicas} outside [minReplicas={lo}, "
              f"maxReplicas={hi}]", file=sys.stderr)
        return 1
    jd["spec"]["replicaSpecs"][role]["replicas"] = replicas
    api.update_job(namespace, name, jd)
    print(f"{name}: {role} -> {replicas} replicas "
          f"(controller restarts the world at the new size)", file=out)
    return 0


def cmd_logs(api, namespace: str, name: str, replica: str,
             tail: int, out=None) -> int:
    out = out or sys.stdout
    from .kube import objects as ko
    pods = api.list_pods(namespace, ko.job_selector(name))
    if replica:
        pods = [p for p in pods
                if ko.name_of(p) == f"{name}-{replica}"]
    if not pods:
        print(f"error: no pods for job {name!r}"
              + (f" replica {replica!r}" if replica else ""),
              file=sys.stderr)
        return 1
    for p in sorted(pods, key=ko.name_of):
        pname = ko.name_of(p)
        if len(pods) > 1:
            print(f"==> {pname} <==", file=out)
        print(api.read_pod_log(namespace, pname, tail_lines=tail or None),
              file=out)
    return 0


def cmd_delete(api, namespace: str, name: str, out=None) -> int:
    out = out or sys.stdout
    api.delete_job(namespace, name)
    print(f"{name} deleted", file=out)
    return 0


def cmd_nodes(api, out=None) -> int:
    """Per-node GPU health as published by the node agent (EDLGPUHealthy
    condition + the gpu-health annotation detail)."""
    import json

    from .agent.node_agent import GPU_HEALTH_ANNOTATION, GPU_HEALTH_CONDITION
    out = out or sys.stdout
    rows = [["NODE", "READY", "GPU-HEALTH", "GPUS", "DETAIL"]]
    for node in api.list_nodes():
        name = node.get("metadata", {}).get("name", "?")
        conds = {c.get("type"): c.get("status")
                 for c in node.get("status", {}).get("conditions") or []}
        ready = conds.get("Ready", "?")
        gh = conds.get(GPU_HEALTH_CONDITION, "-")
        ngpus, detail = "-", ""
        ann = (node.get("metadata", {}).get("annotations") or {}).get(
            GPU_HEALTH_ANNOTATION)
        if ann:
            try:
                rep = json.loads(ann)
                ngpus = str(len(rep.get("gpus", [])))
                detail = rep.get("summary", "")[:50]
            except ValueError:
                detail = "unparseable annotation"
        rows.append([name, ready, gh, ngpus, detail])
    print(_fmt_table(rows), file=out)
    return 0


def main(argv=None, api=None) -> int:
    common = argparse.ArgumentParser(add_help=False)
    common.add_argument("-n", "--namespace", default="default")
    common.add_argument("-A", "--all-namespaces", action="store_true")
    common.add_argument("--master", default="",
                        help="API server URL (default: kube proxy / "
                             "in-cluster)")
    ap = argparse.ArgumentParser(prog="aitjctl", description=__doc__)
    sub = ap.add_subparsers(dest="cmd", required=True)
    sub.add_parser("get", parents=[common])
    d = sub.add_parser("describe", parents=[common])
    d.add_argument("name")
    r = sub.add_parser("resize", parents=[common])
    r.add_argument("name")
    r.add_argument("replicas", type=int)
    r.add_argument("--role", default="trainer")
    x = sub.add_parser("delete", parents=[common])
    x.add_argument("name")
    sub.add_parser("nodes", parents=[common])
    lg = sub.add_parser("logs", parents=[common])
    lg.add_argument("name")
    lg.add_argument("--replica", default="",
                    help="single replica, e.g. trainer-0 (default: all)")
    lg.add_argument("--tail", type=int, default=0)
    args = ap.parse_args(argv)

    if api is None:
        from .kube.client import RealKubeApi
        api = RealKubeApi(base_url=args.master or None)
    if args.cmd == "get":
        return cmd_get(api, "" if args.all_namespaces else args.namespace)
    if args.cmd == "describe":
        return cmd_describe(api, args.namespace, args.name)
    if args.cmd == "resize":
        return cmd_resize(api, args.namespace, args.name, args.replicas,
                          args.role)
    if args.cmd == "delete":
        return cmd_delete(api, args.namespace, args.name)
    if args.cmd == "logs":
        return cmd_logs(api, args.namespace, args.name, args.replica,
                        args.tail)
    if args.cmd == "nodes":
        return cmd_nodes(api)
    return 2


if __name__ == "__main__":
    sys.exit(main())
