"""Command-line entry points.

The reference deploys its services through entry scripts in an
`ols/test/` tree that is absent from the open-source drop
(README.md:780-783, 942-945 reference the Dockerfile entrypoints);
this CLI plays that role on one node:

    python -m olearning_sim_amd serve [--port 60061] [--svc 0]
    python -m olearning_sim_amd submit task.json [--wait]
    python -m olearning_sim_amd status <task_id>
    python -m olearning_sim_amd stop <task_id>
    python -m olearning_sim_amd queue

`submit/status/stop/queue` talk to a running server when `--server
http://host:port` is given, otherwise they run a one-shot in-process
session over the persistent data directory.
"""

from __future__ import annotations

import argparse
import json
import sys
import time


def _session(args):
    from .session import SimulatorSession
    return SimulatorSession(svc=0, data_dir=args.data_dir or None,
                            auto_start_threads=True)


def _http(args, method: str, path: str, body=None):
    import urllib.request
    url = args.server.rstrip("/") + path
    data = json.dumps(body).encode() if body is not None else None
    req = urllib.request.Request(
        url, data=data, method=method,
        headers={"Content-Type": "application/json"})
    with urllib.request.urlopen(req, timeout=30) as resp:
        return json.loads(resp.read())


def cmd_serve(args) -> int:
    sess = _session(args)
    print(f"serving JSON API on {args.host}:{args.port} (svc={sess.svc})")
    sess.serve(host=args.host, port=args.port, block=True)
    return 0


def cmd_submit(args) -> int:
    with open(args.task_json) as f:
        task = json.load(f)
    if args.server:
        out = _http(args, "POST", "/taskmgr/submitTask", {"task": task})
        print(json.dumps(out))
        return 0 if out.get("is_success") else 1
    sess = _session(args)
    ok, msg = sess.task_mgr.submit_task(json.dumps(task))
    print(json.dumps({"is_success": ok, "message": msg}))
    if not ok:
        sess.shutdown()
        return 1
    if args.wait:
        task_id = task["task_id"]
        while True:
            st = sess.task_mgr.get_task_status(task_id)
            if st.is_terminal():
                break
            time.sleep(0.5)
        sess.task_mgr.step_release()
        row = sess.task_mgr.table.get_row(task_id)
        print(json.dumps({"task_status": st.value,
                          "logical_result": row.get("logical_result")}))
    sess.shutdown()
    return 0


def cmd_status(args) -> int:
    if args.server:
        print(json.dumps(_http(args, "GET",
                               f"/taskmgr/getTaskStatus/{args.task_id}")))
        return 0
    sess = _session(args)
    st = sess.task_mgr.get_task_status(args.task_id)
    print(json.dumps({"task_status": st.value}))
    sess.shutdown()
    return 0


def cmd_stop(args) -> int:
    if args.server:
        print(json.dumps(_http(args, "POST",
                               f"/taskmgr/stopTask/{args.task_id}")))
        return 0
    sess = _session(args)
    ok, msg = sess.task_mgr.stop_task(args.task_id)
    print(json.dumps({"is_success": ok, "message": msg}))
    sess.shutdown()
    return 0 if ok else 1


def cmd_queue(args) -> int:
    if args.server:
        print(json.dumps(_http(args, "GET", "/taskmgr/getTaskQueue")))
        return 0
    sess = _session(args)
    print(json.dumps({"tasks": sess.task_mgr.get_task_queue()}))
    sess.shutdown()
    return 0


def cmd_result(args) -> int:
    if args.server:
        print(json.dumps(_http(args, "GET",
                               f"/taskmgr/getTaskResult/{args.task_id}")))
        return 0
    sess = _session(args)
    row = sess.task_mgr.table.get_row(args.task_id)
    if row is None:
        print(json.dumps({"error": "task not found"}))
        sess.shutdown()
        return 1
    print(json.dumps({
        "task_id": args.task_id,
        "task_status": sess.task_mgr.get_task_status(args.task_id).value,
        "logical_round": row.get("logical_round"),
        "logical_operator": row.get("logical_operator"),
        "logical_result": json.loads(row["logical_result"])
        if row.get("logical_result") else None,
        "device_result": json.loads(row["device_result"])
        if row.get("device_result") else None,
    }))
    sess.shutdown()
    return 0


def cmd_perf(args) -> int:
    if args.server:
        print(json.dumps(_http(args, "GET",
                               f"/performancemgr/summary/{args.task_id}")))
        return 0
    sess = _session(args)
    print(json.dumps(sess.performance_mgr.summary(args.task_id)))
    sess.shutdown()
    return 0


def main(argv=None) -> int:
    ap = argparse.ArgumentParser(prog="olearning_sim_amd")
    ap.add_argument("--server", default="",
                    help="base URL of a running API server (else in-process)")
    ap.add_argument("--data-dir", default="")
    sub = ap.add_subparsers(dest="cmd", required=True)

    p = sub.add_parser("serve")
    p.add_argument("--host", default="127.0.0.1")
    p.add_argument("--port", type=int, default=60061)
    p.set_defaults(fn=cmd_serve)

    p = sub.add_parser("submit")
    p.add_argument("task_json")
    p.add_argument("--wait", action="store_true")
    p.set_defaults(fn=cmd_submit)

    p = sub.add_parser("status")
    p.add_argument("task_id")
    p.set_defaults(fn=cmd_status)

    p = sub.add_parser("stop")
    p.add_argument("task_id")
    p.set_defaults(fn=cmd_stop)

    p = sub.add_parser("queue")
    p.set_defaults(fn=cmd_queue)

    p = sub.add_parser("result")
    p.add_argument("task_id")
    p.set_defaults(fn=cmd_result)

    p = sub.add_parser("perf")
    p.add_argument("task_id")
    p.set_defaults(fn=cmd_perf)

    args = ap.parse_args(argv)
    return args.fn(args)


if __name__ == "__main__":
    sys.exit(main())
