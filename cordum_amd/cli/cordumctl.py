"""cordumctl — the operator CLI (compat command surface).

Oracle: cmd/cordumctl/main.go:16-279 — `init` (scaffold), `serve` (replaces
`dev`/`up`: one process instead of docker compose), `status`,
`workflow create/delete/list`, `run start/get/delete/timeline/cancel/rerun`,
`approval list/approve/reject`, `dlq list/retry/delete`,
`pack create/install/uninstall/list/show/verify`, `job submit/status/cancel`.

Usage: python -m cordum_amd.cli.cordumctl <command> ... (or the `cordumctl`
entry point from setup.py).
"""
from __future__ import annotations

import argparse
import io
import json
import os
import sys
import tarfile
from pathlib import Path


def make_client(args):
    from ..sdk.client import Client

    return Client(
        base_url=args.server,
        api_key=args.api_key or os.environ.get("CORDUM_API_KEY", ""),
        principal_id=os.environ.get("CORDUM_PRINCIPAL", "cordumctl"),
        role=os.environ.get("CORDUM_ROLE", "admin"),
    )


def out(data):
    print(json.dumps(data, indent=2, default=str))


def cmd_status(args):
    c = make_client(args)
    out(c.status())


def cmd_serve(args):
    """Run the single-process control-plane node (replaces `cordumctl up`)."""
    from ..cli.serve import serve

    serve(host=args.host, port=args.port, config_dir=args.config_dir,
          workers=args.workers, checkpoint_dir=args.checkpoint_dir,
          checkpoint_interval_s=args.checkpoint_interval,
          dashboard_dir=args.dashboard_dir, bridge_port=args.bridge_port)


def cmd_init(args):
    """Scaffold a project directory with config + an example workflow."""
    root = Path(args.dir)
    (root / "config").mkdir(parents=True, exist_ok=True)
    (root / "workflows").mkdir(parents=True, exist_ok=True)
    defaults = Path(__file__).resolve().parent.parent.parent / "config"
    for name in ("pools.yaml", "timeouts.yaml", "safety.yaml", "system.yaml"):
        src = defaults / name
        dst = root / "config" / name
        if src.exists() and not dst.exists():
            dst.write_text(src.read_text())
    example = root / "workflows" / "hello.json"
    if not example.exists():
        example.write_text(json.dumps({
            "id": "hello-workflow",
            "name": "Hello",
            "steps": {"echo": {"type": "worker", "topic": "job.echo",
                               "input": {"message": "${input.message}"}}},
        }, indent=2))
    print(f"initialized {root}")


def cmd_workflow(args):
    c = make_client(args)
    if args.action == "scaffold":
        cmd_pack_scaffold(Path(args.dir), args.id or "my-pack", force=args.force)
    elif args.action == "create":
        import yaml

        path = Path(args.file or args.id)
        doc = yaml.safe_load(path.read_text())  # YAML or JSON, like the Go CLI
        out(c.create_workflow(doc))
    elif args.action == "list":
        out(c.list_workflows())
    elif args.action == "get":
        out(c.get_workflow(args.id))
    elif args.action == "delete":
        out(c.delete_workflow(args.id))


def cmd_run(args):
    c = make_client(args)
    if args.action == "start":
        inp = json.loads(args.input) if args.input else {}
        wf_id = args.workflow or args.id  # `run start <wf>` or -w <wf>
        out(c.start_run(wf_id, inp, dry_run=args.dry_run))
    elif args.action == "get":
        out(c.get_run(args.id))
    elif args.action == "list":
        out(c.list_runs())
    elif args.action == "timeline":
        out(c.run_timeline(args.id))
    elif args.action == "delete":
        out(c.delete_run(args.id))
    elif args.action == "rerun":
        out(c.rerun(args.id, step_id=args.step or ""))
    elif args.action == "step":
        out(c.approve_step(args.workflow, args.id, args.step, approved=not args.reject))


def cmd_approval(args):
    c = make_client(args)
    if args.action == "list":
        out(c.list_approvals())
    elif args.action == "approve":
        out(c.approve_job(args.job_id, reason=args.reason or ""))
    elif args.action == "reject":
        out(c.reject_job(args.job_id, reason=args.reason or ""))


def cmd_dlq(args):
    c = make_client(args)
    if args.action == "list":
        out(c.list_dlq())
    elif args.action == "retry":
        out(c.retry_dlq(args.job_id))
    elif args.action == "delete":
        out(c.delete_dlq(args.job_id))


def cmd_job(args):
    c = make_client(args)
    if args.action == "submit":
        out(c.submit_job(args.prompt, topic=args.topic))
    elif args.action == "status":
        out(c.get_job(args.id))
    elif args.action == "cancel":
        out(c.cancel_job(args.id))
    elif args.action == "logs":
        job = c.get_job(args.id)
        out({"events": job.get("result"), "state": job.get("state")})


def build_pack_archive(directory: Path) -> bytes:
    buf = io.BytesIO()
    with tarfile.open(fileobj=buf, mode="w:gz") as tf:
        for path in sorted(directory.rglob("*")):
            if path.is_file():
                tf.add(path, arcname=str(path.relative_to(directory)))
    return buf.getvalue()



PACK_SCAFFOLD_MANIFEST = """apiVersion: cordum.io/v1alpha1
kind: Pack

metadata:
  id: {pid}
  version: 0.1.0
  title: {pid} pack
  description: {pid} workflow pack (scaffolded).

compatibility:
  protocolVersion: 1

topics:
  - topic: job.{pid}.echo
    capability: {pid}.echo

resources:
  schemas:
    - schemas/echo-input.json
  workflows:
    - workflows/echo.json

overlays:
  config:
    - scope: system
      key: default
      json_merge_patch:
        pools:
          topics:
            job.{pid}.echo: [default]
          pools:
            default: {{}}
  policy:
    - id: {pid}/policy
      bundle_fragment: |
        version: {pid}-v1
        rules:
          - id: {pid}-allow-echo
            decision: allow
            match:
              topics: ["job.{pid}.echo"]
"""


def cmd_pack_scaffold(root: Path, pack_id: str, force: bool = False) -> None:
    """`cordumctl pack scaffold` — generate a working pack skeleton
    (reference: cmd/cordumctl/pack_create.go scaffoldPack templates)."""

    def write(rel: str, content: str):
        path = root / rel
        if path.exists() and not force:
            raise SystemExit(f"file exists: {path} (use --force)")
        path.parent.mkdir(parents=True, exist_ok=True)
        path.write_text(content)

    write("pack.yaml", PACK_SCAFFOLD_MANIFEST.format(pid=pack_id))
    write("schemas/echo-input.json", json.dumps({
        "$id": f"{pack_id}/echo-input",
        "type": "object",
        "properties": {"message": {"type": "string"},
                       "author": {"type": "string"}},
        "required": ["message"],
        "additionalProperties": False,
    }, indent=2))
    write("workflows/echo.json", json.dumps({
        "id": f"{pack_id}.echo",
        "name": f"{pack_id} echo",
        "org_id": "default",
        "steps": {"echo": {"type": "worker", "topic": f"job.{pack_id}.echo",
                           "input_schema_id": f"{pack_id}/echo-input",
                           "input": {"message": "${input.message}",
                                     "author": "${input.author}"},
                           "meta": {"pack_id": pack_id,
                                    "capability": f"{pack_id}.echo"}}},
    }, indent=2))
    write("README.md", f"# {pack_id}\n\nScaffolded pack. Install with:\n\n"
                       f"    cordumctl pack install --file .\n")
    print(f"scaffolded pack {pack_id!r} in {root}")


def cmd_pack(args):
    c = make_client(args)
    if args.action == "scaffold":
        cmd_pack_scaffold(Path(args.dir), args.id or "my-pack", force=args.force)
    elif args.action == "create":
        blob = build_pack_archive(Path(args.dir))
        Path(args.output).write_bytes(blob)
        print(f"wrote {args.output} ({len(blob)} bytes)")
    elif args.action == "install":
        p = Path(args.file)
        blob = build_pack_archive(p) if p.is_dir() else p.read_bytes()
        out(c.install_pack(blob))
    elif args.action == "list":
        out(c.list_packs())
    elif args.action == "show":
        packs = c.list_packs()["items"]
        out(next((x for x in packs if x["id"] == args.id), {"error": "not found"}))
    elif args.action == "uninstall":
        out(c.uninstall_pack(args.id))
    elif args.action == "verify":
        out(c.verify_pack(args.id))



def cmd_dev(args):
    """`cordumctl dev` — local dev stack: the single-process node with echo
    workers and the hello-pack example installed (the reference's dev mode
    brings up the compose stack; one process replaces it here)."""
    import threading
    import time as _time

    from ..cli.serve import serve

    def install_example():
        _time.sleep(1.5)
        try:
            repo = Path(__file__).resolve().parent.parent.parent
            pack_dir = repo / "examples" / "hello-pack"
            if pack_dir.is_dir():
                blob = build_pack_archive(pack_dir)
                from ..sdk.client import Client

                Client(base_url=f"http://127.0.0.1:{args.port}",
                       role="admin", principal_id="dev").install_pack(blob)
                print("hello-pack installed")
        except Exception as e:
            print(f"(dev) example pack install skipped: {e}", file=sys.stderr)

    threading.Thread(target=install_example, daemon=True).start()
    print("Cordum stack started (dev mode).")
    print(f"Gateway: http://127.0.0.1:{args.port}")
    serve(port=args.port, workers=2)


def cmd_up(args):
    """`cordumctl up` — the production-shaped single-node launch: durable
    WAL/checkpoints under --state-dir, echo workers on configured pools."""
    from ..cli.serve import serve

    Path(args.state_dir).mkdir(parents=True, exist_ok=True)
    print("Cordum stack started.")
    print(f"Gateway: http://127.0.0.1:{args.port}")
    print(f"State: {args.state_dir}")
    serve(port=args.port, workers=2, checkpoint_dir=args.state_dir)


def main(argv=None) -> int:
    ap = argparse.ArgumentParser(prog="cordumctl")
    ap.add_argument("--server", default=os.environ.get("CORDUM_SERVER", "http://127.0.0.1:8080"))
    ap.add_argument("--api-key", default="")
    sub = ap.add_subparsers(dest="cmd", required=True)

    sub.add_parser("status").set_defaults(fn=cmd_status)

    p = sub.add_parser("serve", help="run the single-process control-plane node")
    p.add_argument("--dashboard-dir", default="",
                   help="serve a dashboard build statically at /")
    p.add_argument("--bridge-port", type=int, default=-1,
                   help="TCP bus-bridge port for external CAP workers "
                        "(0 = auto, -1 = disabled)")
    p.add_argument("--host", default="127.0.0.1")
    p.add_argument("--port", type=int, default=8080)
    p.add_argument("--config-dir", default="")
    p.add_argument("--workers", type=int, default=2)
    p.add_argument("--checkpoint-dir", default="")
    p.add_argument("--checkpoint-interval", type=float, default=30.0)
    p.set_defaults(fn=cmd_serve)

    p = sub.add_parser("init")
    p.add_argument("dir", nargs="?", default=".")
    p.set_defaults(fn=cmd_init)

    p = sub.add_parser("workflow")
    p.add_argument("action", choices=["create", "list", "get", "delete"])
    p.add_argument("id", nargs="?")
    p.add_argument("--file", "-f")
    p.set_defaults(fn=cmd_workflow)

    p = sub.add_parser("run")
    p.add_argument("action", choices=["start", "get", "list", "timeline", "delete", "rerun", "step"])
    p.add_argument("id", nargs="?")
    p.add_argument("--workflow", "-w")
    p.add_argument("--input", "-i")
    p.add_argument("--step")
    p.add_argument("--reject", action="store_true")
    p.add_argument("--dry-run", action="store_true")
    p.set_defaults(fn=cmd_run)

    p = sub.add_parser("approval")
    p.add_argument("action", choices=["list", "approve", "reject"])
    p.add_argument("job_id", nargs="?")
    p.add_argument("--reason")
    p.set_defaults(fn=cmd_approval)

    p = sub.add_parser("dlq")
    p.add_argument("action", choices=["list", "retry", "delete"])
    p.add_argument("job_id", nargs="?")
    p.set_defaults(fn=cmd_dlq)

    p = sub.add_parser("job")
    p.add_argument("action", choices=["submit", "status", "cancel", "logs"])
    p.add_argument("id", nargs="?")
    p.add_argument("--prompt", default="")
    p.add_argument("--topic", default="job.default")
    p.set_defaults(fn=cmd_job)

    p = sub.add_parser("pack")
    p.add_argument("action", choices=["scaffold", "create", "install", "uninstall", "list", "show", "verify"])
    p.add_argument("id", nargs="?")
    p.add_argument("--dir", default=".")
    p.add_argument("--file", "-f")
    p.add_argument("--output", "-o", default="pack.tgz")
    p.add_argument("--force", action="store_true")
    p.set_defaults(fn=cmd_pack)

    p = sub.add_parser("dev", help="run a local dev stack (serve + example pack)")
    p.add_argument("--port", type=int, default=8081)
    p.set_defaults(fn=cmd_dev)

    p = sub.add_parser("up", help="run the production-shaped single-node stack "
                                  "(serve + durable checkpoints)")
    p.add_argument("--port", type=int, default=8081)
    p.add_argument("--state-dir", default=str(Path.home() / ".cordum" / "state"))
    p.set_defaults(fn=cmd_up)

    args = ap.parse_args(argv)
    try:
        args.fn(args)
        return 0
    except Exception as e:  # surface API errors as exit codes
        print(f"error: {e}", file=sys.stderr)
        return 1


if __name__ == "__main__":
    sys.exit(main())
