#!/usr/bin/env python3
"""External echo worker (the reference's examples/python-worker analog):
runs in its OWN process and attaches to a running node over the TCP bus
bridge, speaking CAP v2 BusPacket frames on the wire.

Start the node with a bridge:   cordumctl serve --bridge-port 4230
Then:                           python examples/python-worker/worker.py

Falls back to the in-process attach when no bridge is reachable (handy for
a single-file demo)."""
import json
import os
import sys
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parent.parent.parent))


def echo(req, ctx_blob):
    payload = {}
    if ctx_blob:
        try:
            payload = json.loads(ctx_blob)
        except ValueError:
            payload = {"raw": ctx_blob.decode("utf-8", "replace")}
    return json.dumps({"echo": payload.get("prompt") or payload,
                       "worker": "python-echo-1"}).encode()


def main():
    bridge_port = int(os.environ.get("CORDUM_BRIDGE_PORT", "4230"))
    api_base = os.environ.get("CORDUM_API", "http://127.0.0.1:8081")
    topics = (os.environ.get("CORDUM_TOPICS") or "job.default,job.echo").split(",")
    try:
        from cordum_amd.sdk.remote_worker import RemoteWorker

        w = RemoteWorker("python-echo-1", handler=echo, topics=topics,
                         bridge_port=bridge_port, api_base=api_base)
        print(f"external worker attached via bridge :{bridge_port}, "
              f"topics={topics}", flush=True)
        w.run_forever()
    except ConnectionRefusedError:
        print("no bridge reachable; attaching in-process instead", flush=True)
        import time

        from cordum_amd.runtime.node import Node
        from cordum_amd.runtime.worker import echo_handler

        node = Node().start()
        node.add_worker("python-echo-1", handler=echo_handler, topics=topics)
        while True:
            node.tick()
            time.sleep(0.1)


if __name__ == "__main__":
    main()
