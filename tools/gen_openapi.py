#!/usr/bin/env python3
"""Dump the gateway's OpenAPI spec (replaces tools/scripts/gen_openapi.sh)."""
import json
import sys
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))

from cordum_amd.gateway import create_app
from cordum_amd.runtime.node import Node
from cordum_amd.utils.clock import ManualClock


def main():
    node = Node(clock=ManualClock()).start()
    app = create_app(node)
    spec = app.openapi()
    out = Path(__file__).resolve().parent.parent / "docs" / "api" / "openapi.json"
    out.parent.mkdir(parents=True, exist_ok=True)
    out.write_text(json.dumps(spec, indent=2))
    print(f"wrote {out} ({len(spec.get('paths', {}))} paths)")


if __name__ == "__main__":
    main()
