"""Node server entrypoint: assemble the single-process control plane from
config files and serve the gateway (replaces the reference's five service
mains + docker-compose)."""
from __future__ import annotations

import threading
import time
from pathlib import Path


def _cuda() -> bool:
    try:
        import torch

        return torch.cuda.is_available()
    except ImportError:
        return False


def build_node(config_dir: str = "", checkpoint_dir: str = ""):
    from ..config import (
        NodeConfig,
        load_pools,
        load_safety_yaml,
        load_timeouts,
        seed_system_config,
    )
    from ..runtime.node import Node

    cfg = NodeConfig.from_env(Path(config_dir) if config_dir else None)
    routing = load_pools(cfg.pool_config_path)
    timeouts = load_timeouts(cfg.timeout_config_path)
    policy_yaml = load_safety_yaml(cfg.safety_policy_path)
    # dispatch plane: on a GPU host the batched K1/K2 device engine is the
    # default (the north star); CORDUM_DISPATCH=host|device overrides
    import os as _os

    dispatch = _os.environ.get("CORDUM_DISPATCH", "")
    if not dispatch:
        try:
            import torch

            from ..utils.threads import cap_torch_threads

            cap_torch_threads()
            dispatch = "device" if torch.cuda.is_available() else "host"
        except ImportError:
            dispatch = "host"
    node = Node(routing=routing, policy_yaml=policy_yaml,
                safety_cache_ttl_s=cfg.safety_decision_cache_ttl_s,
                dispatch=dispatch,
                device="cuda:0" if dispatch == "device" and _cuda() else "cpu",
                backend="ext" if dispatch == "device" and _cuda() else "ref").start()
    node.scheduler_reconciler.update_timeouts(timeouts.dispatch_timeout_s, timeouts.running_timeout_s)
    seed_system_config(node.configsvc, cfg.system_config_path)

    checkpointer = None
    if checkpoint_dir:
        from ..store.wal import Checkpointer

        checkpointer = Checkpointer(node, checkpoint_dir)
        if checkpointer.restore():
            checkpointer.replay_wal()
        node.wal = checkpointer
    return node, cfg, checkpointer


def serve(host: str = "127.0.0.1", port: int = 8080, config_dir: str = "",
          workers: int = 2, checkpoint_dir: str = "", checkpoint_interval_s: float = 30.0,
          dashboard_dir: str = "", bridge_port: int = -1):
    import uvicorn

    from ..gateway import create_app
    from ..runtime.worker import echo_handler

    node, cfg, checkpointer = build_node(config_dir, checkpoint_dir)
    topics = sorted(node.strategy.current_routing().topics)
    if node.dispatch_mode == "device":
        node.add_device_worker_pool(n_workers=max(4, workers), topics=topics)
    else:
        for i in range(workers):
            node.add_worker(f"worker-{i}", handler=echo_handler, topics=topics)

    app = create_app(node, rate_limit_rps=cfg.api_rate_limit_rps,
                     rate_limit_burst=cfg.api_rate_limit_burst,
                     dashboard_dir=dashboard_dir)

    # TCP bus bridge for external CAP workers (the NATS attach seam;
    # --bridge-port 0 picks a free port, -1 disables)
    import os as _os2

    if bridge_port < 0:
        bridge_port = int(_os2.environ.get("CORDUM_BRIDGE_PORT", "-1"))
    bridge = None
    if bridge_port >= 0:
        from ..bus.tcp_bridge import BusBridgeServer

        bridge = BusBridgeServer(node, host=host, port=bridge_port).start()
        print(f"bus bridge listening on {host}:{bridge.port}", flush=True)

    stop = threading.Event()

    def control_loop():
        last_reconcile = 0.0
        last_checkpoint = 0.0
        while not stop.is_set():
            node.tick()
            now = time.time()
            if now - last_reconcile > 5.0:
                node.reconcile()
                last_reconcile = now
            if checkpointer is not None and now - last_checkpoint > checkpoint_interval_s:
                checkpointer.checkpoint()
                last_checkpoint = now
            time.sleep(0.05)

    t = threading.Thread(target=control_loop, daemon=True)
    t.start()
    try:
        uvicorn.run(app, host=host, port=port, log_level="warning")
    finally:
        stop.set()
        if bridge is not None:
            bridge.stop()
