"""dnet-api entry point (reference: src/cli/api.py).

Usage:
  python -m dnet_amd.cli.api --hostfile hosts.txt [--host 0.0.0.0]
      [--port 8080] [--wire-port 50051]
"""
from __future__ import annotations

import argparse
import asyncio

from ..config import get_settings
from ..utils.hostfile import StaticDiscovery
from ..utils.logger import get_logger


def main(argv=None):
    ap = argparse.ArgumentParser("dnet-api")
    s = get_settings()
    ap.add_argument("--hostfile", default=None,
                    help="static discovery hostfile; omit for UDP discovery")
    ap.add_argument("--discovery", choices=["hostfile", "udp"], default=None,
                    help="default: hostfile if --hostfile given, else udp")
    ap.add_argument("--discovery-port", type=int, default=52525)
    ap.add_argument("--host", default=s.api.host)
    ap.add_argument("--port", type=int, default=s.api.port)
    ap.add_argument("--wire-port", type=int, default=s.api.grpc_port)
    ap.add_argument("--callback-addr", default=s.api.callback_addr)
    ap.add_argument("--tui", action="store_true")
    args = ap.parse_args(argv)
    if args.callback_addr:
        s.api.callback_addr = args.callback_addr
    s.api.grpc_port = args.wire_port
    asyncio.run(serve(args))


async def serve(args):
    import uvicorn

    from ..api.cluster import ClusterManager
    from ..api.server import ApiState, api_wire_handler, build_api_app
    from ..protos.wire import WireServer

    log = get_logger("api")
    s = get_settings()
    mode = args.discovery or ("hostfile" if args.hostfile else "udp")
    if mode == "hostfile":
        assert args.hostfile, "--hostfile required for hostfile discovery"
        discovery = StaticDiscovery(args.hostfile, own_instance="api",
                                    own_is_manager=True)
    else:
        from ..discovery import UdpDiscovery
        discovery = UdpDiscovery("api", args.port, args.wire_port,
                                 is_manager=True, port=args.discovery_port)
    await discovery.async_start()
    cluster = ClusterManager(discovery)
    state = ApiState(cluster, s)
    app = build_api_app(state)
    handler = await api_wire_handler(state)
    wire = WireServer(args.host, args.wire_port, handler)
    await wire.start()
    log.info("dnet-api on http://%s:%d (wire %d)", args.host, args.port,
             args.wire_port)
    if getattr(args, "tui", False):
        import threading

        from ..tui import DnetTUI, HAS_RICH
        if HAS_RICH:
            tui = DnetTUI("api", lambda: {
                "model": state.models.loaded_model or "-",
                "devices": len(cluster.devices),
                "pending": len(state.inference.pending)})
            threading.Thread(target=tui.run_forever, daemon=True,
                             name="tui").start()
    config = uvicorn.Config(app, host=args.host, port=args.port,
                            log_level="warning")
    await uvicorn.Server(config).serve()


if __name__ == "__main__":
    main()
