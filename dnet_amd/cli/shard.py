"""dnet-shard entry point (reference: src/cli/shard.py).

Usage:
  python -m dnet_amd.cli.shard --name shard0 [--host 0.0.0.0]
      [--http-port 8081] [--wire-port 50052]

The driver loop (model + process group owner) runs on the main thread;
HTTP control + wire data servers run on a background thread.
"""
from __future__ import annotations

import argparse
import faulthandler
import signal

from ..config import get_settings
from ..utils.logger import get_logger


def main(argv=None):
    ap = argparse.ArgumentParser("dnet-shard")
    s = get_settings()
    ap.add_argument("--name", default="shard0")
    ap.add_argument("--host", default=s.shard.host)
    ap.add_argument("--http-port", type=int, default=s.shard.http_port)
    ap.add_argument("--wire-port", type=int, default=s.shard.grpc_port)
    ap.add_argument("--tui", action="store_true")
    ap.add_argument("--announce", action="store_true",
                    help="announce this shard over UDP multicast discovery")
    ap.add_argument("--discovery-port", type=int, default=52525)
    ap.add_argument("--gpu-index", type=int, default=0)
    args = ap.parse_args(argv)

    from ..shard.runtime import ShardRuntime
    from ..shard.server import start_servers

    log = get_logger("shard")
    faulthandler.register(signal.SIGUSR1)
    rt = ShardRuntime(instance=args.name)
    start_servers(rt, args.host, args.http_port, args.wire_port)
    if args.announce:
        from ..discovery import _load_ext
        p2p = _load_ext().P2PInstance(
            instance=args.name, http_port=args.http_port,
            shard_port=args.wire_port, is_manager=False,
            gpu_index=args.gpu_index, port=args.discovery_port)
        p2p.start()
        log.info("announcing %s on UDP discovery port %d", args.name,
                 args.discovery_port)
    log.info("dnet-shard %s on http://%s:%d (wire %d)", args.name, args.host,
             args.http_port, args.wire_port)
    if args.tui:
        import threading

        from ..tui import DnetTUI, HAS_RICH
        if HAS_RICH:
            tui = DnetTUI("shard", lambda: {
                "instance": rt.instance, "status": rt.status,
                "model": rt.model_name or "-",
                "queue": rt.infer_q.qsize(), "error": rt.last_error or "-"})
            threading.Thread(target=tui.run_forever, daemon=True,
                             name="tui").start()
    try:
        rt.run()
    except KeyboardInterrupt:
        pass


if __name__ == "__main__":
    main()
