"""In-process Rich TUI for dnet-api / dnet-shard serve loops.

Reference counterpart: src/dnet/tui.py (DnetTUI: log panel + status/model/
memory panels refreshed live). Optional — enabled with --tui on the CLIs.
"""
from __future__ import annotations

import logging
import time
from collections import deque
from typing import Callable, Optional

try:
    from rich.console import Console, Group
    from rich.layout import Layout
    from rich.live import Live
    from rich.panel import Panel
    from rich.table import Table
    from rich.text import Text
    HAS_RICH = True
except ImportError:  # pragma: no cover
    HAS_RICH = False


class TuiLogHandler(logging.Handler):
    def __init__(self, buf: deque):
        super().__init__()
        self.buf = buf

    def emit(self, record):
        self.buf.append(self.format(record))


class DnetTUI:
    """Live status panel; ``status_fn`` returns a dict of fields to show."""

    def __init__(self, role: str, status_fn: Callable[[], dict]):
        assert HAS_RICH, "rich not available"
        self.role = role
        self.status_fn = status_fn
        self.logs: deque = deque(maxlen=200)
        handler = TuiLogHandler(self.logs)
        handler.setFormatter(logging.Formatter("%(asctime)s %(message)s",
                                               datefmt="%H:%M:%S"))
        logging.getLogger("dnet").addHandler(handler)
        self._live: Optional[Live] = None

    def _render(self):
        status = self.status_fn()
        table = Table.grid(padding=(0, 2))
        for k, v in status.items():
            table.add_row(Text(str(k), style="bold cyan"), str(v))
        mem = self._memory_line()
        panels = Group(
            Panel(table, title=f"dnet_amd {self.role}", border_style="green"),
            Panel(Text("\n".join(list(self.logs)[-18:])), title="log",
                  border_style="blue"),
            Text(mem, style="dim"),
        )
        return panels

    def _memory_line(self) -> str:
        parts = []
        try:
            import psutil
            vm = psutil.virtual_memory()
            parts.append(f"RAM {vm.used / 1e9:.1f}/{vm.total / 1e9:.0f} GB")
        except ImportError:
            pass
        try:
            import torch
            if torch.cuda.is_available():
                free, total = torch.cuda.mem_get_info()
                parts.append(f"HBM {(total - free) / 1e9:.1f}/"
                             f"{total / 1e9:.0f} GB")
        except Exception:
            pass
        return "  ".join(parts)

    def run_forever(self, tick: float = 0.5):
        with Live(self._render(), refresh_per_second=2) as live:
            self._live = live
            while True:
                time.sleep(tick)
                live.update(self._render())
