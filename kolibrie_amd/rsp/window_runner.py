"""WindowRunner — push/drain wrapper over CSPARQLWindow
(ref: kolibrie/src/rsp/window_runner.rs:37-106)."""
from __future__ import annotations

from typing import Hashable, List

from .s2r import CSPARQLWindow, ContentContainer


class WindowRunner:
    def __init__(self, window: CSPARQLWindow):
        self.window = window
        self._queue = window.register()

    def push(self, item: Hashable, ts: int):
        self.window.add_to_window(item, ts)

    def drain(self) -> List[ContentContainer]:
        out = []
        while not self._queue.empty():
            out.append(self._queue.get())
        return out

    def flush(self):
        self.window.flush()
