"""Health pulse ticker (reference: cmd/k8s-device-plugin/main.go:129-137)."""

from __future__ import annotations

import threading
from typing import Callable, List


class HeartbeatTicker:
    """Fires registered callbacks every `pulse` seconds until stopped."""

    def __init__(self, pulse_seconds: float):
        self.pulse = pulse_seconds
        self._callbacks: List[Callable[[], None]] = []
        self._stop = threading.Event()
        self._thread: threading.Thread | None = None

    def subscribe(self, cb: Callable[[], None]) -> None:
        self._callbacks.append(cb)

    def start(self) -> None:
        if self.pulse <= 0 or self._thread is not None:
            return
        self._thread = threading.Thread(target=self._run, daemon=True,
                                        name="health-pulse")
        self._thread.start()

    def _run(self) -> None:
        while not self._stop.wait(self.pulse):
            for cb in list(self._callbacks):
                try:
                    cb()
                except Exception:  # never let one bad beat kill the ticker
                    import logging
                    logging.getLogger(__name__).exception("heartbeat callback failed")

    def stop(self) -> None:
        self._stop.set()
        if self._thread is not None:
            self._thread.join(timeout=2)
            self._thread = None
