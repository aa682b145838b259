"""Retry engine with linear backoff.

Counterpart of the reference's REST retry engine
(pkg/virtual_kubelet/runpod_client.go:268-343): 3 attempts, 500 ms * n linear
backoff, with certain outcomes treated as terminal (the reference treats both
200 and 404 as terminal-valid).
"""

from __future__ import annotations

import time
from typing import Callable, Optional, Tuple, TypeVar

T = TypeVar("T")

DEFAULT_ATTEMPTS = 3
DEFAULT_BACKOFF_S = 0.5


def retry(
    fn: Callable[[], T],
    attempts: int = DEFAULT_ATTEMPTS,
    backoff_s: float = DEFAULT_BACKOFF_S,
    terminal: Optional[Callable[[BaseException], bool]] = None,
    sleep: Callable[[float], None] = time.sleep,
) -> T:
    """Call ``fn`` up to ``attempts`` times with linear backoff.

    ``terminal(exc)`` returning True stops retrying and re-raises immediately
    (e.g. a 404 that the caller maps to NOT_FOUND).
    """
    last: Optional[BaseException] = None
    for attempt in range(attempts):
        try:
            return fn()
        except BaseException as exc:  # noqa: BLE001 - re-raised below
            if terminal is not None and terminal(exc):
                raise
            last = exc
            if attempt + 1 < attempts:
                sleep(backoff_s * (attempt + 1))
    assert last is not None
    raise last


class Ticker:
    """A cancellable periodic ticker (goroutine-ticker analogue).

    The reference's status loop ignores ctx cancellation and leaks on shutdown
    (kubelet.go:296-302); this one stops cleanly via ``stop()``.
    """

    def __init__(self, interval_s: float, fn: Callable[[], None], name: str = "ticker"):
        import threading

        self.interval_s = interval_s
        self.fn = fn
        self.name = name
        self._stop = threading.Event()
        self._thread = threading.Thread(target=self._run, name=name, daemon=True)

    def start(self) -> "Ticker":
        self._thread.start()
        return self

    def _run(self) -> None:
        import logging

        log = logging.getLogger(self.name)
        while not self._stop.wait(self.interval_s):
            try:
                self.fn()
            except Exception:  # noqa: BLE001 - keep ticking
                log.exception("periodic task failed")

    def stop(self, timeout: float = 5.0) -> None:
        self._stop.set()
        if self._thread.is_alive():
            self._thread.join(timeout)

    def trigger_now(self) -> None:
        """Run one iteration synchronously (used by tests and event nudges)."""
        self.fn()


def parse_duration_s(value: object, default: float) -> float:
    """Parse "30s"/"5m"/"1h"/"500ms"/bare numbers; malformed input returns
    ``default`` instead of raising (a flag typo must not crash the kubelet)."""
    if value is None:
        return default
    if isinstance(value, (int, float)):
        return float(value)
    text = str(value).strip()
    try:
        if text.endswith("ms"):
            return float(text[:-2]) / 1000.0
        for suffix, mult in (("s", 1.0), ("m", 60.0), ("h", 3600.0)):
            if text.endswith(suffix):
                return float(text[: -len(suffix)]) * mult
        return float(text)
    except ValueError:
        return default
