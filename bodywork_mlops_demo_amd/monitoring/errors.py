"""Error monitoring — the Sentry capability (reference C8).

The reference initialises ``sentry_sdk`` in every stage with
``traces_sample_rate=1.0``, tags the event stream with the stage name, and
wraps ``main()`` in try/except → log → ``sys.exit(1)`` (reference
``stage_1_train_model.py:161-178`` and copies).

Here:
- if ``SENTRY_DSN`` is set *and* sentry_sdk is importable, events go to
  Sentry exactly as the reference does;
- otherwise events are captured to a durable local event log
  (JSON-lines under ``error-events/`` in the artefact store, or a file),
  so the capability — durable, tagged error records with tracebacks —
  exists without the SaaS dependency;
- :func:`stage_guard` is the single shared top-level exception guard the
  reference copy-pastes four times.
"""
from __future__ import annotations

import json
import os
import sys
import time
import traceback
from contextlib import contextmanager
from typing import Any

from bodywork_mlops_demo_amd.utils.logging import configure_logger

log = configure_logger(__name__)


class ErrorMonitor:
    """Tagged error/event sink with an optional Sentry backend."""

    def __init__(
        self,
        dsn: str | None = None,
        event_log_path: str | None = None,
        traces_sample_rate: float = 1.0,
    ):
        self.tags: dict[str, str] = {}
        self._sentry = None
        dsn = dsn if dsn is not None else os.environ.get("SENTRY_DSN")
        if dsn:
            try:
                import sentry_sdk  # type: ignore

                sentry_sdk.init(dsn, traces_sample_rate=traces_sample_rate)
                self._sentry = sentry_sdk
            except ImportError:
                log.warning("SENTRY_DSN set but sentry_sdk not installed; "
                            "falling back to local event log")
        self.event_log_path = event_log_path or os.environ.get(
            "BODYWORK_AMD_EVENT_LOG", "artefact-store/error-events/events.jsonl"
        )

    def set_tag(self, key: str, value: str) -> None:
        self.tags[key] = value
        if self._sentry:
            self._sentry.set_tag(key, value)

    def capture_exception(self, exc: BaseException) -> None:
        if self._sentry:
            self._sentry.capture_exception(exc)
        self._append_event(
            kind="exception",
            type=type(exc).__name__,
            message=str(exc),
            traceback=traceback.format_exc(),
        )

    def capture_message(self, message: str, level: str = "info") -> None:
        if self._sentry:
            self._sentry.capture_message(message, level=level)
        self._append_event(kind="message", level=level, message=message)

    def _append_event(self, **fields: Any) -> None:
        event = {"ts": time.time(), "tags": dict(self.tags), **fields}
        try:
            path = self.event_log_path
            os.makedirs(os.path.dirname(path) or ".", exist_ok=True)
            with open(path, "a") as f:
                f.write(json.dumps(event) + "\n")
        except OSError as e:  # event logging must never take a stage down
            log.warning(f"could not append error event: {e}")


_MONITOR: ErrorMonitor | None = None


def get_error_monitor() -> ErrorMonitor:
    global _MONITOR
    if _MONITOR is None:
        _MONITOR = ErrorMonitor()
    return _MONITOR


@contextmanager
def stage_guard(stage_name: str, exit_on_error: bool = False):
    """Top-level stage exception guard (reference ``stage_1:170-178``).

    Tags the event stream with the stage name, captures any exception to
    the monitor, logs it, and either re-raises (default — the pipeline
    runner handles retries) or exits 1 (standalone CLI parity with the
    reference's ``sys.exit(1)``).
    """
    mon = get_error_monitor()
    mon.set_tag("stage", stage_name)
    try:
        yield mon
    except Exception as e:
        log.error(e)
        mon.capture_exception(e)
        if exit_on_error:
            sys.exit(1)
        raise
