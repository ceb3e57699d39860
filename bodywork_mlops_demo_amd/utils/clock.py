"""Virtual pipeline clock.

The reference is wall-clock-coupled: every stage stamps artefacts with
``date.today()`` (``stage_1_train_model.py:86``, ``stage_3:35,48``), which
caps the pipeline at one cycle per real day and makes the concept-drift
loop untestable.  This framework parameterises the date: every stage takes
a ``date`` argument defaulting to the virtual clock, so a 30-"day" drift
loop runs in seconds and golden-value tests are deterministic.
"""
from __future__ import annotations

import os
from datetime import date, timedelta


class VirtualClock:
    """A settable, advanceable pipeline date.

    The starting date can be pinned with the ``BODYWORK_AMD_DATE``
    environment variable (``YYYY-MM-DD``); otherwise it is today.
    """

    def __init__(self, start: date | str | None = None):
        if start is None:
            start = os.environ.get("BODYWORK_AMD_DATE") or date.today()
        if isinstance(start, str):
            start = date.fromisoformat(start)
        self._today = start

    def today(self) -> date:
        return self._today

    def advance(self, days: int = 1) -> date:
        self._today += timedelta(days=days)
        return self._today

    def set(self, d: date | str) -> None:
        self._today = date.fromisoformat(d) if isinstance(d, str) else d

    def day_of_year(self) -> int:
        return self._today.timetuple().tm_yday


#: process-global default clock used by stages when no date is passed
CLOCK = VirtualClock()
