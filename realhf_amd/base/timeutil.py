"""Frequency control + timers (reference: realhf/base/timeutil.py:98 EpochStepTimeFreqCtl)."""
import dataclasses
import time
from typing import Optional


@dataclasses.dataclass
class FrequencyControl:
    """Triggers on epoch / step / wall-clock frequency, whichever fires."""

    freq_epoch: Optional[int] = None
    freq_step: Optional[int] = None
    freq_sec: Optional[float] = None
    initial_value: bool = False

    def __post_init__(self):
        self._last_epoch = 0
        self._last_step = 0
        self._last_time = time.monotonic()
        self._initial = self.initial_value

    def check(self, epochs: int = 0, steps: int = 0) -> bool:
        self._last_epoch += epochs
        self._last_step += steps
        fire = False
        if self._initial:
            fire, self._initial = True, False
        if self.freq_epoch is not None and self._last_epoch >= self.freq_epoch:
            fire = True
        if self.freq_step is not None and self._last_step >= self.freq_step:
            fire = True
        if (
            self.freq_sec is not None
            and time.monotonic() - self._last_time >= self.freq_sec
        ):
            fire = True
        if fire:
            self._last_epoch = 0
            self._last_step = 0
            self._last_time = time.monotonic()
        return fire


class Timer:
    """Accumulating wall-clock timer."""

    def __init__(self):
        self.total = 0.0
        self.count = 0
        self._t0 = None

    def __enter__(self):
        self._t0 = time.monotonic()
        return self

    def __exit__(self, *exc):
        self.total += time.monotonic() - self._t0
        self.count += 1
        self._t0 = None

    @property
    def mean(self):
        return self.total / max(1, self.count)
