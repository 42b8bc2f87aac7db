"""Windowed metric smoothing (capability parity with the reference's
SmoothedValue, /root/reference/utils.py:60-102, itself adapted from
facebookresearch/mmf).  Re-implemented without numpy: the window is tiny
(default 20, the trainer uses 5) so plain python is faster than round-
tripping through arrays."""

from collections import deque


class SmoothedValue:
    """Track a series of (value, batch_size) updates and expose smoothed
    views over a fixed window plus the global series average.

    API kept identical to the reference (median / avg / global_avg
    properties, get_latest, update(value, batch_size)) so the trainer
    and any downstream log parsing are drop-in compatible.
    """

    def __init__(self, window_size=20):
        self.window_size = window_size
        self.reset()

    def reset(self):
        self._weighted = deque(maxlen=self.window_size)
        self._values = deque(maxlen=self.window_size)
        self._batch_sizes = deque(maxlen=self.window_size)
        self.total_samples = 0
        self.total = 0.0
        self.count = 0

    def update(self, value, batch_size=1):
        self._weighted.append(value * batch_size)
        self._values.append(value)
        self._batch_sizes.append(batch_size)
        self.count += 1
        self.total_samples += batch_size
        self.total += value * batch_size

    @property
    def median(self):
        vals = sorted(self._values)
        n = len(vals)
        if n == 0:
            return float("nan")
        mid = n // 2
        if n % 2 == 1:
            return vals[mid]
        return 0.5 * (vals[mid - 1] + vals[mid])

    @property
    def avg(self):
        denom = sum(self._batch_sizes)
        if denom == 0:
            return float("nan")
        return sum(self._weighted) / denom

    @property
    def global_avg(self):
        if self.total_samples == 0:
            return float("nan")
        return self.total / self.total_samples

    def get_latest(self):
        return self._values[-1]
