"""Python-side latency histogram aggregation.

Mirrors the native engine's LatencyHistogram wire format
(csrc/histogram.h: [numValues, sumMicroSecs, min, max, buckets...]) for
merging across workers/services and computing percentiles. Reference
analogue: LatencyHistogram.{h,cpp} merge + percentile logic.
"""

from __future__ import annotations

NUM_BUCKETS = 256  # 4 * 64 quarter-log2 buckets
HEADER = 4  # numValues, sum, min, max


def empty() -> list[int]:
    v = [0] * (HEADER + NUM_BUCKETS)
    v[2] = 2**64 - 1  # min sentinel
    return v


def merge(a: list[int], b: list[int]) -> list[int]:
    out = list(a)
    out[0] += b[0]
    out[1] += b[1]
    if b[0]:
        out[2] = min(out[2], b[2])
        out[3] = max(out[3], b[3])
    for i in range(HEADER, HEADER + NUM_BUCKETS):
        out[i] += b[i]
    return out


def bucket_lower_bound(idx: int) -> int:
    if idx < 4:
        return idx
    log2v = (idx + 4) // 4
    frac = (idx + 4) % 4
    return (1 << log2v) + (frac << (log2v - 2))


class Histogram:
    def __init__(self, vec: list[int] | None = None):
        self.vec = list(vec) if vec else empty()

    def merge(self, other: "Histogram | list[int]") -> "Histogram":
        ov = other.vec if isinstance(other, Histogram) else other
        self.vec = merge(self.vec, ov)
        return self

    @property
    def num_values(self) -> int:
        return self.vec[0]

    @property
    def sum_us(self) -> int:
        return self.vec[1]

    @property
    def min_us(self) -> int:
        return self.vec[2] if self.num_values else 0

    @property
    def max_us(self) -> int:
        return self.vec[3]

    @property
    def avg_us(self) -> float:
        return self.vec[1] / self.vec[0] if self.vec[0] else 0.0

    def percentile(self, p: float) -> int:
        if not self.num_values:
            return 0
        target = int(p / 100.0 * self.num_values)
        target = min(target, self.num_values - 1)
        cum = 0
        for i in range(NUM_BUCKETS):
            cum += self.vec[HEADER + i]
            if cum > target:
                return bucket_lower_bound(i)
        return self.max_us

    def nonzero_buckets(self) -> list[tuple[int, int]]:
        """(bucket lower bound us, count) for buckets with entries."""
        out = []
        for i in range(NUM_BUCKETS):
            c = self.vec[HEADER + i]
            if c:
                out.append((bucket_lower_bound(i), c))
        return out
