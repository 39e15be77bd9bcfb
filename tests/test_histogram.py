"""Latency histogram: Python/C++ bucket math parity, merge, percentiles."""

from elbencho_amd import histogram as H


def test_bucket_bounds_match_native(core):
    assert core.hist_num_buckets() == H.NUM_BUCKETS
    for i in range(0, 200):
        assert core.hist_bucket_lower_bound(i) == H.bucket_lower_bound(i)


def test_bucket_bounds_monotonic():
    prev = -1
    for i in range(H.NUM_BUCKETS):
        b = H.bucket_lower_bound(i)
        assert b > prev
        prev = b


def test_merge_and_stats():
    a = H.empty()
    a[0], a[1], a[2], a[3] = 2, 30, 10, 20
    a[H.HEADER + 10] = 2
    b = H.empty()
    b[0], b[1], b[2], b[3] = 1, 5, 5, 5
    b[H.HEADER + 5] = 1
    h = H.Histogram(a).merge(b)
    assert h.num_values == 3
    assert h.sum_us == 35
    assert h.min_us == 5
    assert h.max_us == 20
    assert abs(h.avg_us - 35 / 3) < 1e-9


def test_percentiles():
    h = H.Histogram()
    # 100 values in the bucket whose lower bound is 8us, 1 value at 1024us
    i8 = next(i for i in range(H.NUM_BUCKETS) if H.bucket_lower_bound(i) == 8)
    i1024 = next(i for i in range(H.NUM_BUCKETS) if H.bucket_lower_bound(i) == 1024)
    h.vec[0] = 101
    h.vec[1] = 100 * 8 + 1024
    h.vec[2], h.vec[3] = 8, 1024
    h.vec[H.HEADER + i8] = 100
    h.vec[H.HEADER + i1024] = 1
    assert h.percentile(50) == 8
    assert h.percentile(99) == 8
    assert h.percentile(99.9) == 1024


def test_empty_histogram():
    h = H.Histogram()
    assert h.num_values == 0
    assert h.min_us == 0
    assert h.max_us == 0
    assert h.percentile(99) == 0
