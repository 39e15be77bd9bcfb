// Latency histogram with logarithmic quarter-log2 buckets.
//
// Behavior parity with reference LatencyHistogram.{h,cpp} (log2 buckets with
// quarter-step refinement, min/avg/max, percentiles, merge, wire transfer as
// a flat bucket vector). Independent design: bucket index of a microsecond
// value v is 4*floor(log2(v)) + the two bits below the leading bit, so each
// power of two splits into 4 sub-buckets; 256 buckets cover v < 2^64.

#pragma once

#include <cmath>
#include <cstdint>
#include <vector>

namespace eb {

class LatencyHistogram {
public:
    static constexpr int NUM_BUCKETS = 4 * 64; // quarter-log2 over u64 range

    void add(uint64_t microSecs)
    {
        numValues++;
        sumMicroSecs += microSecs;
        if (microSecs < minMicroSecs) minMicroSecs = microSecs;
        if (microSecs > maxMicroSecs) maxMicroSecs = microSecs;
        buckets[bucketIndex(microSecs)]++;
    }

    static int bucketIndex(uint64_t v)
    {
        if (v < 4) return (int)v; // 0,1,2,3 map to buckets 0..3 exactly
        int log2v = 63 - __builtin_clzll(v);
        int frac = (int)((v >> (log2v - 2)) & 3); // two bits below leading bit
        return log2v * 4 + frac - 4; // -4: v in [4,8) starts after exact 0..3
    }

    // Lower bound (µs) of a bucket — inverse of bucketIndex.
    static uint64_t bucketLowerBound(int idx)
    {
        if (idx < 4) return (uint64_t)idx;
        int log2v = (idx + 4) / 4;
        int frac = (idx + 4) % 4;
        return (1ULL << log2v) + ((uint64_t)frac << (log2v - 2));
    }

    void merge(const LatencyHistogram& other)
    {
        numValues += other.numValues;
        sumMicroSecs += other.sumMicroSecs;
        if (other.numValues) {
            if (other.minMicroSecs < minMicroSecs) minMicroSecs = other.minMicroSecs;
            if (other.maxMicroSecs > maxMicroSecs) maxMicroSecs = other.maxMicroSecs;
        }
        for (int i = 0; i < NUM_BUCKETS; i++) buckets[i] += other.buckets[i];
    }

    void reset()
    {
        numValues = 0;
        sumMicroSecs = 0;
        minMicroSecs = UINT64_MAX;
        maxMicroSecs = 0;
        for (auto& b : buckets) b = 0;
    }

    uint64_t getNumValues() const { return numValues; }
    uint64_t getMin() const { return numValues ? minMicroSecs : 0; }
    uint64_t getMax() const { return maxMicroSecs; }
    uint64_t getSum() const { return sumMicroSecs; }
    double getAvg() const { return numValues ? (double)sumMicroSecs / numValues : 0.0; }

    // Percentile estimate: lower bound of the bucket containing the p-quantile.
    uint64_t getPercentile(double p) const
    {
        if (!numValues) return 0;
        uint64_t target = (uint64_t)(p / 100.0 * numValues);
        if (target >= numValues) target = numValues - 1;
        uint64_t cum = 0;
        for (int i = 0; i < NUM_BUCKETS; i++) {
            cum += buckets[i];
            if (cum > target) return bucketLowerBound(i);
        }
        return maxMicroSecs;
    }

    // Wire format: [numValues, sum, min, max, buckets...]; merged across
    // workers/ranks by elementwise sum of buckets (min/max folded on merge).
    std::vector<uint64_t> toVec() const
    {
        std::vector<uint64_t> v;
        v.reserve(4 + NUM_BUCKETS);
        v.push_back(numValues);
        v.push_back(sumMicroSecs);
        v.push_back(numValues ? minMicroSecs : UINT64_MAX);
        v.push_back(maxMicroSecs);
        for (int i = 0; i < NUM_BUCKETS; i++) v.push_back(buckets[i]);
        return v;
    }

    static LatencyHistogram fromVec(const std::vector<uint64_t>& v)
    {
        LatencyHistogram h;
        if (v.size() < 4 + NUM_BUCKETS) return h;
        h.numValues = v[0];
        h.sumMicroSecs = v[1];
        h.minMicroSecs = v[2];
        h.maxMicroSecs = v[3];
        for (int i = 0; i < NUM_BUCKETS; i++) h.buckets[i] = v[4 + i];
        return h;
    }

private:
    uint64_t numValues = 0;
    uint64_t sumMicroSecs = 0;
    uint64_t minMicroSecs = UINT64_MAX;
    uint64_t maxMicroSecs = 0;
    uint64_t buckets[NUM_BUCKETS] = {};
};

} // namespace eb
