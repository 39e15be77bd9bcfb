// Random number algorithms for offset generation and buffer fill.
//
// Behavior parity with the reference's RandAlgo family
// (/root/reference/source/toolkits/random/*.h — user-visible algo names
// "fast" / "balanced" / "balanced_single" / "strong"); implementations are
// independent, straight from the public xoshiro256++/xoshiro256** and
// splitmix64 algorithm definitions.

#pragma once

#include <cstdint>
#include <cstring>
#include <random>
#include <stdexcept>
#include <string>

namespace eb {

inline uint64_t splitmix64(uint64_t& state)
{
    uint64_t z = (state += 0x9E3779B97F4A7C15ULL);
    z = (z ^ (z >> 30)) * 0xBF58476D1CE4E5B9ULL;
    z = (z ^ (z >> 27)) * 0x94D049BB133111EBULL;
    return z ^ (z >> 31);
}

inline uint64_t rotl64(uint64_t x, int k) { return (x << k) | (x >> (64 - k)); }

// Abstract RNG used by offset generators and CPU buffer fill.
class RandAlgo {
public:
    virtual ~RandAlgo() = default;
    virtual uint64_t next() = 0;

    // Fill an arbitrary-size buffer with random bytes.
    virtual void fillBuf(char* buf, uint64_t len)
    {
        while (len >= sizeof(uint64_t)) {
            uint64_t v = next();
            std::memcpy(buf, &v, sizeof(v));
            buf += sizeof(v);
            len -= sizeof(v);
        }
        if (len) {
            uint64_t v = next();
            std::memcpy(buf, &v, len);
        }
    }
};

// "fast": golden-prime multiplicative sequence — cheapest per value.
class RandAlgoGoldenPrime final : public RandAlgo {
public:
    explicit RandAlgoGoldenPrime(uint64_t seed) { state = seed ? seed : 0x9E3779B97F4A7C15ULL; }

    uint64_t next() override
    {
        state = state * 0x9E3779B97F4A7C15ULL + 0xD1B54A32D192ED03ULL;
        return state ^ (state >> 29);
    }

    // tight non-virtual fill loop (the per-block variance refill is on the
    // CPU write hot path; one virtual call per 8 bytes costs GB/s)
    void fillBuf(char* buf, uint64_t len) override
    {
        uint64_t s = state;
        while (len >= 32) { // 4-way unroll, dependency chain stays serial
            uint64_t v0 = s * 0x9E3779B97F4A7C15ULL + 0xD1B54A32D192ED03ULL;
            uint64_t v1 = v0 * 0x9E3779B97F4A7C15ULL + 0xD1B54A32D192ED03ULL;
            uint64_t v2 = v1 * 0x9E3779B97F4A7C15ULL + 0xD1B54A32D192ED03ULL;
            uint64_t v3 = v2 * 0x9E3779B97F4A7C15ULL + 0xD1B54A32D192ED03ULL;
            uint64_t o0 = v0 ^ (v0 >> 29), o1 = v1 ^ (v1 >> 29);
            uint64_t o2 = v2 ^ (v2 >> 29), o3 = v3 ^ (v3 >> 29);
            std::memcpy(buf, &o0, 8);
            std::memcpy(buf + 8, &o1, 8);
            std::memcpy(buf + 16, &o2, 8);
            std::memcpy(buf + 24, &o3, 8);
            s = v3;
            buf += 32;
            len -= 32;
        }
        state = s;
        while (len) {
            uint64_t v = next();
            uint64_t n = len < 8 ? len : 8;
            std::memcpy(buf, &v, n);
            buf += n;
            len -= n;
        }
    }

private:
    uint64_t state;
};

// "balanced_single": xoshiro256** — good quality, still fast.
class RandAlgoXoshiro256ss final : public RandAlgo {
public:
    explicit RandAlgoXoshiro256ss(uint64_t seed)
    {
        uint64_t sm = seed;
        for (auto& w : s) w = splitmix64(sm);
    }

    uint64_t next() override
    {
        const uint64_t result = rotl64(s[1] * 5, 7) * 9;
        const uint64_t t = s[1] << 17;
        s[2] ^= s[0];
        s[3] ^= s[1];
        s[1] ^= s[2];
        s[0] ^= s[3];
        s[2] ^= t;
        s[3] = rotl64(s[3], 45);
        return result;
    }

    void fillBuf(char* buf, uint64_t len) override
    {
        uint64_t a = s[0], b = s[1], c = s[2], d = s[3];
        while (len >= 8) {
            const uint64_t result = rotl64(b * 5, 7) * 9;
            const uint64_t t = b << 17;
            c ^= a;
            d ^= b;
            b ^= c;
            a ^= d;
            c ^= t;
            d = rotl64(d, 45);
            std::memcpy(buf, &result, 8);
            buf += 8;
            len -= 8;
        }
        s[0] = a; s[1] = b; s[2] = c; s[3] = d;
        if (len) {
            uint64_t v = next();
            std::memcpy(buf, &v, len);
        }
    }

private:
    uint64_t s[4];
};

// "balanced": N-way interleaved xoshiro256++ streams for buffer fill —
// independent streams let the compiler keep N states in registers and
// auto-vectorize the fill loop (same idea as the reference's SIMD variant;
// fresh implementation). The GPU fill kernel in gpu_kernels.hip uses the
// same per-stream step function, so CPU and GPU fills are testable against
// each other.
template <int N>
class RandAlgoXoshiro256ppSIMD final : public RandAlgo {
public:
    explicit RandAlgoXoshiro256ppSIMD(uint64_t seed)
    {
        uint64_t sm = seed;
        for (int i = 0; i < N; i++)
            for (int j = 0; j < 4; j++) s[j][i] = splitmix64(sm);
    }

    static inline uint64_t stepOne(uint64_t st[4])
    {
        const uint64_t result = rotl64(st[0] + st[3], 23) + st[0];
        const uint64_t t = st[1] << 17;
        st[2] ^= st[0];
        st[3] ^= st[1];
        st[1] ^= st[2];
        st[0] ^= st[3];
        st[2] ^= t;
        st[3] = rotl64(st[3], 45);
        return result;
    }

    uint64_t next() override
    {
        // scalar path: advance lane 0 only
        uint64_t lane[4] = {s[0][0], s[1][0], s[2][0], s[3][0]};
        uint64_t r = stepOne(lane);
        s[0][0] = lane[0]; s[1][0] = lane[1]; s[2][0] = lane[2]; s[3][0] = lane[3];
        return r;
    }

    void fillBuf(char* buf, uint64_t len) override
    {
        uint64_t chunk[N];
        while (len >= sizeof(chunk)) {
            for (int i = 0; i < N; i++) { // independent streams -> vectorizable
                const uint64_t result = rotl64(s[0][i] + s[3][i], 23) + s[0][i];
                const uint64_t t = s[1][i] << 17;
                s[2][i] ^= s[0][i];
                s[3][i] ^= s[1][i];
                s[1][i] ^= s[2][i];
                s[0][i] ^= s[3][i];
                s[2][i] ^= t;
                s[3][i] = rotl64(s[3][i], 45);
                chunk[i] = result;
            }
            std::memcpy(buf, chunk, sizeof(chunk));
            buf += sizeof(chunk);
            len -= sizeof(chunk);
        }
        while (len) {
            uint64_t v = next();
            uint64_t n = len < 8 ? len : 8;
            std::memcpy(buf, &v, n);
            buf += n;
            len -= n;
        }
    }

private:
    uint64_t s[4][N]; // struct-of-arrays for vectorization
};

// "strong": MT19937-64 via libstdc++.
class RandAlgoMT19937 final : public RandAlgo {
public:
    explicit RandAlgoMT19937(uint64_t seed) : gen(seed) {}
    uint64_t next() override { return gen(); }

private:
    std::mt19937_64 gen;
};

inline RandAlgo* makeRandAlgo(const std::string& name, uint64_t seed)
{
    if (name == "fast") return new RandAlgoGoldenPrime(seed);
    if (name == "balanced_single") return new RandAlgoXoshiro256ss(seed);
    if (name == "balanced") return new RandAlgoXoshiro256ppSIMD<8>(seed);
    if (name == "strong") return new RandAlgoMT19937(seed);
    throw std::runtime_error("unknown random algorithm: " + name);
}

// Bounded-range helper (reference analogue: RandAlgoRange).
class RandRange {
public:
    RandRange(RandAlgo& algo, uint64_t min, uint64_t max) : algo(algo), min(min), span(max - min + 1) {}
    uint64_t next() { return min + (algo.next() % span); }

private:
    RandAlgo& algo;
    uint64_t min;
    uint64_t span;
};

} // namespace eb
