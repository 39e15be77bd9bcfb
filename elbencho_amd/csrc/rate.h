// Per-thread byte rate limiter (reference analogue: toolkits/RateLimiter.h).
// Independent implementation: token budget per 1-second window; when the
// window budget is exhausted, sleep to the end of the window.

#pragma once

#include <chrono>
#include <cstdint>
#include <thread>

namespace eb {

class RateLimiter {
public:
    void init(uint64_t bytesPerSec)
    {
        limit = bytesPerSec;
        windowStart = Clock::now();
        spent = 0;
    }

    // Account `bytes` about to be transferred; sleep if over budget.
    void wait(uint64_t bytes)
    {
        if (!limit) return;

        auto now = Clock::now();
        auto elapsed = std::chrono::duration_cast<std::chrono::microseconds>(now - windowStart);

        if (elapsed.count() >= 1000000) { // new window
            windowStart = now;
            spent = 0;
        }

        if (spent >= limit) { // budget exhausted: sleep to end of window
            auto windowEnd = windowStart + std::chrono::seconds(1);
            std::this_thread::sleep_until(windowEnd);
            windowStart = Clock::now();
            spent = 0;
        }

        spent += bytes;
    }

private:
    using Clock = std::chrono::steady_clock;
    uint64_t limit = 0; // 0 = unlimited
    uint64_t spent = 0;
    Clock::time_point windowStart;
};

} // namespace eb
