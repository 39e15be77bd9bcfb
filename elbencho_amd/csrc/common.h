// elbencho_amd core engine — common types and phase codes.
//
// MI355X-native rebuild of the behavior surface of breuner/elbencho
// (reference: /root/reference/source/Common.h — phase name vocabulary only;
// all code here is an independent implementation).

#pragma once

#include <atomic>
#include <cstdint>
#include <cstring>
#include <stdexcept>
#include <string>
#include <vector>

namespace eb {

// Benchmark phases. Numeric codes are part of the service wire protocol
// (mirrors the vocabulary of reference Common.h:42-72; values are our own).
enum class Phase : int {
    IDLE = 0,
    TERMINATE = 1,
    MKDIRS = 2,      // "MKDIRS"
    WRITE = 3,       // "WRITE"  (create files / write blocks)
    READ = 4,        // "READ"
    STAT = 5,        // "STAT"
    RMFILES = 6,     // "RMFILES"
    RMDIRS = 7,      // "RMDIRS"
    SYNC = 8,        // "SYNC"
    DROPCACHES = 9,  // "DROPCACHE"
    NETBENCH = 10,   // netbench transfer phase (service mode)
    PUTOBJS = 11,    // S3 object upload    (engine-level alias of WRITE)
    GETOBJS = 12,    // S3 object download  (engine-level alias of READ)
};

inline const char* phaseName(Phase p)
{
    switch (p) {
        case Phase::IDLE: return "IDLE";
        case Phase::TERMINATE: return "QUIT";
        case Phase::MKDIRS: return "MKDIRS";
        case Phase::WRITE: return "WRITE";
        case Phase::READ: return "READ";
        case Phase::STAT: return "STAT";
        case Phase::RMFILES: return "RMFILES";
        case Phase::RMDIRS: return "RMDIRS";
        case Phase::SYNC: return "SYNC";
        case Phase::DROPCACHES: return "DROPCACHE";
        case Phase::NETBENCH: return "NETBENCH";
        case Phase::PUTOBJS: return "PUTOBJS";
        case Phase::GETOBJS: return "GETOBJS";
    }
    return "UNKNOWN";
}

enum class PathType : int {
    DIR = 0,
    FILE = 1,
    BLOCKDEV = 2,
};

// Live op counters, updated with relaxed atomics from the hot loop and read
// by the Python-side live-stats poller (reference analogue: LiveOps.h).
struct alignas(64) AtomicLiveOps {
    std::atomic<uint64_t> entries{0};
    std::atomic<uint64_t> bytes{0};
    std::atomic<uint64_t> iops{0};

    void reset()
    {
        entries.store(0, std::memory_order_relaxed);
        bytes.store(0, std::memory_order_relaxed);
        iops.store(0, std::memory_order_relaxed);
    }
};

struct LiveOpsSnapshot {
    uint64_t entries = 0;
    uint64_t bytes = 0;
    uint64_t iops = 0;

    void takeFrom(const AtomicLiveOps& a)
    {
        entries = a.entries.load(std::memory_order_relaxed);
        bytes = a.bytes.load(std::memory_order_relaxed);
        iops = a.iops.load(std::memory_order_relaxed);
    }
};

class WorkerError : public std::runtime_error {
public:
    explicit WorkerError(const std::string& msg) : std::runtime_error(msg) {}
};

class InterruptedError : public std::runtime_error {
public:
    InterruptedError() : std::runtime_error("interrupted") {}
};

} // namespace eb
