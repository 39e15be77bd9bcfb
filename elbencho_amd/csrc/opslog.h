// Per-operation JSON-lines trace log.
//
// Reference analogue: toolkits/OpsLogger.{h,cpp} (--opslog/--opsloglock):
// one JSON object per I/O syscall with op name, target, offset, length and
// error flag, written pre/post op with errno preserved. Independent
// implementation: a process-wide logger with an internal mutex (the
// reference serializes across PROCESSES with flock; we support that too).

#pragma once

#include <fcntl.h>
#include <sys/file.h>
#include <unistd.h>

#include <cerrno>
#include <cstdio>
#include <cstring>
#include <mutex>
#include <string>

namespace eb {

class OpsLogger {
public:
    void open(const std::string& path, bool useFlock)
    {
        std::lock_guard<std::mutex> lk(mtx);
        if (file) fclose(file);
        file = fopen(path.c_str(), "a");
        flockEnabled = useFlock;
        enabled = (file != nullptr);
    }

    void close()
    {
        std::lock_guard<std::mutex> lk(mtx);
        if (file) fclose(file);
        file = nullptr;
        enabled = false;
    }

    bool isEnabled() const { return enabled; }

    // pre==true logs the op start; err is meaningful on post entries.
    void log(int rank, const char* op, const std::string& target, uint64_t offset,
             uint64_t len, bool pre, bool isError)
    {
        if (!enabled) return;
        int savedErrno = errno; // preserve errno across logging (reference contract)

        std::lock_guard<std::mutex> lk(mtx);
        if (!file) return;
        if (flockEnabled) flock(fileno(file), LOCK_EX);

        fprintf(file,
                "{\"rank\":%d,\"op\":\"%s\",\"entry\":\"%s\",\"offset\":%llu,"
                "\"len\":%llu,\"type\":\"%s\"%s}\n",
                rank, op, target.c_str(), (unsigned long long)offset,
                (unsigned long long)len, pre ? "pre" : "post",
                isError ? ",\"error\":true" : "");
        fflush(file);

        if (flockEnabled) flock(fileno(file), LOCK_UN);
        errno = savedErrno;
    }

    ~OpsLogger()
    {
        if (file) fclose(file);
    }

private:
    FILE* file = nullptr;
    bool enabled = false;
    bool flockEnabled = false;
    std::mutex mtx;
};

} // namespace eb
