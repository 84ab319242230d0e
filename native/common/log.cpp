#include "log.hpp"

#include <time.h>
#include <unistd.h>

#include <cstdio>
#include <cstring>

namespace bamd {

LogLevel logLevelFromName(const std::string& name, LogLevel dflt) {
    if (name == "trace") return LogLevel::Trace;
    if (name == "debug") return LogLevel::Debug;
    if (name == "info") return LogLevel::Info;
    if (name == "warn") return LogLevel::Warn;
    if (name == "error") return LogLevel::Error;
    if (name == "fatal") return LogLevel::Fatal;
    return dflt;
}

std::string isoTimeNow() {
    struct timespec ts;
    clock_gettime(CLOCK_REALTIME, &ts);
    /* gmtime_r + strftime per line is measurable at >1M logged
     * queries/s; the second-granular prefix only changes once per
     * second, so cache it per thread. */
    thread_local time_t cachedSec = 0;
    thread_local char prefix[32];
    thread_local size_t prefixLen = 0;
    if (ts.tv_sec != cachedSec || prefixLen == 0) {
        struct tm tm;
        gmtime_r(&ts.tv_sec, &tm);
        prefixLen = strftime(prefix, sizeof(prefix),
                             "%Y-%m-%dT%H:%M:%S", &tm);
        cachedSec = ts.tv_sec;
    }
    char buf[40];
    memcpy(buf, prefix, prefixLen);
    snprintf(buf + prefixLen, sizeof(buf) - prefixLen, ".%03ldZ",
             ts.tv_nsec / 1000000);
    return buf;
}

int64_t monotonicMillis() {
    struct timespec ts;
    clock_gettime(CLOCK_MONOTONIC, &ts);
    return (int64_t)ts.tv_sec * 1000 + ts.tv_nsec / 1000000;
}

int64_t wallMillis() {
    struct timespec ts;
    clock_gettime(CLOCK_REALTIME, &ts);
    return (int64_t)ts.tv_sec * 1000 + ts.tv_nsec / 1000000;
}

static std::string hostName() {
    char buf[256];
    if (gethostname(buf, sizeof(buf)) != 0) return "unknown";
    buf[sizeof(buf) - 1] = '\0';
    return buf;
}

Logger::Logger(std::string name, LogLevel level, int fd)
    : name_(std::move(name)),
      level_(std::make_shared<LogLevel>(level)),
      fd_(fd) {}

Logger Logger::child(const JsonObject& fields) const {
    Logger c(*this);
    for (const auto& [k, v] : fields) {
        c.boundPrefix_.push_back(',');
        jsonEscape(k, c.boundPrefix_);
        c.boundPrefix_.push_back(':');
        v.dumpTo(c.boundPrefix_);
    }
    return c;
}

void Logger::write(LogLevel lv, const std::string& msg,
                   const JsonObject* extra) const {
    static const std::string kHost = hostName();
    std::string line;
    line.reserve(256 + msg.size() + boundPrefix_.size());
    line += "{\"v\":0,\"name\":";
    jsonEscape(name_, line);
    line += ",\"hostname\":";
    jsonEscape(kHost, line);
    char buf[64];
    snprintf(buf, sizeof(buf), ",\"pid\":%d,\"level\":%d", (int)getpid(),
             (int)lv);
    line += buf;
    line += boundPrefix_;
    if (extra != nullptr) {
        for (const auto& [k, v] : *extra) {
            line.push_back(',');
            jsonEscape(k, line);
            line.push_back(':');
            v.dumpTo(line);
        }
    }
    line += ",\"msg\":";
    jsonEscape(msg, line);
    line += ",\"time\":\"";
    line += isoTimeNow();
    line += "\"}\n";
    // Single write(2) keeps lines atomic for typical sizes.
    ssize_t rv = ::write(fd_, line.data(), line.size());
    (void)rv;
}

void Logger::logRaw(LogLevel lv, const char* msg,
                    std::string_view rawFields) const {
    if (!enabled(lv)) return;
    static const std::string kHost = hostName();
    thread_local std::string line;  // reused: no steady-state allocs
    line.clear();
    line += "{\"v\":0,\"name\":";
    jsonEscape(name_, line);
    line += ",\"hostname\":";
    jsonEscape(kHost, line);
    char buf[64];
    snprintf(buf, sizeof(buf), ",\"pid\":%d,\"level\":%d", (int)getpid(),
             (int)lv);
    line += buf;
    line += boundPrefix_;
    line += rawFields;
    line += ",\"msg\":\"";
    line += msg;  // caller passes a literal needing no escaping
    line += "\",\"time\":\"";
    line += isoTimeNow();
    line += "\"}\n";
    ssize_t rv = ::write(fd_, line.data(), line.size());
    (void)rv;
}

}  // namespace bamd
