/*
 * binder-amd: structured JSON logging, bunyan line format.
 *
 * The reference emits bunyan JSON lines ({"name","hostname","pid","level",
 * "msg","time","v":0, ...}) with child loggers carrying bound fields
 * (/root/reference/main.js:40-47, lib/server.js:484-490). Levels use the
 * bunyan numeric scale (trace 10 .. fatal 60) so downstream tooling that
 * understands bunyan keeps working.
 */
#pragma once

#include <cstdint>
#include <memory>
#include <string>
#include <vector>

#include "json.hpp"

namespace bamd {

enum class LogLevel : int {
    Trace = 10,
    Debug = 20,
    Info = 30,
    Warn = 40,
    Error = 50,
    Fatal = 60,
};

LogLevel logLevelFromName(const std::string& name, LogLevel dflt);

class Logger {
  public:
    /* Root logger writing to fd (1 = stdout). */
    Logger(std::string name, LogLevel level, int fd = 1);

    /* Child logger with bound fields (rendered into every line). */
    Logger child(const JsonObject& fields) const;

    LogLevel level() const { return *level_; }
    void setLevel(LogLevel lv) { *level_ = lv; }
    bool enabled(LogLevel lv) const { return (int)lv >= (int)*level_; }

    void log(LogLevel lv, const std::string& msg) const {
        if (enabled(lv)) write(lv, msg, nullptr);
    }
    void log(LogLevel lv, const std::string& msg,
             const JsonObject& extra) const {
        if (enabled(lv)) write(lv, msg, &extra);
    }

    void trace(const std::string& m) const { log(LogLevel::Trace, m); }
    void debug(const std::string& m) const { log(LogLevel::Debug, m); }
    void info(const std::string& m) const { log(LogLevel::Info, m); }
    void warn(const std::string& m) const { log(LogLevel::Warn, m); }
    void error(const std::string& m) const { log(LogLevel::Error, m); }
    void fatal(const std::string& m) const { log(LogLevel::Fatal, m); }
    void info(const JsonObject& extra, const std::string& m) const {
        log(LogLevel::Info, m, extra);
    }
    void warn(const JsonObject& extra, const std::string& m) const {
        log(LogLevel::Warn, m, extra);
    }
    void error(const JsonObject& extra, const std::string& m) const {
        log(LogLevel::Error, m, extra);
    }

    /*
     * Zero-DOM fast path for hot log lines (the per-query line is
     * ~35% of binderd CPU at info level when built through JsonObject:
     * gprof showed 25 jsonEscape calls + a std::map build/teardown per
     * query). The caller appends pre-serialized `,"k":v` fragments
     * into rawFields (escaping only what needs it via jsonEscape) and
     * this emits one bunyan line with no intermediate DOM.
     */
    void logRaw(LogLevel lv, const char* msg,
                std::string_view rawFields) const;

  private:
    void write(LogLevel lv, const std::string& msg,
               const JsonObject* extra) const;

    std::string name_;
    std::shared_ptr<LogLevel> level_;  // shared: children follow root -v
    int fd_;
    std::string boundPrefix_;  // pre-rendered ',"k":v,...' for bound fields
};

/* ISO-8601 UTC timestamp with millisecond precision. */
std::string isoTimeNow();
int64_t monotonicMillis();
int64_t wallMillis();

}  // namespace bamd
