// log.hpp — bunyan-compatible JSON-lines structured logger.
//
// The reference emits bunyan records (main.js:23-28): one JSON object per
// line on stdout with name/hostname/pid/level/msg/time/v fields plus
// arbitrary structured context, so existing `| bunyan` tooling keeps working
// against this daemon. Child loggers carry bound fields
// (component=..., reference: lib/index.js:35, lib/zk.js:87).
#pragma once

#include <sys/types.h>
#include <time.h>
#include <unistd.h>

#include <atomic>
#include <cstdio>
#include <memory>
#include <mutex>
#include <string>
#include <utility>
#include <vector>

#include "json.hpp"

namespace registrar {

enum class LogLevel : int {
  Trace = 10,
  Debug = 20,
  Info = 30,
  Warn = 40,
  Error = 50,
  Fatal = 60,
};

inline const char* log_level_name(LogLevel l) {
  switch (l) {
    case LogLevel::Trace:
      return "trace";
    case LogLevel::Debug:
      return "debug";
    case LogLevel::Info:
      return "info";
    case LogLevel::Warn:
      return "warn";
    case LogLevel::Error:
      return "error";
    case LogLevel::Fatal:
      return "fatal";
  }
  return "info";
}

inline bool log_level_from_name(const std::string& s, LogLevel* out) {
  if (s == "trace")
    *out = LogLevel::Trace;
  else if (s == "debug")
    *out = LogLevel::Debug;
  else if (s == "info")
    *out = LogLevel::Info;
  else if (s == "warn")
    *out = LogLevel::Warn;
  else if (s == "error")
    *out = LogLevel::Error;
  else if (s == "fatal")
    *out = LogLevel::Fatal;
  else
    return false;
  return true;
}

// ISO-8601 UTC with milliseconds, bunyan style: 2026-01-02T03:04:05.678Z
inline std::string iso8601_now() {
  struct timespec ts;
  clock_gettime(CLOCK_REALTIME, &ts);
  struct tm tm_utc;
  gmtime_r(&ts.tv_sec, &tm_utc);
  char buf[40];
  snprintf(buf, sizeof(buf), "%04d-%02d-%02dT%02d:%02d:%02d.%03ldZ", tm_utc.tm_year + 1900, tm_utc.tm_mon + 1,
           tm_utc.tm_mday, tm_utc.tm_hour, tm_utc.tm_min, tm_utc.tm_sec, ts.tv_nsec / 1000000);
  return buf;
}

// Shared sink + level; children share both with their root.
struct LogCore {
  std::mutex mu;
  FILE* stream = stdout;
  std::atomic<int> level{static_cast<int>(LogLevel::Info)};
  std::string name = "registrar";
  std::string hostname;

  LogCore() {
    char buf[256] = {0};
    if (gethostname(buf, sizeof(buf) - 1) == 0)
      hostname = buf;
    else
      hostname = "localhost";
  }
};

class Logger {
 public:
  Logger() : core_(std::make_shared<LogCore>()) {}
  explicit Logger(std::string name) : core_(std::make_shared<LogCore>()) { core_->name = std::move(name); }

  Logger child(std::vector<JsonMember> fields) const {
    Logger l(*this);
    for (auto& f : fields) l.bound_.push_back(std::move(f));
    return l;
  }
  Logger child(const std::string& component) const { return child({{"component", Json(component)}}); }

  void set_level(LogLevel l) { core_->level.store(static_cast<int>(l)); }
  LogLevel level() const { return static_cast<LogLevel>(core_->level.load()); }
  bool enabled(LogLevel l) const { return static_cast<int>(l) >= core_->level.load(); }
  void set_stream(FILE* f) { core_->stream = f; }
  const std::string& name() const { return core_->name; }

  void log(LogLevel lvl, const std::string& msg) const { log(lvl, msg, {}); }

  void log(LogLevel lvl, const std::string& msg, std::vector<JsonMember> fields,
           const char* file = nullptr, int line = 0) const {
    if (!enabled(lvl)) return;
    Json rec = Json::object();
    rec.set("name", Json(core_->name));
    rec.set("hostname", Json(core_->hostname));
    rec.set("pid", Json(static_cast<int64_t>(getpid())));
    for (const auto& b : bound_) rec.set(b.first, b.second);
    for (auto& f : fields) rec.set(f.first, std::move(f.second));
    rec.set("level", Json(static_cast<int64_t>(lvl)));
    rec.set("msg", Json(msg));
    rec.set("time", Json(iso8601_now()));
    // bunyan `src:true` parity: when the logger runs at debug/trace
    // verbosity, every record carries the caller's file:line
    // (reference: main.js:75-76)
    if (file && core_->level.load() <= static_cast<int>(LogLevel::Debug)) {
      const char* base = file;
      for (const char* p = file; *p; p++)
        if (*p == '/') base = p + 1;
      Json src = Json::object();
      src.set("file", Json(std::string(base)));
      src.set("line", Json(static_cast<int64_t>(line)));
      rec.set("src", std::move(src));
    }
    rec.set("v", Json(static_cast<int64_t>(0)));
    std::string line_out = rec.dump();
    line_out += '\n';
    std::lock_guard<std::mutex> g(core_->mu);
    fwrite(line_out.data(), 1, line_out.size(), core_->stream);
    fflush(core_->stream);
  }

  // __builtin_FILE/__builtin_LINE default args evaluate at the CALL site
  // (gcc and ROCm clang both support them in C++17), giving bunyan-style
  // src without a macro layer
  void trace(const std::string& msg, std::vector<JsonMember> f = {},
             const char* file = __builtin_FILE(), int line = __builtin_LINE()) const {
    log(LogLevel::Trace, msg, std::move(f), file, line);
  }
  void debug(const std::string& msg, std::vector<JsonMember> f = {},
             const char* file = __builtin_FILE(), int line = __builtin_LINE()) const {
    log(LogLevel::Debug, msg, std::move(f), file, line);
  }
  void info(const std::string& msg, std::vector<JsonMember> f = {},
            const char* file = __builtin_FILE(), int line = __builtin_LINE()) const {
    log(LogLevel::Info, msg, std::move(f), file, line);
  }
  void warn(const std::string& msg, std::vector<JsonMember> f = {},
            const char* file = __builtin_FILE(), int line = __builtin_LINE()) const {
    log(LogLevel::Warn, msg, std::move(f), file, line);
  }
  void error(const std::string& msg, std::vector<JsonMember> f = {},
             const char* file = __builtin_FILE(), int line = __builtin_LINE()) const {
    log(LogLevel::Error, msg, std::move(f), file, line);
  }
  void fatal(const std::string& msg, std::vector<JsonMember> f = {},
             const char* file = __builtin_FILE(), int line = __builtin_LINE()) const {
    log(LogLevel::Fatal, msg, std::move(f), file, line);
  }

 private:
  std::shared_ptr<LogCore> core_;
  std::vector<JsonMember> bound_;
};

}  // namespace registrar
