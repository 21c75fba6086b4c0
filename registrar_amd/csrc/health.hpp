// health.hpp — periodic health-check command runner with flap damping.
//
// Re-implements the reference's lib/health.js contract (SURVEY.md §2.1
// "Health checker"): run `command` through the shell every `interval` ms with
// an exec `timeout` (SIGTERM, then SIGKILL), capture up to 1 MiB of stdout;
// a run fails on non-zero exit (unless ignoreExitStatus) or when stdout does
// not match stdoutMatch.pattern. Emits {type:'ok'|'fail', ...} records.
//
// Deliberate fixes over the reference:
//   - flap window is a true sliding window: a run counts as "down" when
//     `threshold` failures accumulated within the trailing `period` ms
//     (the reference cleared its failure list exactly once at 2×period and
//     never again, lib/health.js:60-64).
//   - stdoutMatch.invert is honored (validated-but-ignored in the reference,
//     lib/health.js:32-33 vs 87-112): invert=true fails when the pattern DOES
//     match.
//   - after recovery (ok while down) the window resets, so re-marking down
//     requires `threshold` fresh failures (the reference latched `down`
//     forever).
//
// Runs on its own thread; records are delivered to an optional callback
// (checker thread) and to a thread-safe poll queue.
#pragma once

#include <atomic>
#include <condition_variable>
#include <cstdint>
#include <deque>
#include <functional>
#include <memory>
#include <mutex>
#include <optional>
#include <regex>
#include <string>
#include <thread>
#include <vector>

#include "json.hpp"
#include "log.hpp"

namespace registrar {

struct StdoutMatch {
  std::string pattern;  // ECMAScript-dialect std::regex (see docs/config.md
                        // for deltas vs JS RegExp: no lookbehind, etc.)
  std::string flags;    // "i" (icase), "m" (multiline), "g" (no-op for a
                        // single search) — anything else is rejected at
                        // config parse, where JS would have thrown too
  bool invert = false;
};

struct HealthCheckConfig {
  std::string command;            // required; run via /bin/sh -c
  int64_t interval_ms = 60000;    // between runs (lib/health.js:43)
  int64_t timeout_ms = 1000;      // exec timeout (lib/health.js:52)
  int64_t period_ms = 300000;     // flap window (lib/health.js:56)
  int64_t threshold = 5;          // failures within period ⇒ down (lib/health.js:58)
  bool ignore_exit_status = false;
  std::optional<StdoutMatch> stdout_match;
  size_t max_buffer = 1024 * 1024;  // stdout cap (lib/health.js:50)
};

HealthCheckConfig parse_health_check(const Json& j);

// Compile pattern+flags (throws std::runtime_error on unsupported flags or an
// invalid pattern — used by parse_health_check to fail fast at config time).
std::regex compile_stdout_match(const StdoutMatch& m);

struct HealthRecord {
  bool ok = false;
  std::string command;
  std::string error;     // failure description ("exit 1", "timeout", ...)
  int64_t failures = 0;  // failures currently inside the window
  bool is_down = false;
  int64_t threshold = 0;
  int exit_status = 0;
  std::string stdout_tail;  // last bytes of captured stdout (diagnostics)
  std::string stderr_tail;  // last bytes of captured stderr (diagnostics)

  Json to_json() const;
};

// One-shot command execution with timeout; building block for the checker and
// directly usable (e.g. GPU liveness probes).
struct ExecResult {
  int exit_status = -1;   // -1 ⇒ killed / failed to run
  bool timed_out = false;
  std::string out;        // captured stdout (capped) — the ONLY regex input
  std::string err;        // captured stderr (capped) — diagnostics only
};
ExecResult exec_with_timeout(const std::string& command, int64_t timeout_ms, size_t max_buffer = 1024 * 1024);

class HealthCheck {
 public:
  using RecordCallback = std::function<void(const HealthRecord&)>;

  HealthCheck(HealthCheckConfig cfg, Logger log);
  ~HealthCheck();

  HealthCheck(const HealthCheck&) = delete;
  HealthCheck& operator=(const HealthCheck&) = delete;

  void set_callback(RecordCallback cb);  // set before start()
  void start();
  void stop();

  // Run one check immediately on the calling thread (used by tests and by
  // start() for the initial run). Returns the emitted record.
  HealthRecord check_once();

  std::vector<HealthRecord> poll_records();
  bool is_down() const { return down_.load(); }

 private:
  void run_loop();
  HealthRecord evaluate(const ExecResult& res);
  void emit(const HealthRecord& rec);

  HealthCheckConfig cfg_;
  Logger log_;
  RecordCallback cb_;
  std::thread thread_;
  std::mutex mu_;
  std::condition_variable cv_;
  bool running_ = false;
  std::atomic<bool> down_{false};
  std::deque<int64_t> fail_times_;  // monotonic ms of failures in the window
  std::mutex rec_mu_;
  std::vector<HealthRecord> records_;
};

}  // namespace registrar
