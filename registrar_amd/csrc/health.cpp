// health.cpp — health checker implementation (see health.hpp).
#include "health.hpp"

#include <fcntl.h>
#include <poll.h>
#include <signal.h>
#include <sys/wait.h>
#include <unistd.h>

#include <algorithm>
#include <cerrno>
#include <chrono>
#include <regex>

#include "loop.hpp"  // now_ms

namespace registrar {

// Compile a stdoutMatch pattern+flags into a std::regex. Throws
// std::runtime_error (with the offending flag / regex error) on anything the
// ECMAScript std::regex dialect cannot honor — called at CONFIG PARSE time so
// a bad pattern fails `registrard check`, not a 3 a.m. health run (the
// reference's assert-plus throws at construction too, lib/health.js:23-38).
std::regex compile_stdout_match(const StdoutMatch& m) {
  auto syn = std::regex::ECMAScript;
  for (char f : m.flags) {
    switch (f) {
      case 'i': syn |= std::regex::icase; break;
      case 'm': syn |= std::regex::multiline; break;
      case 'g': break;  // single regex_search: global is a no-op, accept it
      default:
        throw std::runtime_error(std::string("healthCheck.stdoutMatch.flags: unsupported flag '") +
                                 f + "' (supported: i, m, g)");
    }
  }
  try {
    return std::regex(m.pattern, syn);
  } catch (const std::regex_error& e) {
    throw std::runtime_error(std::string("healthCheck.stdoutMatch.pattern: invalid regex: ") + e.what());
  }
}

HealthCheckConfig parse_health_check(const Json& j) {
  if (!j.is_object()) throw std::runtime_error("healthCheck: must be an object");
  HealthCheckConfig cfg;
  const Json* cmd = j.find("command");
  if (!cmd || !cmd->is_string()) throw std::runtime_error("healthCheck.command: string required");
  cfg.command = cmd->as_string();
  cfg.interval_ms = j.get_int("interval", cfg.interval_ms);
  cfg.timeout_ms = j.get_int("timeout", cfg.timeout_ms);
  cfg.period_ms = j.get_int("period", cfg.period_ms);
  cfg.threshold = j.get_int("threshold", cfg.threshold);
  cfg.ignore_exit_status = j.get_bool("ignoreExitStatus", false);
  if (const Json* sm = j.find("stdoutMatch")) {
    if (!sm->is_object()) throw std::runtime_error("healthCheck.stdoutMatch: object required");
    StdoutMatch m;
    m.pattern = sm->get_string("pattern", "");
    m.flags = sm->get_string("flags", "");
    m.invert = sm->get_bool("invert", false);
    if (!m.pattern.empty()) {
      compile_stdout_match(m);  // fail fast on bad pattern / unknown flags
      cfg.stdout_match = m;
    }
  }
  return cfg;
}

Json HealthRecord::to_json() const {
  Json rec = Json::object();
  rec.set("type", Json(ok ? "ok" : "fail"));
  rec.set("command", Json(command));
  if (!ok) {
    Json err = Json::object();
    err.set("message", Json(error));
    err.set("code", Json(static_cast<int64_t>(exit_status)));
    if (!stderr_tail.empty()) err.set("stderr", Json(stderr_tail));
    rec.set("err", std::move(err));
    rec.set("failures", Json(failures));
    rec.set("isDown", Json(is_down));
    rec.set("threshold", Json(threshold));
  }
  return rec;
}

ExecResult exec_with_timeout(const std::string& command, int64_t timeout_ms, size_t max_buffer) {
  ExecResult result;
  // Separate pipes: stdoutMatch must see STDOUT ONLY (the reference matches
  // child.exec's stdout arg, /root/reference/lib/health.js:89-101 — a
  // chatty-on-stderr command must not satisfy or spoil the pattern).
  int outfd[2], errfd[2];
  if (pipe2(outfd, O_CLOEXEC) != 0) {
    result.out = "pipe failed";
    return result;
  }
  if (pipe2(errfd, O_CLOEXEC) != 0) {
    close(outfd[0]);
    close(outfd[1]);
    result.out = "pipe failed";
    return result;
  }
  pid_t pid = fork();
  if (pid < 0) {
    close(outfd[0]); close(outfd[1]);
    close(errfd[0]); close(errfd[1]);
    result.out = "fork failed";
    return result;
  }
  if (pid == 0) {
    // child: own process group so the whole shell pipeline can be signaled
    setpgid(0, 0);
    dup2(outfd[1], STDOUT_FILENO);
    dup2(errfd[1], STDERR_FILENO);
    execl("/bin/sh", "sh", "-c", command.c_str(), static_cast<char*>(nullptr));
    _exit(127);
  }
  close(outfd[1]);
  close(errfd[1]);
  setpgid(pid, pid);  // race-safe double setpgid
  // non-blocking reads everywhere: a backgrounded grandchild keeping the
  // write end open must never wedge the health thread
  fcntl(outfd[0], F_SETFL, O_NONBLOCK);
  fcntl(errfd[0], F_SETFL, O_NONBLOCK);

  int64_t deadline = now_ms() + timeout_ms;
  bool sent_term = false;
  int64_t kill_deadline = 0;
  struct pollfd pfds[2];
  pfds[0] = {outfd[0], POLLIN, 0};
  pfds[1] = {errfd[0], POLLIN, 0};
  bool open_[2] = {true, true};
  std::string* sinks[2] = {&result.out, &result.err};

  auto drain_ready = [&]() {
    // read whatever poll flagged; returns true if any fd made progress
    bool progress = false;
    for (int i = 0; i < 2; i++) {
      if (!open_[i] || !(pfds[i].revents & (POLLIN | POLLHUP | POLLERR))) continue;
      char buf[16384];
      ssize_t n;
      while ((n = read(pfds[i].fd, buf, sizeof(buf))) > 0) {
        progress = true;
        std::string& s = *sinks[i];
        if (s.size() < max_buffer)
          s.append(buf, static_cast<size_t>(std::min<size_t>(static_cast<size_t>(n), max_buffer - s.size())));
      }
      if (n == 0 || (n < 0 && errno != EAGAIN && errno != EWOULDBLOCK)) {
        open_[i] = false;
        pfds[i].fd = -1;  // poll ignores negative fds
        progress = true;
      }
    }
    return progress;
  };

  bool reaped = false;
  int status = 0;
  while (!reaped) {
    int64_t now = now_ms();
    if (!result.timed_out && now >= deadline) {
      // SIGTERM the process group (reference killSignal: lib/health.js:48),
      // escalate to SIGKILL if it lingers
      result.timed_out = true;
      sent_term = true;
      kill(-pid, SIGTERM);
      kill_deadline = now + 1000;
    }
    if (sent_term && now >= kill_deadline) {
      kill(-pid, SIGKILL);
      sent_term = false;  // only escalate once
    }
    int64_t wait_until = result.timed_out ? (sent_term ? kill_deadline : now + 50) : deadline;
    int poll_ms = static_cast<int>(std::max<int64_t>(1, std::min<int64_t>(wait_until - now, 100)));

    if (open_[0] || open_[1]) {
      if (poll(pfds, 2, poll_ms) > 0) drain_ready();
    } else {
      std::this_thread::sleep_for(std::chrono::milliseconds(poll_ms));
    }

    pid_t r = waitpid(pid, &status, WNOHANG);
    if (r == pid) reaped = true;
  }

  // bounded post-reap drain: give surviving writers (grandchildren) a short
  // grace to flush, then close regardless — EOF may never come
  int64_t grace_deadline = now_ms() + 250;
  while ((open_[0] || open_[1]) && now_ms() < grace_deadline) {
    int pr = poll(pfds, 2, 25);
    if (pr > 0) drain_ready();
    if (pr == 0) break;  // nothing pending and the child is gone: done
  }
  close(outfd[0]);
  close(errfd[0]);
  if (WIFEXITED(status))
    result.exit_status = WEXITSTATUS(status);
  else
    result.exit_status = -1;
  return result;
}

HealthCheck::HealthCheck(HealthCheckConfig cfg, Logger log) : cfg_(std::move(cfg)), log_(log.child("HealthCheck")) {}

HealthCheck::~HealthCheck() {
  try {
    stop();
  } catch (...) {
  }
}

void HealthCheck::set_callback(RecordCallback cb) { cb_ = std::move(cb); }

void HealthCheck::start() {
  std::lock_guard<std::mutex> g(mu_);
  if (running_) return;
  running_ = true;
  thread_ = std::thread([this] { run_loop(); });
}

void HealthCheck::stop() {
  {
    std::lock_guard<std::mutex> g(mu_);
    if (!running_) return;
    running_ = false;
  }
  cv_.notify_all();
  if (thread_.joinable()) thread_.join();
}

void HealthCheck::run_loop() {
  while (true) {
    {
      std::unique_lock<std::mutex> g(mu_);
      if (!running_) return;
    }
    check_once();
    std::unique_lock<std::mutex> g(mu_);
    cv_.wait_for(g, std::chrono::milliseconds(cfg_.interval_ms), [this] { return !running_; });
    if (!running_) return;
  }
}

HealthRecord HealthCheck::check_once() {
  log_.debug("check: running command", {{"command", Json(cfg_.command)}});
  ExecResult res = exec_with_timeout(cfg_.command, cfg_.timeout_ms, cfg_.max_buffer);
  HealthRecord rec = evaluate(res);
  emit(rec);
  return rec;
}

HealthRecord HealthCheck::evaluate(const ExecResult& res) {
  HealthRecord rec;
  rec.command = cfg_.command;
  rec.threshold = cfg_.threshold;
  rec.exit_status = res.exit_status;
  rec.stdout_tail = res.out.size() > 256 ? res.out.substr(res.out.size() - 256) : res.out;
  rec.stderr_tail = res.err.size() > 256 ? res.err.substr(res.err.size() - 256) : res.err;

  bool ok = true;
  if (res.timed_out) {
    ok = false;
    rec.error = "command timed out after " + std::to_string(cfg_.timeout_ms) + "ms";
  } else if (res.exit_status != 0 && !cfg_.ignore_exit_status) {
    // non-zero exit ⇒ failure unless ignoreExitStatus (lib/health.js:90-95)
    ok = false;
    rec.error = "command exited " + std::to_string(res.exit_status);
  } else if (cfg_.stdout_match) {
    // pattern+flags were validated by parse_health_check; match STDOUT only
    std::regex re = compile_stdout_match(*cfg_.stdout_match);
    bool matched = std::regex_search(res.out, re);
    // invert honored here (accepted-but-ignored in the reference, §2.2.4)
    bool want_match = !cfg_.stdout_match->invert;
    if (matched != want_match) {
      ok = false;
      rec.error = std::string("stdout match (") + cfg_.stdout_match->pattern + ") failed";
      rec.exit_status = -1;  // reference sets code -1 on regex failure (lib/health.js:107)
    }
  }

  if (ok) {
    rec.ok = true;
    if (down_.load()) {
      // recovery: reset the flap window so re-marking down needs `threshold`
      // fresh failures (reference latched down forever — fixed, §2.2.2)
      fail_times_.clear();
      down_.store(false);
    }
    log_.debug("healthCheck: ok", {{"command", Json(cfg_.command)}});
    return rec;
  }

  rec.ok = false;
  int64_t now = now_ms();
  fail_times_.push_back(now);
  // true sliding window: evict failures older than `period` (fix of §2.2.2)
  while (!fail_times_.empty() && now - fail_times_.front() > cfg_.period_ms) fail_times_.pop_front();
  rec.failures = static_cast<int64_t>(fail_times_.size());
  if (!down_.load() && rec.failures >= cfg_.threshold) down_.store(true);
  rec.is_down = down_.load();
  log_.debug("check: command failed",
             {{"command", Json(cfg_.command)},
              {"err", Json(rec.error)},
              {"failures", Json(rec.failures)},
              {"isDown", Json(rec.is_down)}});
  return rec;
}

void HealthCheck::emit(const HealthRecord& rec) {
  {
    std::lock_guard<std::mutex> g(rec_mu_);
    records_.push_back(rec);
  }
  if (cb_) cb_(rec);
}

std::vector<HealthRecord> HealthCheck::poll_records() {
  std::lock_guard<std::mutex> g(rec_mu_);
  std::vector<HealthRecord> out;
  out.swap(records_);
  return out;
}

}  // namespace registrar
