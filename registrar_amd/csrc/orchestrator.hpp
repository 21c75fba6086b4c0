// orchestrator.hpp — register_plus: ties client + registration + health
// checker together and drives the heartbeat loop.
//
// Re-implements the reference's lib/index.js `register_plus` (SURVEY.md §2.1
// "Orchestrator") with the same observable event surface — register,
// unregister, ok, fail, error, heartbeat, heartbeatFailure (7 types,
// lib/index.js:33-177) plus sessionExpired for the expiry policy — and the
// same cadences: heartbeat every `heartbeatInterval` (default 3000 ms,
// lib/index.js:132), degraded to max(interval, 60000) after a failure
// (lib/index.js:146).
//
// Deliberate changes vs the reference:
//   - session expiry: the reference crashes (process.exit(1), main.js:141-144)
//     and relies on SMF to restart it; here the DEFAULT policy re-registers
//     in-process (new session + full register pipeline, which starts with
//     stale-entry cleanup — preserving the observable contract per SURVEY.md
//     §1), with exitOnExpiry=true for supervisor parity.
//   - heartbeat retry policy is plumbed from config (`heartbeat.retry`),
//     which the reference read but never used (§2.2.5).
//   - the registration-failure path can't crash on an undefined variable
//     (§2.2.3): it emits 'error' with the failure detail.
#pragma once

#include <atomic>
#include <condition_variable>
#include <cstdint>
#include <functional>
#include <memory>
#include <mutex>
#include <optional>
#include <string>
#include <thread>
#include <vector>

#include "health.hpp"
#include "json.hpp"
#include "log.hpp"
#include "registrar.hpp"
#include "zkclient.hpp"

namespace registrar {

struct OrchestratorConfig {
  zk::ZkClientConfig zk;
  RegistrationConfig registration;
  std::optional<HealthCheckConfig> health;
  int64_t heartbeat_interval_ms = 3000;  // lib/index.js:132
  // degraded cadence after a heartbeat failure: max(interval, floor)
  // (lib/index.js:142-146; configurable so tests can assert the reschedule)
  int64_t heartbeat_failure_floor_ms = 60000;
  zk::RetryPolicy heartbeat_retry;       // lib/zk.js:38-42 defaults
  bool exit_on_expiry = false;           // false ⇒ in-process re-register
  std::string log_level;                 // config `logLevel`
};

// Parse a full config-file JSON (schema SURVEY.md §2.5), including the
// back-compat hoist of top-level adminIp into registration (main.js:147) and
// the MI355X extras (gpuIndex; healthCheck.command "gpu-liveness" preset).
// Throws std::runtime_error on schema violations.
OrchestratorConfig parse_config(const Json& cfg);

struct OrchEvent {
  enum class Type { Register, Unregister, Ok, Fail, Error, Heartbeat, HeartbeatFailure, SessionExpired, Stopped };
  Type type;
  std::string detail;                // error text / context
  std::vector<std::string> znodes;   // Register/Unregister/Heartbeat
  int64_t rtt_us = 0;                // Heartbeat
};

const char* orch_event_name(OrchEvent::Type t);

struct OrchMetrics {
  uint64_t registers = 0;
  uint64_t unregisters = 0;
  uint64_t heartbeats = 0;
  uint64_t heartbeat_failures = 0;
  uint64_t session_expiries = 0;
  uint64_t errors = 0;
  std::vector<int64_t> recent_heartbeat_rtt_us;  // ring of the last 1024
};

class Orchestrator {
 public:
  using EventCallback = std::function<void(const OrchEvent&)>;

  Orchestrator(OrchestratorConfig cfg, Logger log);
  ~Orchestrator();

  Orchestrator(const Orchestrator&) = delete;
  Orchestrator& operator=(const Orchestrator&) = delete;

  void set_event_callback(EventCallback cb);  // before start()

  // Non-blocking: spawns the control thread (connect → register → loops).
  void start();
  // Block until the first successful register (true) or failure/stop (false).
  bool wait_registered(int64_t timeout_ms = -1);
  void stop();

  // Trigger one heartbeat immediately (bench/tests); returns rc, fills rtt.
  int heartbeat_now(int64_t* rtt_us = nullptr);

  std::vector<std::string> znodes() const;
  std::vector<OrchEvent> poll_events();
  OrchMetrics metrics() const;
  int64_t session_id() const;
  bool expired() const { return expired_flag_.load(); }  // exit_on_expiry case

 private:
  void control_loop();
  bool connect_and_register(bool initial);
  void heartbeat_loop();
  void on_health_record(const HealthRecord& rec);
  void emit(OrchEvent ev);

  OrchestratorConfig cfg_;
  Logger log_;
  EventCallback cb_;

  std::thread control_;
  std::atomic<bool> running_{false};
  std::atomic<bool> expired_flag_{false};

  mutable std::mutex mu_;            // guards client_, znodes_, down_
  std::shared_ptr<zk::ZkClient> client_;
  std::vector<std::string> znodes_;
  bool down_ = false;                // health-driven unregistered state

  std::condition_variable wake_cv_;  // control/heartbeat sleep interrupt
  std::mutex wake_mu_;
  bool expiry_signal_ = false;

  std::unique_ptr<HealthCheck> health_;

  std::mutex ev_mu_;
  std::condition_variable ev_cv_;
  std::vector<OrchEvent> ev_queue_;
  bool registered_once_ = false;
  bool failed_ = false;

  mutable std::mutex metrics_mu_;
  OrchMetrics metrics_;
};

}  // namespace registrar
