// ensemble.hpp — synthetic in-process ZooKeeper ensemble.
//
// The reference's tests require a real ZooKeeper on localhost
// (test/helper.js:57-62 in the reference). This build instead
// ships a from-scratch in-process ensemble that speaks the real jute wire
// protocol over TCP (so the native client, or any ZK client, can talk to
// it), with:
//   - a shared data tree across N "servers" (1- and 3-node modes),
//   - real session semantics: negotiation, touch-on-request, expiry sweep,
//     ephemeral-node deletion on expiry/close, reconnect-with-same-session,
//   - leader-kill with an optional election pause (storm testing,
//     BASELINE.json config 4),
//   - per-response latency injection,
//   - one-shot data/child watches (Binder-style readers).
//
// Threading: an IO-loop pool (like ZooKeeper's selector threads) with
// connections assigned round-robin, over 64-way sharded tree/watch state —
// see ensemble.cpp's header comment for the full model and lock order. The
// control/introspection API is thread-safe.
#pragma once

#include <atomic>
#include <condition_variable>
#include <cstdint>
#include <map>
#include <memory>
#include <mutex>
#include <set>
#include <string>
#include <thread>
#include <unordered_map>
#include <unordered_set>
#include <vector>

#include "jute.hpp"
#include "log.hpp"
#include "loop.hpp"

namespace registrar {
namespace zk {

struct EnsembleConfig {
  std::vector<int> ports{0};        // 0 = auto-assign; one entry per server
  std::string bind_host = "127.0.0.1";
  int tick_ms = 100;                // expiry sweep granularity
  int min_session_timeout_ms = 400;  // negotiated floor (2*tick in real ZK)
  int max_session_timeout_ms = 60000;
  int latency_ms = 0;               // fixed artificial delay per response
  int election_ms = 0;              // pause after leader kill before failover serves
  // IO loop threads (0 = auto). Like real ZooKeeper's selector threads, IO
  // parallelism is independent of ensemble size: connections are assigned
  // round-robin across the pool regardless of which server accepted them.
  int io_threads = 0;
  LogLevel log_level = LogLevel::Warn;
};

struct NodeInfo {  // introspection result
  bool exists = false;
  std::string data;
  Stat stat;
};

class Ensemble {
 public:
  explicit Ensemble(EnsembleConfig cfg);
  ~Ensemble();

  Ensemble(const Ensemble&) = delete;
  Ensemble& operator=(const Ensemble&) = delete;

  void start();
  void stop();

  // Bound ports after start() (auto-assigned resolved).
  std::vector<int> ports() const;
  std::string connect_string() const;  // "host:port,host:port,..."

  // --- fault injection / control (thread-safe) ---
  void kill_server(size_t idx);      // close listener + connections; sessions persist
  void restart_server(size_t idx);   // re-listen on the same port
  bool server_up(size_t idx) const;
  size_t leader() const;
  // Kill the current leader; remaining servers refuse connects for
  // election_ms, then one becomes leader. Returns the killed index.
  size_t kill_leader();
  void expire_session(int64_t session_id);  // force-expire (ephemerals vanish)
  void set_latency_ms(int ms);

  // --- introspection (thread-safe) ---
  NodeInfo get(const std::string& path) const;
  std::vector<std::string> children(const std::string& path) const;
  size_t node_count() const;             // all znodes (excl. root)
  size_t ephemeral_count() const;
  std::vector<int64_t> session_ids() const;
  int64_t zxid() const;
  // cumulative request counters by opcode name ("create", "exists", ...)
  std::map<std::string, uint64_t> counters() const;

 private:
  struct Impl;
  std::unique_ptr<Impl> impl_;
};

}  // namespace zk
}  // namespace registrar
