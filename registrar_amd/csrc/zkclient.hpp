// zkclient.hpp — native ZooKeeper client: session state machine + pipelined ops.
//
// This replaces the reference's zkplus → node-zookeeper → libzookeeper_mt
// chain with a from-scratch epoll client speaking the jute
// wire protocol (jute.hpp). The public verb surface is the zkplus surface the
// reference actually uses: stat/put/create/mkdirp/unlink/
// close plus connect events, with `heartbeat` (parallel exists over all
// registered znodes with bounded retry, reference lib/zk.js:21-59) built in.
//
// Concurrency model: one epoll loop thread owns the socket and all protocol
// state. Public methods are thread-safe; async ops run their callbacks on the
// loop thread; sync wrappers block the calling thread (never call them from a
// callback). Where the reference fans out independent ZK RPCs through
// vasync.forEachParallel (lib/register.js:85-171), this client pipelines the
// whole batch onto the single session socket in one writev-sized flush —
// that pipelining is the throughput core of the registrations/sec metric.
//
// Session semantics: CONNECTING → CONNECTED; on connection loss the client
// reconnects to the next server of the ensemble *with the same session id*
// (fast backoff); the server decides whether the session still lives. An
// expired-session handshake is terminal: the client emits Expired and stops,
// and the owner (orchestrator/daemon) chooses the recovery policy — the
// reference's policy is crash-and-restart (main.js:141-144); this build
// defaults to in-process re-register (SURVEY.md §1 "crucial architectural
// fact") with --exit-on-expiry for supervisor parity.
#pragma once

#include <atomic>
#include <condition_variable>
#include <cstdint>
#include <deque>
#include <functional>
#include <memory>
#include <mutex>
#include <string>
#include <vector>

#include "jute.hpp"
#include "log.hpp"
#include "loop.hpp"

namespace registrar {
namespace zk {

struct ServerAddr {
  std::string host;
  int port = 2181;
};

struct RetryPolicy {
  // reference defaults: lib/zk.js:38-42
  int64_t max_attempts = 5;
  int64_t initial_delay_ms = 1000;
  int64_t max_delay_ms = 30000;
};

struct ZkClientConfig {
  std::vector<ServerAddr> servers;
  int session_timeout_ms = 30000;   // `timeout` in the config schema (§2.5)
  int connect_timeout_ms = 4000;    // per-attempt TCP+handshake timeout
  // initial-connect retry (reference lib/zk.js:97-101: infinite, 1 s → 90 s)
  int64_t connect_initial_delay_ms = 1000;
  int64_t connect_max_delay_ms = 90000;
  int64_t connect_max_attempts = -1;  // <0 = infinite
  // session-preserving reconnect retry (fast: the session clock is ticking)
  int64_t reconnect_initial_delay_ms = 10;
  int64_t reconnect_max_delay_ms = 1000;
  // start at a random server (ZK-client convention; set false when the
  // caller pre-ordered the list for deterministic load balance)
  bool randomize_start = true;
  LogLevel log_level = LogLevel::Warn;
};

enum class SessionState { Connecting, Connected, Expired, Closed };

struct SessionEvent {
  enum class Type { Connected, Disconnected, Expired, ConnectAttempt, Closed };
  Type type;
  int64_t session_id = 0;
  int64_t attempt = 0;   // ConnectAttempt
  int64_t delay_ms = 0;  // ConnectAttempt: delay before this retry
};

const char* session_event_name(SessionEvent::Type t);

class ZkClient {
 public:
  using StatCallback = std::function<void(int rc, const Stat& stat)>;
  using StringCallback = std::function<void(int rc, const std::string& value)>;
  using VoidCallback = std::function<void(int rc)>;
  using DataCallback = std::function<void(int rc, const std::string& data, const Stat& stat)>;
  using ChildrenCallback = std::function<void(int rc, const std::vector<std::string>& children)>;
  using EventCallback = std::function<void(const SessionEvent&)>;
  using WatchCallback = std::function<void(const WatcherEvent&)>;

  ZkClient(ZkClientConfig cfg, Logger log);
  ~ZkClient();

  ZkClient(const ZkClient&) = delete;
  ZkClient& operator=(const ZkClient&) = delete;

  // Begin connecting (starts the loop thread). Non-blocking.
  void start();
  // Block until the first CONNECTED (true) or abort/expiry/attempt-exhaustion
  // (false). timeout_ms < 0 waits forever (the reference retries forever:
  // lib/zk.js:97-101).
  bool wait_connected(int64_t timeout_ms = -1);
  // Abort a pending initial connect (reference: retry.stop(), lib/zk.js:122-125).
  void abort_connect();
  // Graceful close: closeSession op, then tear down the loop thread.
  void close();

  SessionState state() const;
  int64_t session_id() const;
  int64_t session_timeout_ms() const;  // negotiated
  std::string to_string() const;       // connection description for logs (§2.4)

  // Event observation: callback (invoked on loop thread) and/or polling queue.
  void set_event_callback(EventCallback cb);
  void set_watch_callback(WatchCallback cb);
  std::vector<SessionEvent> poll_events();  // drains the queue

  // ---- async ops (callbacks on loop thread) ----
  void acreate(const std::string& path, const std::string& data, int32_t flags, StringCallback cb);
  void adelete(const std::string& path, int32_t version, VoidCallback cb);
  void aexists(const std::string& path, bool watch, StatCallback cb);
  void aget(const std::string& path, bool watch, DataCallback cb);
  void aset(const std::string& path, const std::string& data, int32_t version, StatCallback cb);
  void achildren(const std::string& path, bool watch, ChildrenCallback cb);

  // ---- sync ops (any thread but the loop thread) ----
  // watch=true registers a one-shot watch on the path (delivered through the
  // watch callback and the poll_watches() queue), Binder-style.
  int create(const std::string& path, const std::string& data, int32_t flags, std::string* created_path = nullptr);
  int del(const std::string& path, int32_t version = -1);
  int exists(const std::string& path, Stat* stat = nullptr, bool watch = false);
  int get(const std::string& path, std::string* data, Stat* stat = nullptr, bool watch = false);
  int set(const std::string& path, const std::string& data, int32_t version = -1, Stat* stat = nullptr);
  int get_children(const std::string& path, std::vector<std::string>* children, bool watch = false);

  // drained queue of watch notifications received so far
  std::vector<WatcherEvent> poll_watches();

  // ---- zkplus-surface verbs ----
  // put: create-or-overwrite persistent node (reference lib/register.js:62)
  int put(const std::string& path, const std::string& data);
  // mkdirp: recursive persistent create, ok-if-exists (lib/register.js:116)
  int mkdirp(const std::string& path);
  // unlink: delete any version; NO_NODE surfaces as kZNoNode (lib/register.js:87)
  int unlink(const std::string& path);

  // ---- pipelined batches (the hot path) ----
  // Submit the whole batch in one flush; returns per-item rcs in order.
  std::vector<int> create_many(const std::vector<std::string>& paths, const std::vector<std::string>& datas,
                               int32_t flags);
  std::vector<int> delete_many(const std::vector<std::string>& paths);
  std::vector<int> exists_many(const std::vector<std::string>& paths, std::vector<Stat>* stats = nullptr);

  // Heterogeneous pipelined batch (delete/create mixes). ZooKeeper processes
  // a session's requests strictly in order, so a stage sequence whose only
  // dependency is that ordering (cleanup unlinks → parent creates →
  // ephemeral creates) can ship as ONE submission — one round trip instead
  // of one per stage.
  struct MixedOp {
    int32_t op = kOpCreate;  // kOpCreate or kOpDelete
    std::string path;
    std::string data;        // create only
    int32_t flags = 0;       // create only
  };
  std::vector<int> submit_mixed(const std::vector<MixedOp>& ops);

  // ZooKeeper multi (op 14): apply the create/delete sequence as ONE atomic
  // transaction. Returns the transaction rc (kZOk = all applied) and fills
  // per_op with each op's result (the failing op's error; others
  // RuntimeInconsistency on abort).
  int multi(const std::vector<MixedOp>& ops, std::vector<int>* per_op = nullptr);

  // Pre-serialized batch: the full framed request stream built once, with
  // per-request xid placeholders patched at submission time. For repeated
  // identical batches (re-register cycles, heartbeats) the per-op cost
  // drops to an xid patch + one buffer append.
  struct BatchTemplate {
    std::string buf;                  // framed requests, xids zeroed
    std::vector<size_t> xid_offsets;  // offset of each request's xid field
    std::vector<int32_t> ops;         // opcode per request (reply matching)
  };
  static BatchTemplate make_template(const std::vector<MixedOp>& ops);
  static BatchTemplate make_exists_template(const std::vector<std::string>& paths);
  // Patches fresh xids into `t` (mutated in place; one submission at a time
  // per template) and pipelines the whole stream. Returns per-request rcs.
  std::vector<int> submit_template(BatchTemplate& t);
  // exists_many over a template with heartbeat-style bounded retry
  int heartbeat_template(BatchTemplate& t, const RetryPolicy& retry, int64_t* rtt_us);

  // App-level heartbeat: parallel exists over `nodes` with bounded retry
  // (reference lib/zk.js:21-44: ≤5 attempts, 1 s → 30 s). Returns kZOk when a
  // round succeeded for every node; fills rtt_us with the successful round's
  // wall time when non-null.
  int heartbeat(const std::vector<std::string>& nodes, const RetryPolicy& retry = RetryPolicy{},
                int64_t* rtt_us = nullptr);

 private:
  struct Impl;
  std::unique_ptr<Impl> impl_;
};

}  // namespace zk
}  // namespace registrar
