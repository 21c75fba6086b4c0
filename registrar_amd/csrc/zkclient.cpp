// zkclient.cpp — native ZooKeeper client implementation (see zkclient.hpp).
#include "zkclient.hpp"

#include <arpa/inet.h>
#include <fcntl.h>
#include <netdb.h>
#include <netinet/in.h>
#include <netinet/tcp.h>
#include <sys/socket.h>
#include <unistd.h>

#include <algorithm>
#include <chrono>
#include <future>
#include <thread>
#include <unordered_set>

namespace registrar {
namespace zk {

const char* session_event_name(SessionEvent::Type t) {
  switch (t) {
    case SessionEvent::Type::Connected:
      return "connect";
    case SessionEvent::Type::Disconnected:
      return "close";
    case SessionEvent::Type::Expired:
      return "session_expired";
    case SessionEvent::Type::ConnectAttempt:
      return "attempt";
    case SessionEvent::Type::Closed:
      return "closed";
  }
  return "?";
}

namespace {
void set_nonblock(int fd) {
  int fl = fcntl(fd, F_GETFL, 0);
  fcntl(fd, F_SETFL, fl | O_NONBLOCK);
}
void set_nodelay(int fd) {
  int one = 1;
  setsockopt(fd, IPPROTO_TCP, TCP_NODELAY, &one, sizeof(one));
}
}  // namespace

struct ZkClient::Impl {
  enum class Phase { Idle, TcpConnecting, Handshaking, Ready, Stopped };

  struct Pending {
    int32_t xid;
    int32_t op;
    // rc + body reader (null unless rc==kZOk and the op has a body)
    std::function<void(int rc, JuteReader* r)> done;
  };

  ZkClientConfig cfg;
  Logger log;
  EventLoop loop;
  std::thread thread;
  std::atomic<bool> started{false};
  std::atomic<bool> closed{false};

  // ---- loop-thread-only state ----
  Phase phase = Phase::Idle;
  int fd = -1;
  std::string inbuf;
  size_t inpos = 0;
  std::string outbuf;
  bool flush_scheduled = false;
  std::deque<Pending> pending;
  int32_t next_xid = 1;
  size_t server_idx = 0;
  Backoff connect_backoff;    // pre-session (1 s → 90 s, infinite by default)
  Backoff reconnect_backoff;  // session-preserving (10 ms → 1 s)
  bool have_session = false;
  int64_t sid = 0;
  std::string passwd;
  int negotiated_timeout = 0;
  int64_t last_zxid = 0;
  int64_t last_recv = 0;
  EventLoop::TimerId conn_timer = 0;
  EventLoop::TimerId ping_timer = 0;
  EventLoop::TimerId retry_timer = 0;
  std::string current_server;
  // armed one-shot watches (loop thread); re-sent via setWatches after a
  // same-session reconnect so watches survive connection loss
  std::unordered_set<std::string> data_watch_paths;
  std::unordered_set<std::string> exist_watch_paths;
  std::unordered_set<std::string> child_watch_paths;

  // ---- shared state ----
  std::atomic<SessionState> state{SessionState::Connecting};
  std::atomic<int64_t> session_id_pub{0};
  std::atomic<int64_t> negotiated_pub{0};
  std::mutex ev_mu;
  std::condition_variable ev_cv;
  std::vector<SessionEvent> ev_queue;
  EventCallback ev_cb;
  WatchCallback watch_cb;
  std::mutex watch_mu;
  std::vector<WatcherEvent> watch_queue;
  bool settled = false;  // first connect resolved (ok or fail)

  Impl(ZkClientConfig c, Logger l) : cfg(std::move(c)), log(l.child("zookeeper")) {
    log.set_level(cfg.log_level);
    // spread clients across the ensemble: random starting server (real ZK
    // clients shuffle the connect string for the same reason)
    if (cfg.randomize_start)
      server_idx = static_cast<size_t>((static_cast<uint64_t>(getpid()) * 0x9E3779B97F4A7C15ull +
                                        reinterpret_cast<uintptr_t>(this)) >>
                                       17);
    connect_backoff.initial_ms = cfg.connect_initial_delay_ms;
    connect_backoff.max_ms = cfg.connect_max_delay_ms;
    connect_backoff.max_attempts = cfg.connect_max_attempts;
    reconnect_backoff.initial_ms = cfg.reconnect_initial_delay_ms;
    reconnect_backoff.max_ms = cfg.reconnect_max_delay_ms;
    reconnect_backoff.max_attempts = -1;
  }

  // ---------------- events ----------------

  void emit(SessionEvent ev) {
    {
      std::lock_guard<std::mutex> g(ev_mu);
      ev_queue.push_back(ev);
      if (ev.type != SessionEvent::Type::ConnectAttempt) settled_check(ev);
    }
    ev_cv.notify_all();
    if (ev_cb) ev_cb(ev);
  }

  // ev_mu held
  void settled_check(const SessionEvent& ev) {
    if (ev.type == SessionEvent::Type::Connected || ev.type == SessionEvent::Type::Expired ||
        ev.type == SessionEvent::Type::Closed)
      settled = true;
  }

  // ---------------- connect machinery (loop thread) ----------------

  void start_connect() {
    if (phase == Phase::Stopped || closed.load()) return;
    const ServerAddr& srv = cfg.servers[server_idx % cfg.servers.size()];
    server_idx++;
    current_server = srv.host + ":" + std::to_string(srv.port);
    log.debug("connecting", {{"server", Json(current_server)}});

    struct sockaddr_in addr;
    memset(&addr, 0, sizeof(addr));
    addr.sin_family = AF_INET;
    addr.sin_port = htons(static_cast<uint16_t>(srv.port));
    if (inet_pton(AF_INET, srv.host.c_str(), &addr.sin_addr) != 1) {
      struct addrinfo hints;
      memset(&hints, 0, sizeof(hints));
      hints.ai_family = AF_INET;
      hints.ai_socktype = SOCK_STREAM;
      struct addrinfo* res = nullptr;
      if (getaddrinfo(srv.host.c_str(), nullptr, &hints, &res) != 0 || !res) {
        if (res) freeaddrinfo(res);
        on_connect_failed("resolve failed");
        return;
      }
      addr.sin_addr = reinterpret_cast<struct sockaddr_in*>(res->ai_addr)->sin_addr;
      freeaddrinfo(res);
    }

    fd = socket(AF_INET, SOCK_STREAM | SOCK_CLOEXEC, 0);
    if (fd < 0) {
      on_connect_failed("socket failed");
      return;
    }
    set_nonblock(fd);
    set_nodelay(fd);
    int rc = ::connect(fd, reinterpret_cast<struct sockaddr*>(&addr), sizeof(addr));
    phase = Phase::TcpConnecting;
    conn_timer = loop.schedule(cfg.connect_timeout_ms, [this] {
      conn_timer = 0;
      on_connect_failed("connect timeout");
    });
    if (rc == 0) {
      install_fd();
      on_tcp_connected();
    } else if (errno == EINPROGRESS) {
      install_fd();
      loop.mod_fd(fd, EPOLLOUT);
    } else {
      on_connect_failed("connect errno");
    }
  }

  void install_fd() {
    int f = fd;
    loop.add_fd(fd, EPOLLIN, [this, f](uint32_t ev) {
      if (f != fd) return;  // stale event for an already-replaced socket
      on_socket_event(ev);
    });
  }

  void on_socket_event(uint32_t ev) {
    if (phase == Phase::TcpConnecting) {
      if (ev & (EPOLLERR | EPOLLHUP)) {
        on_connect_failed("tcp error");
        return;
      }
      if (ev & EPOLLOUT) {
        int err = 0;
        socklen_t len = sizeof(err);
        getsockopt(fd, SOL_SOCKET, SO_ERROR, &err, &len);
        if (err != 0) {
          on_connect_failed("tcp error");
          return;
        }
        loop.mod_fd(fd, EPOLLIN);
        on_tcp_connected();
      }
      return;
    }
    if (ev & (EPOLLERR | EPOLLHUP)) {
      on_connection_lost("socket error");
      return;
    }
    if (ev & EPOLLIN) {
      if (!read_frames()) return;
    }
    if (ev & EPOLLOUT) flush(true);
  }

  void on_tcp_connected() {
    phase = Phase::Handshaking;
    ConnectRequest req;
    req.last_zxid_seen = have_session ? last_zxid : 0;
    req.time_out_ms = cfg.session_timeout_ms;
    req.session_id = have_session ? sid : 0;
    req.passwd = have_session ? passwd : std::string(16, '\0');
    std::string pkt;
    begin_packet(&pkt);
    JuteWriter w(&pkt);
    req.serialize(w);
    frame_packet(&pkt);
    outbuf += pkt;
    flush(false);
  }

  void on_connect_failed(const char* why) {
    teardown_socket();
    if (closed.load()) return;
    Backoff& bo = have_session ? reconnect_backoff : connect_backoff;
    if (bo.exhausted()) {
      log.error("zookeeper: connect attempts exhausted", {{"reason", Json(why)}});
      phase = Phase::Stopped;
      state.store(SessionState::Closed);
      emit({SessionEvent::Type::Closed, 0, 0, 0});
      return;
    }
    int64_t delay = bo.next_delay();
    int64_t attempt = bo.attempt;  // 1-based after next_delay()
    // per-attempt log level escalates info→warn→error (reference lib/zk.js:104-119)
    std::vector<JsonMember> fields{{"attempt", Json(attempt)}, {"delay", Json(delay)}, {"reason", Json(why)},
                                   {"server", Json(current_server)}};
    if (attempt <= 1)
      log.info("zookeeper: connection attempted (failed)", fields);
    else if (attempt < 5)
      log.warn("zookeeper: connection attempted (failed)", fields);
    else
      log.error("zookeeper: connection attempted (failed)", fields);
    emit({SessionEvent::Type::ConnectAttempt, 0, attempt, delay});
    phase = Phase::Idle;
    retry_timer = loop.schedule(delay, [this] {
      retry_timer = 0;
      start_connect();
    });
  }

  void on_connection_lost(const char* why) {
    log.warn("zookeeper: disconnected", {{"reason", Json(why)}});
    teardown_socket();
    fail_all_pending(kZConnectionLoss);
    if (closed.load()) return;
    state.store(SessionState::Connecting);
    emit({SessionEvent::Type::Disconnected, sid, 0, 0});
    reconnect_backoff.reset();
    phase = Phase::Idle;
    start_connect();
  }

  void teardown_socket() {
    if (conn_timer) {
      loop.cancel(conn_timer);
      conn_timer = 0;
    }
    if (ping_timer) {
      loop.cancel(ping_timer);
      ping_timer = 0;
    }
    if (fd >= 0) {
      loop.del_fd(fd);
      ::close(fd);
      fd = -1;
    }
    inbuf.clear();
    inpos = 0;
    outbuf.clear();
    flush_scheduled = false;
  }

  void fail_all_pending(int rc) {
    std::deque<Pending> q;
    q.swap(pending);
    for (auto& p : q)
      if (p.done) p.done(rc, nullptr);
  }

  // ---------------- frame processing ----------------

  bool read_frames() {
    bool eof = false;
    char buf[65536];
    while (true) {
      ssize_t n = read(fd, buf, sizeof(buf));
      if (n > 0) {
        inbuf.append(buf, static_cast<size_t>(n));
      } else if (n == 0) {
        // process buffered frames first: the server sends its final frames
        // (e.g. an expired-handshake ConnectResponse) right before closing
        eof = true;
        break;
      } else {
        if (errno == EAGAIN || errno == EWOULDBLOCK) break;
        if (errno == EINTR) continue;
        eof = true;
        break;
      }
    }
    last_recv = now_ms();
    while (true) {
      size_t avail = inbuf.size() - inpos;
      if (avail < 4) break;
      const unsigned char* p = reinterpret_cast<const unsigned char*>(inbuf.data() + inpos);
      uint32_t len = (static_cast<uint32_t>(p[0]) << 24) | (static_cast<uint32_t>(p[1]) << 16) |
                     (static_cast<uint32_t>(p[2]) << 8) | static_cast<uint32_t>(p[3]);
      if (len > 64 * 1024 * 1024) {
        on_connection_lost("oversized frame");
        return false;
      }
      if (avail < 4 + len) break;
      // parse in place — no per-reply copy; handlers never touch inbuf
      const char* body = inbuf.data() + inpos + 4;
      inpos += 4 + len;
      if (!handle_frame(body, len)) return false;  // handler may reconnect/teardown
    }
    if (inpos > 0 && fd >= 0) {
      inbuf.erase(0, inpos);
      inpos = 0;
    }
    if (eof) {
      // a handler may already have torn the socket down (expiry path)
      if (fd >= 0) on_connection_lost("eof");
      return false;
    }
    return true;
  }

  bool handle_frame(const char* body, size_t len) {
    try {
      JuteReader r(body, len);
      if (phase == Phase::Handshaking) {
        ConnectResponse resp;
        resp.deserialize(r);
        if (conn_timer) {
          loop.cancel(conn_timer);
          conn_timer = 0;
        }
        if (resp.session_id == 0 || resp.time_out_ms <= 0) {
          if (have_session) {
            log.warn("zookeeper: session expired by server", {{"session", Json(sid)}});
            teardown_socket();
            fail_all_pending(kZSessionExpired);
            data_watch_paths.clear();
            exist_watch_paths.clear();
            child_watch_paths.clear();
            phase = Phase::Stopped;
            state.store(SessionState::Expired);
            emit({SessionEvent::Type::Expired, sid, 0, 0});
          } else {
            on_connect_failed("handshake rejected");
          }
          return false;
        }
        bool reconnected = have_session;
        have_session = true;
        sid = resp.session_id;
        passwd = resp.passwd;
        negotiated_timeout = resp.time_out_ms;
        session_id_pub.store(sid);
        negotiated_pub.store(negotiated_timeout);
        phase = Phase::Ready;
        connect_backoff.reset();
        reconnect_backoff.reset();
        last_recv = now_ms();
        if (reconnected &&
            (!data_watch_paths.empty() || !exist_watch_paths.empty() || !child_watch_paths.empty())) {
          // re-arm watches; the server fires synthetic events for changes
          // missed past last_zxid (ZooKeeper setWatches semantics)
          SetWatchesRequest sw;
          sw.relative_zxid = last_zxid;
          sw.data_watches.assign(data_watch_paths.begin(), data_watch_paths.end());
          sw.exist_watches.assign(exist_watch_paths.begin(), exist_watch_paths.end());
          sw.child_watches.assign(child_watch_paths.begin(), child_watch_paths.end());
          submit(kXidSetWatches, kOpSetWatches, [sw](JuteWriter& w) { sw.serialize(w); }, nullptr);
        }
        schedule_ping();
        state.store(SessionState::Connected);
        log.info(reconnected ? "zookeeper: reconnected" : "ZK: connected",
                 {{"server", Json(current_server)},
                  {"session", Json(sid)},
                  {"timeout_ms", Json(static_cast<int64_t>(negotiated_timeout))}});
        emit({SessionEvent::Type::Connected, sid, 0, 0});
        return true;
      }

      ReplyHeader hdr;
      hdr.deserialize(r);
      if (hdr.zxid > 0) last_zxid = hdr.zxid;
      if (hdr.xid == kXidWatcherEvent) {
        WatcherEvent ev;
        ev.deserialize(r);
        log.debug("watch event", {{"path", Json(ev.path)}, {"type", Json(static_cast<int64_t>(ev.type))}});
        // one-shot consumption matches the event TYPE (ZooKeeper semantics):
        // Created/DataChanged consume data+exist watches only, ChildrenChanged
        // consumes only the child watch, Deleted consumes all three. Erasing
        // unrelated sets would drop still-armed watches from setWatches re-arm.
        switch (ev.type) {
          case kEventNodeChildrenChanged:
            child_watch_paths.erase(ev.path);
            break;
          case kEventNodeDeleted:
            data_watch_paths.erase(ev.path);
            exist_watch_paths.erase(ev.path);
            child_watch_paths.erase(ev.path);
            break;
          default:  // NodeCreated / NodeDataChanged
            data_watch_paths.erase(ev.path);
            exist_watch_paths.erase(ev.path);
            break;
        }
        {
          std::lock_guard<std::mutex> g(watch_mu);
          watch_queue.push_back(ev);
        }
        if (watch_cb) watch_cb(ev);
        return true;
      }
      if (pending.empty()) {
        on_connection_lost("unexpected reply");
        return false;
      }
      Pending p = std::move(pending.front());
      pending.pop_front();
      if (p.xid != hdr.xid) {
        log.error("zookeeper: xid mismatch", {{"expected", Json(static_cast<int64_t>(p.xid))},
                                              {"got", Json(static_cast<int64_t>(hdr.xid))}});
        if (p.done) p.done(kZConnectionLoss, nullptr);
        on_connection_lost("xid mismatch");
        return false;
      }
      if (p.done) {
        // the reader is passed even on error replies: multi responses carry
        // per-op error results in the body (callbacks gate on rc themselves)
        p.done(hdr.err, &r);
      }
      return true;
    } catch (const std::exception& e) {
      log.error("zookeeper: malformed frame", {{"err", Json(e.what())}});
      on_connection_lost("malformed frame");
      return false;
    }
  }

  // ---------------- ping ----------------

  void schedule_ping() {
    int64_t interval = std::max<int64_t>(negotiated_timeout / 3, 100);
    ping_timer = loop.schedule(interval, [this] {
      ping_timer = 0;
      if (phase != Phase::Ready) return;
      // server silent for 2/3 of the session timeout ⇒ presume dead, move on
      if (now_ms() - last_recv > negotiated_timeout * 2 / 3) {
        on_connection_lost("ping timeout");
        return;
      }
      submit(kXidPing, kOpPing, [](JuteWriter&) {}, nullptr);
      schedule_ping();
    });
  }

  // ---------------- request submission (loop thread) ----------------

  template <typename SerFn>
  void submit(int32_t xid, int32_t op, SerFn serialize_body, std::function<void(int, JuteReader*)> done) {
    if (phase != Phase::Ready) {
      if (done) done(phase == Phase::Stopped ? kZSessionExpired : kZConnectionLoss, nullptr);
      return;
    }
    // serialize straight into the session output buffer (framed in place);
    // the whole drain cycle flushes as a few large writes
    size_t start = outbuf.size();
    outbuf.append(4, '\0');
    JuteWriter w(&outbuf);
    RequestHeader hdr;
    hdr.xid = xid;
    hdr.type = op;
    hdr.serialize(w);
    serialize_body(w);
    uint32_t n = static_cast<uint32_t>(outbuf.size() - start - 4);
    outbuf[start] = static_cast<char>(n >> 24);
    outbuf[start + 1] = static_cast<char>(n >> 16);
    outbuf[start + 2] = static_cast<char>(n >> 8);
    outbuf[start + 3] = static_cast<char>(n);
    pending.push_back(Pending{xid, op, std::move(done)});
    schedule_flush();
  }

  template <typename SerFn>
  void submit_op(int32_t op, SerFn serialize_body, std::function<void(int, JuteReader*)> done) {
    submit(next_xid++, op, serialize_body, std::move(done));
  }

  // Batch all writes queued in this loop-drain cycle into one flush: every
  // async op posted from other threads lands in the same drain, so a
  // register() of 1k nodes goes out as a handful of large TCP writes instead
  // of 1k small ones.
  void schedule_flush() {
    if (flush_scheduled) return;
    flush_scheduled = true;
    loop.schedule(0, [this] {
      flush_scheduled = false;
      flush(false);
    });
  }

  void flush(bool from_epollout) {
    if (fd < 0) return;
    size_t off = 0;
    while (off < outbuf.size()) {
      ssize_t n = send(fd, outbuf.data() + off, outbuf.size() - off, MSG_NOSIGNAL);
      if (n > 0) {
        off += static_cast<size_t>(n);
      } else if (n < 0 && (errno == EAGAIN || errno == EWOULDBLOCK)) {
        break;
      } else if (n < 0 && errno == EINTR) {
        continue;
      } else {
        on_connection_lost("write error");
        return;
      }
    }
    outbuf.erase(0, off);
    if (!outbuf.empty()) {
      loop.mod_fd(fd, EPOLLIN | EPOLLOUT);
    } else if (from_epollout) {
      loop.mod_fd(fd, EPOLLIN);
    }
  }
};

// ---------------- public API ----------------

ZkClient::ZkClient(ZkClientConfig cfg, Logger log) : impl_(std::make_unique<Impl>(std::move(cfg), log)) {
  if (impl_->cfg.servers.empty()) throw std::runtime_error("ZkClient: options.servers empty");
}

ZkClient::~ZkClient() {
  try {
    close();
  } catch (...) {
  }
}

void ZkClient::start() {
  if (impl_->started.exchange(true)) return;
  impl_->thread = std::thread([this] {
    impl_->loop.post([this] { impl_->start_connect(); });
    impl_->loop.run();
  });
}

bool ZkClient::wait_connected(int64_t timeout_ms) {
  std::unique_lock<std::mutex> g(impl_->ev_mu);
  auto pred = [this] { return impl_->settled; };
  if (timeout_ms < 0) {
    impl_->ev_cv.wait(g, pred);
  } else {
    if (!impl_->ev_cv.wait_for(g, std::chrono::milliseconds(timeout_ms), pred)) return false;
  }
  return impl_->state.load() == SessionState::Connected;
}

void ZkClient::abort_connect() {
  impl_->closed.store(true);
  impl_->loop.post([this] {
    if (impl_->retry_timer) {
      impl_->loop.cancel(impl_->retry_timer);
      impl_->retry_timer = 0;
    }
    impl_->teardown_socket();
    impl_->fail_all_pending(zk::kZConnectionLoss);
    impl_->phase = Impl::Phase::Stopped;
    impl_->state.store(SessionState::Closed);
    impl_->emit({SessionEvent::Type::Closed, 0, 0, 0});
  });
}

void ZkClient::close() {
  if (!impl_->started.load()) return;
  if (!impl_->closed.exchange(true)) {
    std::promise<void> done;
    impl_->loop.post([this, &done] {
      if (impl_->phase == Impl::Phase::Ready) {
        // best-effort closeSession then synchronous drain
        impl_->submit(impl_->next_xid++, kOpCloseSession, [](JuteWriter&) {}, nullptr);
        impl_->flush(false);
      }
      impl_->fail_all_pending(kZConnectionLoss);
      if (impl_->retry_timer) impl_->loop.cancel(impl_->retry_timer);
      impl_->teardown_socket();
      impl_->phase = Impl::Phase::Stopped;
      impl_->state.store(SessionState::Closed);
      impl_->emit({SessionEvent::Type::Closed, impl_->sid, 0, 0});
      done.set_value();
    });
    done.get_future().wait();
  }
  impl_->loop.stop();
  if (impl_->thread.joinable()) impl_->thread.join();
  impl_->started.store(false);
}

SessionState ZkClient::state() const { return impl_->state.load(); }
int64_t ZkClient::session_id() const { return impl_->session_id_pub.load(); }
int64_t ZkClient::session_timeout_ms() const { return impl_->negotiated_pub.load(); }

std::string ZkClient::to_string() const {
  std::string s = "zk://";
  for (size_t i = 0; i < impl_->cfg.servers.size(); i++) {
    if (i) s += ',';
    s += impl_->cfg.servers[i].host + ":" + std::to_string(impl_->cfg.servers[i].port);
  }
  char buf[32];
  snprintf(buf, sizeof(buf), " (session 0x%llx)", static_cast<unsigned long long>(session_id()));
  s += buf;
  return s;
}

void ZkClient::set_event_callback(EventCallback cb) { impl_->ev_cb = std::move(cb); }
void ZkClient::set_watch_callback(WatchCallback cb) { impl_->watch_cb = std::move(cb); }

std::vector<WatcherEvent> ZkClient::poll_watches() {
  std::lock_guard<std::mutex> g(impl_->watch_mu);
  std::vector<WatcherEvent> out;
  out.swap(impl_->watch_queue);
  return out;
}

std::vector<SessionEvent> ZkClient::poll_events() {
  std::lock_guard<std::mutex> g(impl_->ev_mu);
  std::vector<SessionEvent> out;
  out.swap(impl_->ev_queue);
  return out;
}

// ---- async ops ----

void ZkClient::acreate(const std::string& path, const std::string& data, int32_t flags, StringCallback cb) {
  impl_->loop.post([this, path, data, flags, cb = std::move(cb)] {
    CreateRequest req;
    req.path = path;
    req.data = data;
    req.flags = flags;
    impl_->submit_op(kOpCreate, [req](JuteWriter& w) { req.serialize(w); },
                     [cb](int rc, JuteReader* r) {
                       if (!cb) return;
                       std::string created;
                       if (rc == kZOk && r) {
                         CreateResponse resp;
                         resp.deserialize(*r);
                         created = resp.path;
                       }
                       cb(rc, created);
                     });
  });
}

void ZkClient::adelete(const std::string& path, int32_t version, VoidCallback cb) {
  impl_->loop.post([this, path, version, cb = std::move(cb)] {
    DeleteRequest req;
    req.path = path;
    req.version = version;
    impl_->submit_op(kOpDelete, [req](JuteWriter& w) { req.serialize(w); },
                     [cb](int rc, JuteReader*) {
                       if (cb) cb(rc);
                     });
  });
}

void ZkClient::aexists(const std::string& path, bool watch, StatCallback cb) {
  impl_->loop.post([this, path, watch, cb = std::move(cb)] {
    ExistsRequest req;
    req.path = path;
    req.watch = watch;
    Impl* impl = impl_.get();
    impl_->submit_op(kOpExists, [req](JuteWriter& w) { req.serialize(w); },
                     [cb, impl, req](int rc, JuteReader* r) {
                       if (req.watch) {
                         // exists registers a watch on present AND absent
                         // nodes (created-event semantics)
                         if (rc == kZOk)
                           impl->data_watch_paths.insert(req.path);
                         else if (rc == kZNoNode)
                           impl->exist_watch_paths.insert(req.path);
                       }
                       if (!cb) return;
                       Stat st;
                       if (rc == kZOk && r) {
                         ExistsResponse resp;
                         resp.deserialize(*r);
                         st = resp.stat;
                       }
                       cb(rc, st);
                     });
  });
}

void ZkClient::aget(const std::string& path, bool watch, DataCallback cb) {
  impl_->loop.post([this, path, watch, cb = std::move(cb)] {
    GetDataRequest req;
    req.path = path;
    req.watch = watch;
    Impl* impl = impl_.get();
    impl_->submit_op(kOpGetData, [req](JuteWriter& w) { req.serialize(w); },
                     [cb, impl, req](int rc, JuteReader* r) {
                       if (req.watch && rc == kZOk) impl->data_watch_paths.insert(req.path);
                       if (!cb) return;
                       std::string data;
                       Stat st;
                       if (rc == kZOk && r) {
                         GetDataResponse resp;
                         resp.deserialize(*r);
                         data = resp.data;
                         st = resp.stat;
                       }
                       cb(rc, data, st);
                     });
  });
}

void ZkClient::aset(const std::string& path, const std::string& data, int32_t version, StatCallback cb) {
  impl_->loop.post([this, path, data, version, cb = std::move(cb)] {
    SetDataRequest req;
    req.path = path;
    req.data = data;
    req.version = version;
    impl_->submit_op(kOpSetData, [req](JuteWriter& w) { req.serialize(w); },
                     [cb](int rc, JuteReader* r) {
                       if (!cb) return;
                       Stat st;
                       if (rc == kZOk && r) {
                         SetDataResponse resp;
                         resp.deserialize(*r);
                         st = resp.stat;
                       }
                       cb(rc, st);
                     });
  });
}

void ZkClient::achildren(const std::string& path, bool watch, ChildrenCallback cb) {
  impl_->loop.post([this, path, watch, cb = std::move(cb)] {
    GetChildrenRequest req;
    req.path = path;
    req.watch = watch;
    Impl* impl = impl_.get();
    impl_->submit_op(kOpGetChildren, [req](JuteWriter& w) { req.serialize(w); },
                     [cb, impl, req](int rc, JuteReader* r) {
                       if (req.watch && rc == kZOk) impl->child_watch_paths.insert(req.path);
                       if (!cb) return;
                       std::vector<std::string> children;
                       if (rc == kZOk && r) {
                         GetChildrenResponse resp;
                         resp.deserialize(*r);
                         children = std::move(resp.children);
                       }
                       cb(rc, children);
                     });
  });
}

// ---- sync wrappers ----

int ZkClient::create(const std::string& path, const std::string& data, int32_t flags, std::string* created_path) {
  std::promise<std::pair<int, std::string>> p;
  acreate(path, data, flags, [&p](int rc, const std::string& cp) { p.set_value({rc, cp}); });
  auto [rc, cp] = p.get_future().get();
  if (created_path) *created_path = cp;
  return rc;
}

int ZkClient::del(const std::string& path, int32_t version) {
  std::promise<int> p;
  adelete(path, version, [&p](int rc) { p.set_value(rc); });
  return p.get_future().get();
}

int ZkClient::exists(const std::string& path, Stat* stat, bool watch) {
  std::promise<std::pair<int, Stat>> p;
  aexists(path, watch, [&p](int rc, const Stat& st) { p.set_value({rc, st}); });
  auto [rc, st] = p.get_future().get();
  if (stat) *stat = st;
  return rc;
}

int ZkClient::get(const std::string& path, std::string* data, Stat* stat, bool watch) {
  std::promise<int> p;
  aget(path, watch, [&](int rc, const std::string& d, const Stat& st) {
    if (data) *data = d;
    if (stat) *stat = st;
    p.set_value(rc);
  });
  return p.get_future().get();
}

int ZkClient::set(const std::string& path, const std::string& data, int32_t version, Stat* stat) {
  std::promise<int> p;
  aset(path, data, version, [&](int rc, const Stat& st) {
    if (stat) *stat = st;
    p.set_value(rc);
  });
  return p.get_future().get();
}

int ZkClient::get_children(const std::string& path, std::vector<std::string>* children, bool watch) {
  std::promise<int> p;
  achildren(path, watch, [&](int rc, const std::vector<std::string>& ch) {
    if (children) *children = ch;
    p.set_value(rc);
  });
  return p.get_future().get();
}

// ---- zkplus-surface verbs ----

int ZkClient::put(const std::string& path, const std::string& data) {
  // create-or-overwrite persistent node (zkplus `put`, used for the service
  // record at lib/register.js:62)
  int rc = set(path, data, -1);
  if (rc == kZNoNode) {
    rc = create(path, data, 0);
    if (rc == kZNodeExists) rc = set(path, data, -1);  // lost the race; overwrite
  }
  return rc;
}

int ZkClient::mkdirp(const std::string& path) {
  if (path.empty() || path[0] != '/') return kZMarshallingError;
  if (path == "/") return kZOk;
  // Pipeline the whole prefix chain in one flush: in-order processing
  // guarantees each parent exists (or already existed) by the time its child
  // create is handled.
  std::vector<std::string> prefixes;
  size_t pos = 0;
  while ((pos = path.find('/', pos + 1)) != std::string::npos) prefixes.push_back(path.substr(0, pos));
  prefixes.push_back(path);
  std::vector<std::string> datas(prefixes.size());
  std::vector<int> rcs = create_many(prefixes, datas, 0);
  for (int rc : rcs)
    if (rc != kZOk && rc != kZNodeExists) return rc;
  return kZOk;
}

int ZkClient::unlink(const std::string& path) { return del(path, -1); }

// ---- pipelined batches ----

namespace {
// Blocking batch context: lives on the caller's stack for the whole batch,
// so callbacks capture a raw pointer + index (fits std::function's SSO — no
// per-op heap allocation).
struct BatchState {
  std::mutex mu;
  std::condition_variable cv;
  size_t done = 0;
  size_t total = 0;
  std::vector<int>* rcs = nullptr;
  std::vector<Stat>* stats = nullptr;

  void complete(size_t i, int rc) {
    std::lock_guard<std::mutex> g(mu);
    (*rcs)[i] = rc;
    if (++done == total) cv.notify_all();
  }
  void wait() {
    std::unique_lock<std::mutex> g(mu);
    cv.wait(g, [&] { return done == total; });
  }
};
}  // namespace

std::vector<int> ZkClient::create_many(const std::vector<std::string>& paths, const std::vector<std::string>& datas,
                                       int32_t flags) {
  size_t n = paths.size();
  std::vector<int> rcs(n, kZConnectionLoss);
  if (n == 0) return rcs;
  BatchState st;
  st.total = n;
  st.rcs = &rcs;
  BatchState* stp = &st;
  impl_->loop.post([this, &paths, &datas, flags, stp, n] {
    for (size_t i = 0; i < n; i++) {
      const std::string& path = paths[i];
      const std::string& data = datas[i];
      impl_->submit_op(kOpCreate,
                       [&path, &data, flags](JuteWriter& w) {
                         w.write_string(path);
                         w.write_buffer(data);
                         write_acl_vector(w, {ACL{}});
                         w.write_int(flags);
                       },
                       [stp, i](int rc, JuteReader*) { stp->complete(i, rc); });
    }
  });
  st.wait();
  return rcs;
}

std::vector<int> ZkClient::delete_many(const std::vector<std::string>& paths) {
  size_t n = paths.size();
  std::vector<int> rcs(n, kZConnectionLoss);
  if (n == 0) return rcs;
  BatchState st;
  st.total = n;
  st.rcs = &rcs;
  BatchState* stp = &st;
  impl_->loop.post([this, &paths, stp, n] {
    for (size_t i = 0; i < n; i++) {
      const std::string& path = paths[i];
      impl_->submit_op(kOpDelete,
                       [&path](JuteWriter& w) {
                         w.write_string(path);
                         w.write_int(-1);
                       },
                       [stp, i](int rc, JuteReader*) { stp->complete(i, rc); });
    }
  });
  st.wait();
  return rcs;
}

std::vector<int> ZkClient::exists_many(const std::vector<std::string>& paths, std::vector<Stat>* stats) {
  size_t n = paths.size();
  std::vector<int> rcs(n, kZConnectionLoss);
  if (stats) stats->assign(n, Stat{});
  if (n == 0) return rcs;
  BatchState st;
  st.total = n;
  st.rcs = &rcs;
  st.stats = stats;
  BatchState* stp = &st;
  impl_->loop.post([this, &paths, stp, n] {
    for (size_t i = 0; i < n; i++) {
      const std::string& path = paths[i];
      impl_->submit_op(kOpExists,
                       [&path](JuteWriter& w) {
                         w.write_string(path);
                         w.write_bool(false);
                       },
                       [stp, i](int rc, JuteReader* r) {
                         if (rc == kZOk && r && stp->stats) {
                           ExistsResponse resp;
                           resp.deserialize(*r);
                           (*stp->stats)[i] = resp.stat;
                         }
                         stp->complete(i, rc);
                       });
    }
  });
  st.wait();
  return rcs;
}

std::vector<int> ZkClient::submit_mixed(const std::vector<MixedOp>& ops) {
  size_t n = ops.size();
  std::vector<int> rcs(n, kZConnectionLoss);
  if (n == 0) return rcs;
  BatchState st;
  st.total = n;
  st.rcs = &rcs;
  BatchState* stp = &st;
  impl_->loop.post([this, &ops, stp, n] {
    for (size_t i = 0; i < n; i++) {
      const MixedOp& mo = ops[i];
      if (mo.op == kOpDelete) {
        impl_->submit_op(kOpDelete,
                         [&mo](JuteWriter& w) {
                           w.write_string(mo.path);
                           w.write_int(-1);
                         },
                         [stp, i](int rc, JuteReader*) { stp->complete(i, rc); });
      } else {
        impl_->submit_op(kOpCreate,
                         [&mo](JuteWriter& w) {
                           w.write_string(mo.path);
                           w.write_buffer(mo.data);
                           write_acl_vector(w, {ACL{}});
                           w.write_int(mo.flags);
                         },
                         [stp, i](int rc, JuteReader*) { stp->complete(i, rc); });
      }
    }
  });
  st.wait();
  return rcs;
}

int ZkClient::multi(const std::vector<MixedOp>& mops, std::vector<int>* per_op) {
  if (per_op) per_op->assign(mops.size(), kZConnectionLoss);
  std::promise<int> done;
  impl_->loop.post([this, &mops, per_op, &done] {
    impl_->submit_op(
        kOpMulti,
        [&mops](JuteWriter& w) {
          for (const auto& mo : mops) {
            MultiHeader mh;
            mh.type = mo.op;
            mh.done = false;
            mh.err = -1;
            mh.serialize(w);
            if (mo.op == kOpDelete) {
              w.write_string(mo.path);
              w.write_int(-1);
            } else {
              w.write_string(mo.path);
              w.write_buffer(mo.data);
              write_acl_vector(w, {ACL{}});
              w.write_int(mo.flags);
            }
          }
          MultiHeader end;
          end.serialize(w);
        },
        [per_op, &done, n = mops.size()](int rc, JuteReader* r) {
          // parse per-op results when a body is present (success or txn abort)
          if (r) {
            size_t i = 0;
            try {
              while (r->remaining() > 0 && i < n) {
                MultiHeader mh;
                mh.deserialize(*r);
                if (mh.done) break;
                int op_rc = kZOk;
                if (mh.type == -1) {
                  op_rc = r->read_int();  // ErrorResult
                } else if (mh.type == kOpCreate) {
                  r->read_string();
                } else if (mh.type == kOpSetData) {
                  Stat st;
                  st.deserialize(*r);
                }
                if (per_op && i < per_op->size()) (*per_op)[i] = op_rc;
                i++;
              }
            } catch (const std::exception&) {
            }
          }
          done.set_value(rc);
        });
  });
  return done.get_future().get();
}

ZkClient::BatchTemplate ZkClient::make_template(const std::vector<MixedOp>& mops) {
  BatchTemplate t;
  for (const auto& mo : mops) {
    size_t start = t.buf.size();
    t.buf.append(4, '\0');
    t.xid_offsets.push_back(t.buf.size());
    JuteWriter w(&t.buf);
    w.write_int(0);  // xid placeholder
    if (mo.op == kOpDelete) {
      w.write_int(kOpDelete);
      w.write_string(mo.path);
      w.write_int(-1);
    } else {
      w.write_int(kOpCreate);
      w.write_string(mo.path);
      w.write_buffer(mo.data);
      write_acl_vector(w, {ACL{}});
      w.write_int(mo.flags);
    }
    uint32_t n = static_cast<uint32_t>(t.buf.size() - start - 4);
    t.buf[start] = static_cast<char>(n >> 24);
    t.buf[start + 1] = static_cast<char>(n >> 16);
    t.buf[start + 2] = static_cast<char>(n >> 8);
    t.buf[start + 3] = static_cast<char>(n);
    t.ops.push_back(mo.op);
  }
  return t;
}

ZkClient::BatchTemplate ZkClient::make_exists_template(const std::vector<std::string>& paths) {
  BatchTemplate t;
  for (const auto& path : paths) {
    size_t start = t.buf.size();
    t.buf.append(4, '\0');
    t.xid_offsets.push_back(t.buf.size());
    JuteWriter w(&t.buf);
    w.write_int(0);  // xid placeholder
    w.write_int(kOpExists);
    w.write_string(path);
    w.write_bool(false);
    uint32_t n = static_cast<uint32_t>(t.buf.size() - start - 4);
    t.buf[start] = static_cast<char>(n >> 24);
    t.buf[start + 1] = static_cast<char>(n >> 16);
    t.buf[start + 2] = static_cast<char>(n >> 8);
    t.buf[start + 3] = static_cast<char>(n);
    t.ops.push_back(kOpExists);
  }
  return t;
}

std::vector<int> ZkClient::submit_template(BatchTemplate& t) {
  size_t n = t.xid_offsets.size();
  std::vector<int> rcs(n, kZConnectionLoss);
  if (n == 0) return rcs;
  BatchState st;
  st.total = n;
  st.rcs = &rcs;
  BatchState* stp = &st;
  impl_->loop.post([this, &t, stp, n] {
    if (impl_->phase != Impl::Phase::Ready) {
      for (size_t i = 0; i < n; i++)
        stp->complete(i, impl_->phase == Impl::Phase::Stopped ? kZSessionExpired : kZConnectionLoss);
      return;
    }
    // patch fresh xids in place, register pendings, append the whole stream
    for (size_t i = 0; i < n; i++) {
      int32_t xid = impl_->next_xid++;
      size_t off = t.xid_offsets[i];
      uint32_t u = static_cast<uint32_t>(xid);
      t.buf[off] = static_cast<char>(u >> 24);
      t.buf[off + 1] = static_cast<char>(u >> 16);
      t.buf[off + 2] = static_cast<char>(u >> 8);
      t.buf[off + 3] = static_cast<char>(u);
      impl_->pending.push_back(
          Impl::Pending{xid, t.ops[i], [stp, i](int rc, JuteReader*) { stp->complete(i, rc); }});
    }
    impl_->outbuf += t.buf;
    impl_->schedule_flush();
  });
  st.wait();
  return rcs;
}

int ZkClient::heartbeat_template(BatchTemplate& t, const RetryPolicy& retry, int64_t* rtt_us) {
  Backoff bo;
  bo.initial_ms = retry.initial_delay_ms;
  bo.max_ms = retry.max_delay_ms;
  bo.max_attempts = retry.max_attempts;
  while (true) {
    int64_t t0 = now_us();
    std::vector<int> rcs = submit_template(t);
    int rc = kZOk;
    for (int r : rcs) {
      if (r != kZOk) {
        rc = r;
        break;
      }
    }
    if (rc == kZOk) {
      if (rtt_us) *rtt_us = now_us() - t0;
      return kZOk;
    }
    SessionState st = state();
    if (rc == kZSessionExpired || st == SessionState::Expired || st == SessionState::Closed) return rc;
    if (bo.exhausted()) return rc;
    int64_t delay = bo.next_delay();
    std::unique_lock<std::mutex> g(impl_->ev_mu);
    impl_->ev_cv.wait_for(g, std::chrono::milliseconds(delay), [this] {
      SessionState s2 = impl_->state.load();
      return impl_->closed.load() || s2 == SessionState::Expired || s2 == SessionState::Closed;
    });
    if (impl_->closed.load()) return kZConnectionLoss;
  }
}

int ZkClient::heartbeat(const std::vector<std::string>& nodes, const RetryPolicy& retry, int64_t* rtt_us) {
  Backoff bo;
  bo.initial_ms = retry.initial_delay_ms;
  bo.max_ms = retry.max_delay_ms;
  bo.max_attempts = retry.max_attempts;
  while (true) {
    int64_t t0 = now_us();
    std::vector<int> rcs = exists_many(nodes, nullptr);
    int rc = kZOk;
    for (int r : rcs) {
      if (r != kZOk) {
        rc = r;
        break;
      }
    }
    if (rc == kZOk) {
      if (rtt_us) *rtt_us = now_us() - t0;
      return kZOk;
    }
    // terminal states: retrying cannot help, and sleeping here would stall
    // the owner's expiry handling (the orchestrator must re-register NOW)
    SessionState st = state();
    if (rc == kZSessionExpired || st == SessionState::Expired || st == SessionState::Closed) return rc;
    if (bo.exhausted()) return rc;
    int64_t delay = bo.next_delay();
    impl_->log.debug("heartbeat: retrying", {{"rc", Json(error_name(rc))}, {"delay_ms", Json(delay)}});
    // interruptible: close()/expiry wake ev_cv, so shutdown never waits out
    // a long backoff
    std::unique_lock<std::mutex> g(impl_->ev_mu);
    impl_->ev_cv.wait_for(g, std::chrono::milliseconds(delay), [this] {
      SessionState st = impl_->state.load();
      return impl_->closed.load() || st == SessionState::Expired || st == SessionState::Closed;
    });
    if (impl_->closed.load()) return kZConnectionLoss;
  }
}

}  // namespace zk
}  // namespace registrar
