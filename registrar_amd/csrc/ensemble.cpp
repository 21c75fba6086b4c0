// ensemble.cpp — synthetic in-process ZooKeeper ensemble (see ensemble.hpp).
//
// Threading model (the scaling core of the whole framework): an IO-loop
// thread POOL (connections assigned round-robin, independent of which
// server socket accepted them — like ZooKeeper's selector threads) for
// parallel socket I/O, and SHARDED state so concurrent registrar processes
// (which write disjoint domain subtrees) never contend:
//
//   - znodes + their watch maps live in 64 hash shards, each under its own
//     mutex; an op locks only the shard(s) of the paths it touches (create/
//     delete lock child+parent shards in index order),
//   - sessions live under one session mutex, but per-request touch is a
//     lock-free atomic store through the connection's cached session pointer,
//   - zxid is a global atomic counter,
//   - per-connection output buffers have their own mutex; replies queue per
//     event drain and flush as a few large writes.
//
// Global lock order: shard(s) → session_mu → conns_mu → conn->out_mu.
// kill_session drains the ephemeral set with no other lock held before
// taking shard locks (see the phase comments).
#include "ensemble.hpp"

#include <arpa/inet.h>
#include <fcntl.h>
#include <netinet/in.h>
#include <netinet/tcp.h>
#include <sys/socket.h>
#include <unistd.h>

#include <algorithm>
#include <array>
#include <cstdio>
#include <future>

namespace registrar {
namespace zk {

namespace {

void set_nonblock(int fd) {
  int fl = fcntl(fd, F_GETFL, 0);
  fcntl(fd, F_SETFL, fl | O_NONBLOCK);
}

void set_nodelay(int fd) {
  int one = 1;
  setsockopt(fd, IPPROTO_TCP, TCP_NODELAY, &one, sizeof(one));
}

int64_t wall_ms() {
  struct timespec ts;
  clock_gettime(CLOCK_REALTIME, &ts);
  return static_cast<int64_t>(ts.tv_sec) * 1000 + ts.tv_nsec / 1000000;
}

std::string parent_path(const std::string& p) {
  size_t pos = p.rfind('/');
  if (pos == std::string::npos || pos == 0) return "/";
  return p.substr(0, pos);
}

std::string basename_of(const std::string& p) {
  size_t pos = p.rfind('/');
  return p.substr(pos + 1);
}

bool valid_path(const std::string& p) {
  if (p.empty() || p[0] != '/') return false;
  if (p.size() > 1 && p.back() == '/') return false;
  if (p.find("//") != std::string::npos) return false;
  return true;
}

}  // namespace

struct Ensemble::Impl {
  static constexpr size_t kShards = 64;

  struct ZNode {
    std::string data;
    Stat stat;
    // unordered: create/delete are the hot path; listings sort on demand
    std::unordered_set<std::string> children;
  };

  struct Shard {
    std::mutex mu;
    std::unordered_map<std::string, ZNode> nodes;
    std::unordered_map<std::string, std::set<int64_t>> data_watches;
    std::unordered_map<std::string, std::set<int64_t>> child_watches;
  };

  struct Session {
    int64_t id = 0;
    std::string passwd;
    int timeout_ms = 30000;
    std::atomic<int64_t> last_touch{0};  // monotonic ms, lock-free touch
    std::atomic<uint64_t> conn_id{0};    // 0 = detached
    std::atomic<bool> alive{true};       // false once kill_session starts
    std::mutex eph_mu;
    std::unordered_set<std::string> ephemerals;
  };
  using SessionPtr = std::shared_ptr<Session>;

  struct Conn {
    uint64_t id = 0;
    int fd = -1;
    size_t server_idx = 0;  // which server (port) accepted it — kill semantics
    size_t io_idx = 0;      // which IO loop owns its events
    // owner-loop-thread only:
    std::string inbuf;
    size_t inpos = 0;
    bool handshaken = false;
    SessionPtr session;  // cached after handshake (lock-free touch)
    bool flush_scheduled = false;
    // shared output state:
    std::mutex out_mu;
    std::string outbuf;
    bool epollout_armed = false;
    bool closing = false;  // close once outbuf drains
    std::atomic<bool> dead{false};
  };
  using ConnPtr = std::shared_ptr<Conn>;

  struct Server {
    int listen_fd = -1;
    int port = 0;
    bool up = false;
  };

  struct OpCounters {
    std::atomic<uint64_t> connect{0}, create{0}, del{0}, exists{0}, get_data{0}, set_data{0},
        get_children{0}, ping{0}, close_session{0}, unknown{0};
  };

  EnsembleConfig cfg;
  Logger log;
  std::vector<std::unique_ptr<EventLoop>> loops;  // IO pool (>= #servers)
  std::vector<std::thread> threads;
  std::atomic<size_t> next_io{0};
  std::atomic<bool> started{false};
  std::atomic<int> latency_ms{0};
  std::atomic<int64_t> zxid_counter{0};
  OpCounters ops;

  std::array<Shard, kShards> shards;

  mutable std::mutex session_mu;
  std::unordered_map<int64_t, SessionPtr> sessions;
  // no tombstone set needed: session ids are monotonic and never reused, so
  // an expired/closed id simply misses `sessions` ⇒ expired handshake
  int64_t next_session = 0x100000;

  mutable std::mutex conns_mu;
  std::unordered_map<uint64_t, ConnPtr> conns;
  std::atomic<uint64_t> next_conn_id{1};

  mutable std::mutex admin_mu;  // servers / leader / election
  std::vector<Server> servers;
  size_t leader_idx = 0;
  int64_t election_until = 0;  // monotonic ms; connects refused until then

  explicit Impl(EnsembleConfig c) : cfg(std::move(c)), log(Logger("zk-ensemble").child("ensemble")) {
    log.set_level(cfg.log_level);
    latency_ms.store(cfg.latency_ms);
    shard_of("/").nodes["/"] = ZNode{};
  }

  Shard& shard_of(const std::string& path) {
    return shards[std::hash<std::string>{}(path) % kShards];
  }
  size_t shard_idx(const std::string& path) const {
    return std::hash<std::string>{}(path) % kShards;
  }

  // Lock the shards of two paths without deadlock (index order; may be same).
  // Exposes the shard refs so callers hash each path exactly once.
  struct TwoShardLock {
    std::unique_lock<std::mutex> a, b;
    Shard* first;   // shard of p1
    Shard* second;  // shard of p2
    TwoShardLock(Impl& impl, const std::string& p1, const std::string& p2) {
      size_t i1 = impl.shard_idx(p1), i2 = impl.shard_idx(p2);
      first = &impl.shards[i1];
      second = &impl.shards[i2];
      if (i1 == i2) {
        a = std::unique_lock<std::mutex>(impl.shards[i1].mu);
      } else {
        a = std::unique_lock<std::mutex>(impl.shards[std::min(i1, i2)].mu);
        b = std::unique_lock<std::mutex>(impl.shards[std::max(i1, i2)].mu);
      }
    }
  };

  // ---------------- lifecycle ----------------

  void start() {
    if (started.exchange(true)) return;
    size_t n = cfg.ports.size();
    size_t io = static_cast<size_t>(cfg.io_threads);
    if (io == 0) {
      unsigned hw = std::thread::hardware_concurrency();
      io = std::max(n, std::min<size_t>(8, hw ? hw / 4 : 2));
      if (io == 0) io = 1;
    }
    loops.clear();
    for (size_t i = 0; i < io; i++) loops.push_back(std::make_unique<EventLoop>());
    {
      std::lock_guard<std::mutex> g(admin_mu);
      servers.resize(n);
      for (size_t i = 0; i < n; i++) open_listener(i, cfg.ports[i]);
    }
    for (size_t i = 0; i < io; i++) {
      threads.emplace_back([this, i] {
        if (i == 0) schedule_sweep();
        loops[i]->run();
      });
    }
  }

  void stop() {
    if (!started.load()) return;
    std::vector<std::promise<void>> done(loops.size());
    for (size_t i = 0; i < loops.size(); i++) {
      loops[i]->post([this, i, &done] {
        std::vector<ConnPtr> victims;
        {
          std::lock_guard<std::mutex> g(conns_mu);
          for (auto& kv : conns)
            if (kv.second->io_idx == i) victims.push_back(kv.second);
        }
        for (auto& c : victims) close_conn(c.get());
        {
          std::lock_guard<std::mutex> g(admin_mu);
          for (size_t srv = 0; srv < servers.size(); srv++) {
            if (srv % loops.size() == i && servers[srv].listen_fd >= 0) {
              loops[i]->del_fd(servers[srv].listen_fd);
              ::close(servers[srv].listen_fd);
              servers[srv].listen_fd = -1;
              servers[srv].up = false;
            }
          }
        }
        done[i].set_value();
      });
    }
    for (auto& d : done) d.get_future().wait();
    for (auto& l : loops) l->stop();
    for (auto& t : threads) t.join();
    threads.clear();
    started.store(false);
  }

  // admin_mu held
  void open_listener(size_t idx, int port) {
    int fd = socket(AF_INET, SOCK_STREAM | SOCK_CLOEXEC, 0);
    if (fd < 0) throw std::runtime_error("ensemble: socket() failed");
    int one = 1;
    setsockopt(fd, SOL_SOCKET, SO_REUSEADDR, &one, sizeof(one));
    struct sockaddr_in addr;
    memset(&addr, 0, sizeof(addr));
    addr.sin_family = AF_INET;
    addr.sin_port = htons(static_cast<uint16_t>(port));
    inet_pton(AF_INET, cfg.bind_host.c_str(), &addr.sin_addr);
    if (bind(fd, reinterpret_cast<struct sockaddr*>(&addr), sizeof(addr)) < 0) {
      ::close(fd);
      throw std::runtime_error("ensemble: bind failed on port " + std::to_string(port));
    }
    if (listen(fd, 512) < 0) {
      ::close(fd);
      throw std::runtime_error("ensemble: listen failed");
    }
    socklen_t alen = sizeof(addr);
    getsockname(fd, reinterpret_cast<struct sockaddr*>(&addr), &alen);
    set_nonblock(fd);
    servers[idx].listen_fd = fd;
    servers[idx].port = ntohs(addr.sin_port);
    servers[idx].up = true;
    size_t srv = idx;
    size_t lidx = idx % loops.size();
    auto install = [this, fd, srv, lidx] {
      loops[lidx]->add_fd(fd, EPOLLIN, [this, fd, srv](uint32_t) { on_accept(fd, srv); });
    };
    if (loops[lidx]->on_loop_thread()) {
      install();
    } else if (started.load() && loops[lidx]->running()) {
      loops[lidx]->post(install);
    } else {
      install();  // loop thread not started yet: direct registration is safe
    }
  }

  // ---------------- socket handling (owner loop thread) ----------------

  void on_accept(int listen_fd, size_t server_idx) {
    while (true) {
      int fd = accept4(listen_fd, nullptr, nullptr, SOCK_NONBLOCK | SOCK_CLOEXEC);
      if (fd < 0) break;
      {
        std::lock_guard<std::mutex> g(admin_mu);
        if (now_ms() < election_until) {
          // mid-election: nobody serves (BASELINE config 4 storm realism)
          ::close(fd);
          continue;
        }
        maybe_elect_leader_locked();
      }
      set_nodelay(fd);
      auto conn = std::make_shared<Conn>();
      conn->id = next_conn_id.fetch_add(1);
      conn->fd = fd;
      conn->server_idx = server_idx;
      conn->io_idx = next_io.fetch_add(1) % loops.size();
      uint64_t cid = conn->id;
      size_t io_idx = conn->io_idx;
      {
        std::lock_guard<std::mutex> g(conns_mu);
        conns[cid] = conn;
      }
      loops[io_idx]->add_fd(fd, EPOLLIN, [this, cid](uint32_t ev) { on_conn_event(cid, ev); });
    }
  }

  ConnPtr lookup(uint64_t cid) {
    std::lock_guard<std::mutex> g(conns_mu);
    auto it = conns.find(cid);
    return it == conns.end() ? nullptr : it->second;
  }

  void on_conn_event(uint64_t cid, uint32_t ev) {
    ConnPtr c = lookup(cid);
    if (!c) return;
    if (c->dead.load() || (ev & (EPOLLHUP | EPOLLERR))) {
      close_conn(c.get());
      return;
    }
    if (ev & EPOLLIN) {
      bool eof = false;
      char buf[65536];
      while (true) {
        ssize_t n = read(c->fd, buf, sizeof(buf));
        if (n > 0) {
          c->inbuf.append(buf, static_cast<size_t>(n));
        } else if (n == 0) {
          eof = true;  // process buffered frames (e.g. closeSession) first
          break;
        } else {
          if (errno == EAGAIN || errno == EWOULDBLOCK) break;
          if (errno == EINTR) continue;
          eof = true;
          break;
        }
      }
      if (!process_frames(c.get())) return;
      // replies queued during this drain go out in one flush (syscall
      // batching: 1k pipelined creates ⇒ a handful of large writes)
      flush_out(c.get());
      if (c->dead.load()) return;  // flush_out may close on error/closing
      if (eof) {
        close_conn(c.get());
        return;
      }
      return;
    }
    if (ev & EPOLLOUT) flush_out(c.get());
  }

  // returns false if the conn was closed
  bool process_frames(Conn* c) {
    while (!c->dead.load()) {
      size_t avail = c->inbuf.size() - c->inpos;
      if (avail < 4) break;
      const unsigned char* p = reinterpret_cast<const unsigned char*>(c->inbuf.data() + c->inpos);
      uint32_t len = (static_cast<uint32_t>(p[0]) << 24) | (static_cast<uint32_t>(p[1]) << 16) |
                     (static_cast<uint32_t>(p[2]) << 8) | static_cast<uint32_t>(p[3]);
      if (len > 4 * 1024 * 1024) {  // jute.maxbuffer-ish sanity cap
        close_conn(c);
        return false;
      }
      if (avail < 4 + len) break;
      const char* body = c->inbuf.data() + c->inpos + 4;
      c->inpos += 4 + len;
      if (!handle_frame(c, body, len)) {
        close_conn(c);
        return false;
      }
    }
    if (c->inpos > 0) {
      c->inbuf.erase(0, c->inpos);
      c->inpos = 0;
    }
    return true;
  }

  void touch(Conn* c) {
    if (c->session) c->session->last_touch.store(now_ms(), std::memory_order_relaxed);
  }

  // returns false ⇒ caller closes the conn (protocol error)
  bool handle_frame(Conn* c, const char* body, size_t len) {
    try {
      JuteReader r(body, len);
      if (!c->handshaken) {
        handle_connect(c, r);
        return true;
      }
      RequestHeader hdr;
      hdr.deserialize(r);
      touch(c);
      switch (hdr.type) {
        case kOpPing:
          ops.ping.fetch_add(1, std::memory_order_relaxed);
          send_reply(c, kXidPing, zxid_counter.load(std::memory_order_relaxed), kZOk, nullptr);
          break;
        case kOpCreate:
          handle_create(c, hdr.xid, r);
          break;
        case kOpCreate2:  // create + Stat in the response (3.5+ clients)
          handle_create(c, hdr.xid, r, /*with_stat=*/true);
          break;
        case kOpDelete:
          handle_delete(c, hdr.xid, r);
          break;
        case kOpExists:
          handle_exists(c, hdr.xid, r);
          break;
        case kOpGetData:
          handle_get_data(c, hdr.xid, r);
          break;
        case kOpSetData:
          handle_set_data(c, hdr.xid, r);
          break;
        case kOpGetChildren:
          handle_get_children(c, hdr.xid, r);
          break;
        case kOpGetACL:
          handle_get_acl(c, hdr.xid, r);
          break;
        case kOpSetACL:
          handle_set_acl(c, hdr.xid, r);
          break;
        case kOpGetChildren2:
          handle_get_children2(c, hdr.xid, r);
          break;
        case kOpSync:
          handle_sync(c, hdr.xid, r);
          break;
        case kOpSetWatches:
          handle_set_watches(c, hdr.xid, r);
          break;
        case kOpMulti:
          handle_multi(c, hdr.xid, r);
          break;
        case kOpCloseSession:
          handle_close_session(c, hdr.xid);
          break;
        case 100:  // auth packet (xid -4): ACLs are open here, ack and ignore
                   // so real third-party clients that always send auth work
          send_reply(c, hdr.xid, zxid_counter.load(std::memory_order_relaxed), kZOk, nullptr);
          break;
        default:
          ops.unknown.fetch_add(1, std::memory_order_relaxed);
          send_reply(c, hdr.xid, zxid_counter.load(std::memory_order_relaxed), kZSystemError, nullptr);
          break;
      }
      return true;
    } catch (const std::exception& e) {
      log.warn("ensemble: malformed frame, closing conn", {{"err", Json(e.what())}});
      return false;
    }
  }

  void handle_connect(Conn* c, JuteReader& r) {
    ConnectRequest req;
    req.deserialize(r);
    ops.connect.fetch_add(1, std::memory_order_relaxed);
    ConnectResponse resp;
    resp.has_read_only = req.has_read_only;
    bool expired_handshake = false;
    ConnPtr old_conn;
    {
      std::lock_guard<std::mutex> g(session_mu);
      if (req.session_id != 0) {
        auto sit = sessions.find(req.session_id);
        if (sit == sessions.end() || sit->second->passwd != req.passwd) {
          // unknown/expired/bad-passwd session ⇒ the canonical "expired"
          // ConnectResponse: sessionId=0, timeOut=0
          resp.session_id = 0;
          resp.time_out_ms = 0;
          expired_handshake = true;
          log.info("connect: session expired/unknown", {{"session", Json(req.session_id)}});
        } else {
          SessionPtr s = sit->second;
          uint64_t prev = s->conn_id.exchange(c->id);
          if (prev != 0) {
            std::lock_guard<std::mutex> cg(conns_mu);
            auto old = conns.find(prev);
            if (old != conns.end()) old_conn = old->second;
          }
          s->last_touch.store(now_ms());
          c->session = s;
          resp.session_id = s->id;
          resp.time_out_ms = s->timeout_ms;
          resp.passwd = s->passwd;
          log.info("connect: session re-attached", {{"session", Json(s->id)}});
        }
      } else {
        auto s = std::make_shared<Session>();
        s->id = next_session++;
        s->passwd.resize(16);
        uint64_t seed = static_cast<uint64_t>(s->id) * 0x9E3779B97F4A7C15ull + 0xD1B54A32D192ED03ull;
        for (int i = 0; i < 16; i++) {
          seed ^= seed >> 27;
          seed *= 0x94D049BB133111EBull;
          s->passwd[i] = static_cast<char>(seed >> (8 * (i % 8)));
        }
        int req_to = req.time_out_ms > 0 ? req.time_out_ms : 30000;
        s->timeout_ms = std::max(cfg.min_session_timeout_ms, std::min(cfg.max_session_timeout_ms, req_to));
        s->last_touch.store(now_ms());
        s->conn_id.store(c->id);
        c->session = s;
        resp.session_id = s->id;
        resp.time_out_ms = s->timeout_ms;
        resp.passwd = s->passwd;
        sessions[s->id] = s;
        log.info("connect: new session",
                 {{"session", Json(s->id)}, {"timeout_ms", Json(static_cast<int64_t>(s->timeout_ms))}});
      }
    }
    if (old_conn) {
      // session moved: retire the old connection without detaching
      old_conn->dead.store(true);
      post_close(old_conn);
    }
    c->handshaken = true;
    std::string pkt;
    begin_packet(&pkt);
    JuteWriter w(&pkt);
    resp.serialize(w);
    frame_packet(&pkt);
    if (expired_handshake) {
      std::lock_guard<std::mutex> og(c->out_mu);
      c->closing = true;  // expired handshake: notify, then close
    }
    send_packet(c, std::move(pkt));
  }

  int64_t sid_of(Conn* c) const { return c->session ? c->session->id : 0; }

  // --- ops ---

  void handle_create(Conn* c, int32_t xid, JuteReader& r, bool with_stat = false) {
    ops.create.fetch_add(1, std::memory_order_relaxed);
    CreateRequest req;
    req.deserialize(r);
    int32_t err = kZOk;
    CreateResponse resp;
    int64_t sid = sid_of(c);
    bool made_ephemeral = false;
    std::string created_path;
    std::string child_watch_parent;
    int64_t op_zxid = 0;   // the zxid THIS create committed at
    Stat created_stat;     // snapshot for create2 replies
    if (!valid_path(req.path) || req.path == "/") {
      err = kZMarshallingError;
    } else {
      std::string parent = parent_path(req.path);
      TwoShardLock lk(*this, req.path, parent);
      Shard& psh = *lk.second;
      auto pit = psh.nodes.find(parent);
      if (pit == psh.nodes.end()) {
        err = kZNoNode;
      } else if (pit->second.stat.ephemeral_owner != 0) {
        err = kZNoChildrenForEphemerals;
      } else {
        std::string path = req.path;
        if (req.flags & kSequence) {
          char suffix[16];
          snprintf(suffix, sizeof(suffix), "%010d", pit->second.stat.cversion);
          path += suffix;
        }
        // NB: with SEQUENCE the final path may hash to a different shard than
        // req.path; re-lock correctly in that (registrar-unused) corner
        Shard& csh = (req.flags & kSequence) ? shard_of(path) : *lk.first;
        bool same_lock = (&csh == lk.first) || (&csh == &psh);
        std::unique_lock<std::mutex> extra;
        if (!same_lock) extra = std::unique_lock<std::mutex>(csh.mu, std::try_to_lock);
        if (!same_lock && !extra.owns_lock()) {
          err = kZSystemError;  // pathological shard collision; not reachable
                                // for non-sequence creates
        } else if (auto [nit, inserted] = csh.nodes.try_emplace(path); !inserted) {
          err = kZNodeExists;
        } else {
          // single hash+probe: the node was emplaced above; fill it in
          int64_t zz = zxid_counter.fetch_add(1, std::memory_order_relaxed) + 1;
          op_zxid = zz;
          ZNode& n = nit->second;
          n.data = req.data;
          n.stat.czxid = zz;
          n.stat.mzxid = zz;
          n.stat.ctime = wall_ms();
          n.stat.mtime = n.stat.ctime;
          n.stat.data_length = static_cast<int32_t>(req.data.size());
          if (req.flags & kEphemeral) {
            n.stat.ephemeral_owner = sid;
            made_ephemeral = true;
          }
          ZNode& par = pit->second;
          par.children.insert(basename_of(path));
          par.stat.cversion++;
          par.stat.pzxid = zz;
          par.stat.num_children = static_cast<int32_t>(par.children.size());
          fire_data_watches_locked(csh, path, kEventNodeCreated);
          fire_child_watches_locked(psh, parent);
          resp.path = path;
          created_path = path;
          created_stat = n.stat;
        }
      }
      (void)child_watch_parent;
    }
    if (made_ephemeral) {
      // Registered outside the shard locks — NO global lock on this hot path.
      // kill_session flips `alive` BEFORE draining the set, so: insert seen
      // by the drain ⇒ cleaned there; insert after the drain ⇒ alive==false
      // here ⇒ roll the node back ourselves.
      SessionPtr s = c->session;
      bool ok = false;
      if (s) {
        {
          std::lock_guard<std::mutex> eg(s->eph_mu);
          s->ephemerals.insert(created_path);
        }
        ok = s->alive.load(std::memory_order_acquire);
        if (!ok) {
          std::lock_guard<std::mutex> eg(s->eph_mu);
          ok = s->ephemerals.count(created_path) == 0;  // drain took it: fine
          if (!ok) s->ephemerals.erase(created_path);
        }
      }
      if (!ok) {
        delete_node(created_path, s);
        err = kZSessionExpired;
      }
    }
    // mutating replies carry the op's OWN zxid (a fresh counter read could
    // exceed what this client actually observed — ADVICE r1)
    if (err == kZOk)
      send_reply(c, xid, op_zxid, kZOk, [&](JuteWriter& w) {
        resp.serialize(w);
        if (with_stat) created_stat.serialize(w);  // Create2Response
      });
    else
      send_reply(c, xid, zxid_counter.load(std::memory_order_relaxed), err, nullptr);
  }

  void handle_delete(Conn* c, int32_t xid, JuteReader& r) {
    ops.del.fetch_add(1, std::memory_order_relaxed);
    DeleteRequest req;
    req.deserialize(r);
    int32_t err = kZOk;
    int64_t op_zxid = 0;
    {
      // peek under the target's shard lock; full delete re-locks both shards
      Shard& sh = shard_of(req.path);
      std::unique_lock<std::mutex> lk(sh.mu);
      auto it = sh.nodes.find(req.path);
      if (it == sh.nodes.end() || req.path == "/") {
        err = kZNoNode;
      } else if (!it->second.children.empty()) {
        err = kZNotEmpty;
      } else if (req.version != -1 && req.version != it->second.stat.version) {
        err = kZBadVersion;
      }
    }
    if (err == kZOk) {
      // common case: a session unlinking its own ephemerals (cleanup step)
      if (!delete_node(req.path, c->session, &op_zxid)) err = kZNoNode;  // raced
    }
    send_reply(c, xid,
               op_zxid ? op_zxid : zxid_counter.load(std::memory_order_relaxed),
               err, nullptr);
  }

  // Full node removal: locks child+parent shards, updates parent, fires
  // watches, detaches from the owner session. Returns false if missing.
  // owner_hint avoids the session_mu map lookup on the hot path (a session
  // deleting its own ephemerals — the register-pipeline cleanup case).
  bool delete_node(const std::string& path, const SessionPtr& owner_hint = nullptr,
                   int64_t* out_zxid = nullptr) {
    int64_t owner = 0;
    {
      std::string parent = parent_path(path);
      TwoShardLock lk(*this, path, parent);
      Shard& csh = *lk.first;
      auto it = csh.nodes.find(path);
      if (it == csh.nodes.end()) return false;
      if (!it->second.children.empty()) return false;  // re-check under lock
      int64_t z = zxid_counter.fetch_add(1, std::memory_order_relaxed) + 1;
      if (out_zxid) *out_zxid = z;
      owner = it->second.stat.ephemeral_owner;
      csh.nodes.erase(it);
      Shard& psh = *lk.second;
      auto pit = psh.nodes.find(parent);
      if (pit != psh.nodes.end()) {
        pit->second.children.erase(basename_of(path));
        pit->second.stat.cversion++;
        pit->second.stat.pzxid = z;
        pit->second.stat.num_children = static_cast<int32_t>(pit->second.children.size());
        fire_child_watches_locked(psh, parent);
      }
      fire_deleted_watches_locked(csh, path);
    }
    if (owner != 0) {
      SessionPtr s;
      if (owner_hint && owner_hint->id == owner) {
        s = owner_hint;
      } else {
        std::lock_guard<std::mutex> g(session_mu);
        auto sit = sessions.find(owner);
        if (sit != sessions.end()) s = sit->second;
      }
      if (s) {
        std::lock_guard<std::mutex> eg(s->eph_mu);
        s->ephemerals.erase(path);
      }
    }
    return true;
  }

  void handle_exists(Conn* c, int32_t xid, JuteReader& r) {
    ops.exists.fetch_add(1, std::memory_order_relaxed);
    ExistsRequest req;
    req.deserialize(r);
    int32_t err = kZOk;
    ExistsResponse resp;
    {
      Shard& sh = shard_of(req.path);
      std::lock_guard<std::mutex> lk(sh.mu);
      if (req.watch) sh.data_watches[req.path].insert(sid_of(c));
      auto it = sh.nodes.find(req.path);
      if (it == sh.nodes.end())
        err = kZNoNode;
      else
        resp.stat = it->second.stat;
    }
    if (err == kZOk)
      send_reply(c, xid, zxid_counter.load(std::memory_order_relaxed), kZOk,
                 [&](JuteWriter& w) { resp.serialize(w); });
    else
      send_reply(c, xid, zxid_counter.load(std::memory_order_relaxed), err, nullptr);
  }

  void handle_get_data(Conn* c, int32_t xid, JuteReader& r) {
    ops.get_data.fetch_add(1, std::memory_order_relaxed);
    GetDataRequest req;
    req.deserialize(r);
    int32_t err = kZOk;
    GetDataResponse resp;
    {
      Shard& sh = shard_of(req.path);
      std::lock_guard<std::mutex> lk(sh.mu);
      auto it = sh.nodes.find(req.path);
      if (it == sh.nodes.end()) {
        err = kZNoNode;
      } else {
        if (req.watch) sh.data_watches[req.path].insert(sid_of(c));
        resp.data = it->second.data;
        resp.stat = it->second.stat;
      }
    }
    if (err == kZOk)
      send_reply(c, xid, zxid_counter.load(std::memory_order_relaxed), kZOk,
                 [&](JuteWriter& w) { resp.serialize(w); });
    else
      send_reply(c, xid, zxid_counter.load(std::memory_order_relaxed), err, nullptr);
  }

  void handle_set_data(Conn* c, int32_t xid, JuteReader& r) {
    ops.set_data.fetch_add(1, std::memory_order_relaxed);
    SetDataRequest req;
    req.deserialize(r);
    int32_t err = kZOk;
    SetDataResponse resp;
    int64_t op_zxid = 0;
    {
      Shard& sh = shard_of(req.path);
      std::lock_guard<std::mutex> lk(sh.mu);
      auto it = sh.nodes.find(req.path);
      if (it == sh.nodes.end()) {
        err = kZNoNode;
      } else if (req.version != -1 && req.version != it->second.stat.version) {
        err = kZBadVersion;
      } else {
        int64_t zz = zxid_counter.fetch_add(1, std::memory_order_relaxed) + 1;
        op_zxid = zz;
        it->second.data = req.data;
        it->second.stat.mzxid = zz;
        it->second.stat.mtime = wall_ms();
        it->second.stat.version++;
        it->second.stat.data_length = static_cast<int32_t>(req.data.size());
        fire_data_watches_locked(sh, req.path, kEventNodeDataChanged);
        resp.stat = it->second.stat;
      }
    }
    if (err == kZOk)
      send_reply(c, xid, op_zxid, kZOk,
                 [&](JuteWriter& w) { resp.serialize(w); });
    else
      send_reply(c, xid, zxid_counter.load(std::memory_order_relaxed), err, nullptr);
  }

  void handle_get_children(Conn* c, int32_t xid, JuteReader& r) {
    ops.get_children.fetch_add(1, std::memory_order_relaxed);
    GetChildrenRequest req;
    req.deserialize(r);
    int32_t err = kZOk;
    GetChildrenResponse resp;
    {
      Shard& sh = shard_of(req.path);
      std::lock_guard<std::mutex> lk(sh.mu);
      auto it = sh.nodes.find(req.path);
      if (it == sh.nodes.end()) {
        err = kZNoNode;
      } else {
        if (req.watch) sh.child_watches[req.path].insert(sid_of(c));
        resp.children.assign(it->second.children.begin(), it->second.children.end());
        std::sort(resp.children.begin(), resp.children.end());
      }
    }
    if (err == kZOk)
      send_reply(c, xid, zxid_counter.load(std::memory_order_relaxed), kZOk,
                 [&](JuteWriter& w) { resp.serialize(w); });
    else
      send_reply(c, xid, zxid_counter.load(std::memory_order_relaxed), err, nullptr);
  }

  // getChildren2: children + the node's Stat (3.4+ clients use it by default)
  void handle_get_children2(Conn* c, int32_t xid, JuteReader& r) {
    ops.get_children.fetch_add(1, std::memory_order_relaxed);
    GetChildrenRequest req;
    req.deserialize(r);
    int32_t err = kZOk;
    GetChildrenResponse resp;
    Stat stat;
    {
      Shard& sh = shard_of(req.path);
      std::lock_guard<std::mutex> lk(sh.mu);
      auto it = sh.nodes.find(req.path);
      if (it == sh.nodes.end()) {
        err = kZNoNode;
      } else {
        if (req.watch) sh.child_watches[req.path].insert(sid_of(c));
        resp.children.assign(it->second.children.begin(), it->second.children.end());
        std::sort(resp.children.begin(), resp.children.end());
        stat = it->second.stat;
      }
    }
    if (err == kZOk)
      send_reply(c, xid, zxid_counter.load(std::memory_order_relaxed), kZOk, [&](JuteWriter& w) {
        resp.serialize(w);
        stat.serialize(w);
      });
    else
      send_reply(c, xid, zxid_counter.load(std::memory_order_relaxed), err, nullptr);
  }

  // multi (op 14): apply a create/delete/setData/check transaction
  // atomically. All involved shards are locked in index order (a large multi
  // effectively serializes — transactions are a correctness feature, not the
  // throughput path), ops are validated against an overlay of the
  // transaction's own effects, then applied; watches fire only on success.
  struct MultiOpParsed {
    int32_t type = 0;
    std::string path;
    std::string data;
    int32_t flags = 0;
    int32_t version = -1;
  };

  void handle_multi(Conn* c, int32_t xid, JuteReader& r) {
    std::vector<MultiOpParsed> mops;
    while (true) {
      MultiHeader mh;
      mh.deserialize(r);
      if (mh.done) break;
      MultiOpParsed mo;
      mo.type = mh.type;
      switch (mh.type) {
        case kOpCreate: {
          CreateRequest req;
          req.deserialize(r);
          mo.path = std::move(req.path);
          mo.data = std::move(req.data);
          mo.flags = req.flags;
          break;
        }
        case kOpDelete: {
          DeleteRequest req;
          req.deserialize(r);
          mo.path = std::move(req.path);
          mo.version = req.version;
          break;
        }
        case kOpSetData: {
          SetDataRequest req;
          req.deserialize(r);
          mo.path = std::move(req.path);
          mo.data = std::move(req.data);
          mo.version = req.version;
          break;
        }
        case 13: {  // check(path, version)
          mo.path = r.read_string();
          mo.version = r.read_int();
          break;
        }
        default:
          send_reply(c, xid, zxid_counter.load(std::memory_order_relaxed), kZMarshallingError, nullptr);
          return;
      }
      mops.push_back(std::move(mo));
    }

    int64_t sid = sid_of(c);
    std::vector<int32_t> results(mops.size(), kZOk);
    std::vector<std::string> created_paths(mops.size());
    // setData Stat snapshots taken while the shard locks are held — the
    // reply body must NOT re-read the tree after the locks are released
    // (concurrent create/delete on the shard would race the read)
    std::vector<Stat> setdata_stats(mops.size());
    int64_t txn_zxid = 0;  // zxid of the LAST op applied by this txn
    std::vector<std::pair<std::string, int32_t>> data_events;   // path, event type
    std::vector<std::string> child_events;                      // parent paths
    std::vector<std::string> deleted_paths;                     // NodeDeleted → own child watchers
    // ephemeral bookkeeping: (path, added, owner-session) — a delete may
    // remove ANOTHER session's ephemeral; its owner's set must be updated
    std::vector<std::tuple<std::string, bool, int64_t>> eph_changes;
    int32_t txn_err = kZOk;

    {
      // lock every involved shard (paths + parents) in index order
      std::set<size_t> idxs;
      for (const auto& mo : mops) {
        idxs.insert(shard_idx(mo.path));
        idxs.insert(shard_idx(parent_path(mo.path)));
      }
      std::vector<std::unique_lock<std::mutex>> locks;
      locks.reserve(idxs.size());
      for (size_t i : idxs) locks.emplace_back(shards[i].mu);

      // ---- validate against tree + overlay of txn effects ----
      struct Overlay {
        // path -> (exists, ephemeral); absent = defer to tree
        std::unordered_map<std::string, std::pair<bool, bool>> state;
        std::unordered_map<std::string, int> child_delta;
        std::unordered_map<std::string, int32_t> version;  // post-op versions
      } ov;
      auto tree_node = [&](const std::string& p) -> ZNode* {
        auto& sh = shard_of(p);
        auto it = sh.nodes.find(p);
        return it == sh.nodes.end() ? nullptr : &it->second;
      };
      auto ov_exists = [&](const std::string& p) -> bool {
        auto it = ov.state.find(p);
        if (it != ov.state.end()) return it->second.first;
        return tree_node(p) != nullptr;
      };
      auto ov_ephemeral = [&](const std::string& p) -> bool {
        auto it = ov.state.find(p);
        if (it != ov.state.end()) return it->second.second;
        ZNode* n = tree_node(p);
        return n && n->stat.ephemeral_owner != 0;
      };
      auto ov_child_count = [&](const std::string& p) -> int {
        ZNode* n = tree_node(p);
        int base = n ? static_cast<int>(n->children.size()) : 0;
        auto it = ov.child_delta.find(p);
        return base + (it == ov.child_delta.end() ? 0 : it->second);
      };
      auto ov_version = [&](const std::string& p) -> int32_t {
        auto it = ov.version.find(p);
        if (it != ov.version.end()) return it->second;
        bool created_in_txn = ov.state.count(p) && ov.state[p].first && !tree_node(p);
        if (created_in_txn) return 0;
        ZNode* n = tree_node(p);
        return n ? n->stat.version : 0;
      };

      for (size_t i = 0; i < mops.size(); i++) {
        const auto& mo = mops[i];
        int32_t err = kZOk;
        if (!valid_path(mo.path) || mo.path == "/") {
          err = (mo.type == kOpCreate && mo.path == "/") ? kZNodeExists : kZMarshallingError;
        } else {
          std::string parent = parent_path(mo.path);
          switch (mo.type) {
            case kOpCreate:
              if (mo.flags & kSequence) {
                err = kZMarshallingError;  // sequence-in-multi unsupported
              } else if (!ov_exists(parent)) {
                err = kZNoNode;
              } else if (ov_ephemeral(parent)) {
                err = kZNoChildrenForEphemerals;
              } else if (ov_exists(mo.path)) {
                err = kZNodeExists;
              } else {
                ov.state[mo.path] = {true, (mo.flags & kEphemeral) != 0};
                ov.version[mo.path] = 0;
                ov.child_delta[parent]++;
              }
              break;
            case kOpDelete: {
              if (!ov_exists(mo.path)) {
                err = kZNoNode;
              } else if (ov_child_count(mo.path) > 0) {
                err = kZNotEmpty;
              } else {
                if (mo.version != -1 && mo.version != ov_version(mo.path)) {
                  err = kZBadVersion;
                } else {
                  ov.state[mo.path] = {false, false};
                  ov.version.erase(mo.path);
                  ov.child_delta[parent]--;
                }
              }
              break;
            }
            case kOpSetData: {
              if (!ov_exists(mo.path)) {
                err = kZNoNode;
              } else if (mo.version != -1 && mo.version != ov_version(mo.path)) {
                err = kZBadVersion;
              } else {
                ov.version[mo.path] = ov_version(mo.path) + 1;  // txn-internal increment
              }
              break;
            }
            case 13: {  // check
              if (!ov_exists(mo.path))
                err = kZNoNode;
              else if (mo.version != -1 && mo.version != ov_version(mo.path))
                err = kZBadVersion;
              break;
            }
          }
        }
        if (err != kZOk && txn_err == kZOk) {
          txn_err = err;
          results[i] = err;
        } else if (txn_err != kZOk) {
          results[i] = kZRuntimeInconsistency;
        }
      }
      if (txn_err != kZOk) {
        // everything before the failure reports RuntimeInconsistency too
        for (size_t i = 0; i < mops.size(); i++)
          if (results[i] == kZOk) results[i] = kZRuntimeInconsistency;
      } else {
        // ---- apply (all locks still held) ----
        for (size_t i = 0; i < mops.size(); i++) {
          const auto& mo = mops[i];
          std::string parent = parent_path(mo.path);
          switch (mo.type) {
            case kOpCreate: {
              ops.create.fetch_add(1, std::memory_order_relaxed);
              int64_t zz = zxid_counter.fetch_add(1, std::memory_order_relaxed) + 1;
              txn_zxid = zz;
              Shard& csh = shard_of(mo.path);
              ZNode& n = csh.nodes[mo.path];
              n.data = mo.data;
              n.stat = Stat{};
              n.stat.czxid = zz;
              n.stat.mzxid = zz;
              n.stat.ctime = wall_ms();
              n.stat.mtime = n.stat.ctime;
              n.stat.data_length = static_cast<int32_t>(mo.data.size());
              if (mo.flags & kEphemeral) {
                n.stat.ephemeral_owner = sid;
                eph_changes.push_back({mo.path, true, sid});
              }
              Shard& psh = shard_of(parent);
              auto pit = psh.nodes.find(parent);
              if (pit != psh.nodes.end()) {
                pit->second.children.insert(basename_of(mo.path));
                pit->second.stat.cversion++;
                pit->second.stat.pzxid = zz;
                pit->second.stat.num_children = static_cast<int32_t>(pit->second.children.size());
              }
              data_events.push_back({mo.path, kEventNodeCreated});
              child_events.push_back(parent);
              created_paths[i] = mo.path;
              break;
            }
            case kOpDelete: {
              ops.del.fetch_add(1, std::memory_order_relaxed);
              int64_t zz = zxid_counter.fetch_add(1, std::memory_order_relaxed) + 1;
              txn_zxid = zz;
              Shard& csh = shard_of(mo.path);
              auto it = csh.nodes.find(mo.path);
              if (it != csh.nodes.end()) {
                if (it->second.stat.ephemeral_owner != 0)
                  eph_changes.push_back({mo.path, false, it->second.stat.ephemeral_owner});
                csh.nodes.erase(it);
              }
              Shard& psh = shard_of(parent);
              auto pit = psh.nodes.find(parent);
              if (pit != psh.nodes.end()) {
                pit->second.children.erase(basename_of(mo.path));
                pit->second.stat.cversion++;
                pit->second.stat.pzxid = zz;
                pit->second.stat.num_children = static_cast<int32_t>(pit->second.children.size());
              }
              data_events.push_back({mo.path, kEventNodeDeleted});
              child_events.push_back(parent);
              deleted_paths.push_back(mo.path);
              break;
            }
            case kOpSetData: {
              ops.set_data.fetch_add(1, std::memory_order_relaxed);
              int64_t zz = zxid_counter.fetch_add(1, std::memory_order_relaxed) + 1;
              txn_zxid = zz;
              Shard& csh = shard_of(mo.path);
              auto it = csh.nodes.find(mo.path);
              if (it != csh.nodes.end()) {
                it->second.data = mo.data;
                it->second.stat.mzxid = zz;
                it->second.stat.mtime = wall_ms();
                it->second.stat.version++;
                it->second.stat.data_length = static_cast<int32_t>(mo.data.size());
                setdata_stats[i] = it->second.stat;
              }
              data_events.push_back({mo.path, kEventNodeDataChanged});
              break;
            }
            default:
              break;  // check: no effect
          }
        }
        // fire watches while holding the shard locks (same discipline as
        // the single-op handlers: shard → session_mu → conns_mu → out_mu)
        for (auto& [path, ev] : data_events)
          if (ev != kEventNodeDeleted) fire_data_watches_locked(shard_of(path), path, ev);
        for (auto& path : deleted_paths) fire_deleted_watches_locked(shard_of(path), path);
        std::set<std::string> fired;
        for (auto& parent : child_events)
          if (fired.insert(parent).second) fire_child_watches_locked(shard_of(parent), parent);
      }
    }

    // session ephemeral bookkeeping outside the shard locks (same alive-flag
    // handshake as single-op create); deletes may belong to OTHER sessions
    if (txn_err == kZOk && !eph_changes.empty()) {
      SessionPtr mine = c->session;
      auto owner_session = [&](int64_t owner) -> SessionPtr {
        if (mine && mine->id == owner) return mine;
        std::lock_guard<std::mutex> g(session_mu);
        auto sit = sessions.find(owner);
        return sit == sessions.end() ? nullptr : sit->second;
      };
      for (auto& [path, added, owner] : eph_changes) {
        SessionPtr s = owner_session(owner);
        if (!s) continue;  // owner already dead; its drain saw (or will
                           // no-op on) the node
        std::lock_guard<std::mutex> eg(s->eph_mu);
        if (added)
          s->ephemerals.insert(path);
        else
          s->ephemerals.erase(path);
      }
      if (mine && !mine->alive.load(std::memory_order_acquire)) {
        // our session died mid-txn: roll our created ephemerals back.
        // Collect under eph_mu, delete after releasing it — delete_node's
        // owner bookkeeping re-locks the same eph_mu (self-deadlock
        // otherwise; caught by the chaos stress watchdog).
        std::vector<std::string> rollback;
        {
          std::lock_guard<std::mutex> eg(mine->eph_mu);
          for (auto& [path, added, owner] : eph_changes)
            if (added && owner == mine->id && mine->ephemerals.erase(path)) rollback.push_back(path);
        }
        for (const auto& path : rollback) delete_node(path, mine);
      }
    }

    // reply with THIS txn's last zxid (not a fresh counter read, which may
    // already reflect other connections' later commits)
    int64_t z = txn_zxid ? txn_zxid : zxid_counter.load(std::memory_order_relaxed);
    send_reply_with_body(c, xid, z, txn_err, [&](JuteWriter& w) {
      for (size_t i = 0; i < mops.size(); i++) {
        MultiHeader mh;
        mh.done = false;
        if (txn_err != kZOk) {
          mh.type = -1;
          mh.err = results[i];
          mh.serialize(w);
          w.write_int(results[i]);  // ErrorResult
        } else {
          mh.type = mops[i].type;
          mh.err = 0;
          mh.serialize(w);
          if (mops[i].type == kOpCreate) w.write_string(created_paths[i]);
          if (mops[i].type == kOpSetData) setdata_stats[i].serialize(w);
        }
      }
      MultiHeader end;
      end.type = -1;
      end.done = true;
      end.err = -1;
      end.serialize(w);
    });
  }

  // setWatches: re-arm a reconnected session's watches, firing synthetic
  // events for anything that changed past relative_zxid while it was away
  void handle_set_watches(Conn* c, int32_t xid, JuteReader& r) {
    SetWatchesRequest req;
    req.deserialize(r);
    int64_t sid = sid_of(c);
    for (const auto& path : req.data_watches) {
      bool fire_changed = false, fire_deleted = false;
      {
        Shard& sh = shard_of(path);
        std::lock_guard<std::mutex> lk(sh.mu);
        auto it = sh.nodes.find(path);
        if (it == sh.nodes.end())
          fire_deleted = true;
        else if (it->second.stat.mzxid > req.relative_zxid)
          fire_changed = true;
        else
          sh.data_watches[path].insert(sid);
      }
      if (fire_deleted) deliver_watch({sid}, path, kEventNodeDeleted);
      if (fire_changed) deliver_watch({sid}, path, kEventNodeDataChanged);
    }
    for (const auto& path : req.exist_watches) {
      bool fire_created = false;
      {
        Shard& sh = shard_of(path);
        std::lock_guard<std::mutex> lk(sh.mu);
        auto it = sh.nodes.find(path);
        if (it != sh.nodes.end())
          fire_created = true;  // appeared while the client was away
        else
          sh.data_watches[path].insert(sid);
      }
      if (fire_created) deliver_watch({sid}, path, kEventNodeCreated);
    }
    for (const auto& path : req.child_watches) {
      bool fire_child = false, fire_deleted = false;
      {
        Shard& sh = shard_of(path);
        std::lock_guard<std::mutex> lk(sh.mu);
        auto it = sh.nodes.find(path);
        if (it == sh.nodes.end())
          fire_deleted = true;
        else if (it->second.stat.pzxid > req.relative_zxid)
          fire_child = true;
        else
          sh.child_watches[path].insert(sid);
      }
      if (fire_deleted) deliver_watch({sid}, path, kEventNodeDeleted);
      if (fire_child) deliver_watch({sid}, path, kEventNodeChildrenChanged);
    }
    send_reply(c, xid, zxid_counter.load(std::memory_order_relaxed), kZOk, nullptr);
  }

  // sync: single-copy tree ⇒ always in sync; echo the path back
  void handle_sync(Conn* c, int32_t xid, JuteReader& r) {
    std::string path = r.read_string();
    send_reply(c, xid, zxid_counter.load(std::memory_order_relaxed), kZOk,
               [&](JuteWriter& w) { w.write_string(path); });
  }

  // ACLs: everything is world:anyone (matching the reference's deployments);
  // getACL reports the open ACL, setACL acks without enforcement
  void handle_get_acl(Conn* c, int32_t xid, JuteReader& r) {
    std::string path = r.read_string();
    int32_t err = kZOk;
    Stat stat;
    {
      Shard& sh = shard_of(path);
      std::lock_guard<std::mutex> lk(sh.mu);
      auto it = sh.nodes.find(path);
      if (it == sh.nodes.end())
        err = kZNoNode;
      else
        stat = it->second.stat;
    }
    if (err == kZOk)
      send_reply(c, xid, zxid_counter.load(std::memory_order_relaxed), kZOk, [&](JuteWriter& w) {
        write_acl_vector(w, {ACL{}});
        stat.serialize(w);
      });
    else
      send_reply(c, xid, zxid_counter.load(std::memory_order_relaxed), err, nullptr);
  }

  void handle_set_acl(Conn* c, int32_t xid, JuteReader& r) {
    std::string path = r.read_string();
    (void)read_acl_vector(r);
    r.read_int();  // version
    int32_t err = kZOk;
    Stat stat;
    {
      Shard& sh = shard_of(path);
      std::lock_guard<std::mutex> lk(sh.mu);
      auto it = sh.nodes.find(path);
      if (it == sh.nodes.end()) {
        err = kZNoNode;
      } else {
        it->second.stat.aversion++;
        stat = it->second.stat;
      }
    }
    if (err == kZOk)
      send_reply(c, xid, zxid_counter.load(std::memory_order_relaxed), kZOk,
                 [&](JuteWriter& w) { stat.serialize(w); });
    else
      send_reply(c, xid, zxid_counter.load(std::memory_order_relaxed), err, nullptr);
  }

  void handle_close_session(Conn* c, int32_t xid) {
    ops.close_session.fetch_add(1, std::memory_order_relaxed);
    int64_t sid = sid_of(c);
    {
      std::lock_guard<std::mutex> og(c->out_mu);
      c->closing = true;  // close after the reply drains
    }
    send_reply(c, xid, zxid_counter.load(std::memory_order_relaxed), kZOk, nullptr);
    if (sid != 0) kill_session(sid, /*close_conn_too=*/false);
  }

  // ---------------- session lifecycle ----------------

  // Expire/close a session: remove ephemerals (firing watches), tombstone it,
  // optionally retire its connection. Lock discipline: session_mu alone, then
  // eph_mu alone, then per-path shard locks via delete_node.
  void kill_session(int64_t sid, bool close_conn_too) {
    SessionPtr s;
    {
      std::lock_guard<std::mutex> g(session_mu);
      auto it = sessions.find(sid);
      if (it == sessions.end()) return;
      s = it->second;
      sessions.erase(it);
    }
    s->alive.store(false, std::memory_order_release);
    std::unordered_set<std::string> eph;
    {
      std::lock_guard<std::mutex> eg(s->eph_mu);
      eph.swap(s->ephemerals);
    }
    for (const auto& p : eph) delete_node(p, s);
    // stale watch registrations for this session are skipped at fire time
    uint64_t cid = s->conn_id.load();
    if (close_conn_too && cid != 0) {
      ConnPtr cp = lookup(cid);
      if (cp) {
        cp->dead.store(true);
        post_close(cp);
      }
    }
  }

  void schedule_sweep() {
    loops[0]->schedule(cfg.tick_ms, [this] {
      std::vector<int64_t> expired;
      {
        std::lock_guard<std::mutex> g(session_mu);
        int64_t now = now_ms();
        for (const auto& kv : sessions)
          if (now - kv.second->last_touch.load(std::memory_order_relaxed) > kv.second->timeout_ms)
            expired.push_back(kv.first);
      }
      for (int64_t sid : expired) {
        log.info("session expired", {{"session", Json(sid)}});
        kill_session(sid, /*close_conn_too=*/true);
      }
      schedule_sweep();
    });
  }

  // ---------------- watches (shard lock held) ----------------

  void fire_data_watches_locked(Shard& sh, const std::string& path, int32_t event_type) {
    auto it = sh.data_watches.find(path);
    if (it == sh.data_watches.end()) return;
    std::set<int64_t> watchers = std::move(it->second);
    sh.data_watches.erase(it);
    deliver_watch(watchers, path, event_type);
  }

  void fire_child_watches_locked(Shard& sh, const std::string& path) {
    auto it = sh.child_watches.find(path);
    if (it == sh.child_watches.end()) return;
    std::set<int64_t> watchers = std::move(it->second);
    sh.child_watches.erase(it);
    deliver_watch(watchers, path, kEventNodeChildrenChanged);
  }

  // NodeDeleted goes to data watchers AND child watchers of the deleted
  // node, deduplicated per session — ZooKeeper's DataTree passes the
  // already-triggered set to the child-watch trigger so a session holding
  // both watch kinds gets ONE event
  void fire_deleted_watches_locked(Shard& sh, const std::string& path) {
    std::set<int64_t> watchers;
    if (auto it = sh.data_watches.find(path); it != sh.data_watches.end()) {
      watchers = std::move(it->second);
      sh.data_watches.erase(it);
    }
    if (auto it = sh.child_watches.find(path); it != sh.child_watches.end()) {
      watchers.insert(it->second.begin(), it->second.end());
      sh.child_watches.erase(it);
    }
    if (!watchers.empty()) deliver_watch(watchers, path, kEventNodeDeleted);
  }

  void deliver_watch(const std::set<int64_t>& watchers, const std::string& path, int32_t event_type) {
    for (int64_t sid : watchers) {
      uint64_t cid = 0;
      {
        std::lock_guard<std::mutex> g(session_mu);
        auto sit = sessions.find(sid);
        if (sit == sessions.end()) continue;  // stale registration
        cid = sit->second->conn_id.load();
      }
      if (cid == 0) continue;
      ConnPtr cp = lookup(cid);
      if (!cp) continue;
      std::string pkt;
      begin_packet(&pkt);
      JuteWriter w(&pkt);
      ReplyHeader hdr;
      hdr.xid = kXidWatcherEvent;
      hdr.zxid = -1;
      hdr.err = 0;
      hdr.serialize(w);
      WatcherEvent ev;
      ev.type = event_type;
      ev.state = kStateSyncConnected;
      ev.path = path;
      ev.serialize(w);
      frame_packet(&pkt);
      send_packet(cp.get(), std::move(pkt));
    }
  }

  // ---------------- response sending ----------------

  template <typename BodyFn>
  void send_reply(Conn* c, int32_t xid, int64_t zxid, int32_t err, BodyFn body) {
    // fast path: serialize straight into the connection's output buffer (one
    // allocation-free append per reply, flushed once per event drain)
    if (latency_ms.load(std::memory_order_relaxed) == 0 && loops[c->io_idx]->on_loop_thread()) {
      {
        std::lock_guard<std::mutex> og(c->out_mu);
        if (c->dead.load(std::memory_order_relaxed)) return;
        size_t start = c->outbuf.size();
        c->outbuf.append(4, '\0');
        JuteWriter w(&c->outbuf);
        ReplyHeader hdr;
        hdr.xid = xid;
        hdr.zxid = zxid;
        hdr.err = err;
        hdr.serialize(w);
        if constexpr (!std::is_same_v<BodyFn, std::nullptr_t>) {
          if (err == kZOk) body(w);
        }
        uint32_t n = static_cast<uint32_t>(c->outbuf.size() - start - 4);
        c->outbuf[start] = static_cast<char>(n >> 24);
        c->outbuf[start + 1] = static_cast<char>(n >> 16);
        c->outbuf[start + 2] = static_cast<char>(n >> 8);
        c->outbuf[start + 3] = static_cast<char>(n);
      }
      ensure_flush_scheduled(c);
      return;
    }
    std::string pkt;
    begin_packet(&pkt);
    JuteWriter w(&pkt);
    ReplyHeader hdr;
    hdr.xid = xid;
    hdr.zxid = zxid;
    hdr.err = err;
    hdr.serialize(w);
    if constexpr (!std::is_same_v<BodyFn, std::nullptr_t>) {
      if (err == kZOk) body(w);
    }
    frame_packet(&pkt);
    send_packet(c, std::move(pkt));
  }

  void send_reply(Conn* c, int32_t xid, int64_t zxid, int32_t err, std::nullptr_t) {
    send_reply<std::nullptr_t>(c, xid, zxid, err, nullptr);
  }

  // multi replies carry a body even on failure (per-op error results)
  template <typename BodyFn>
  void send_reply_with_body(Conn* c, int32_t xid, int64_t zxid, int32_t err, BodyFn body) {
    std::string pkt;
    begin_packet(&pkt);
    JuteWriter w(&pkt);
    ReplyHeader hdr;
    hdr.xid = xid;
    hdr.zxid = zxid;
    hdr.err = err;
    hdr.serialize(w);
    body(w);
    frame_packet(&pkt);
    send_packet(c, std::move(pkt));
  }

  // Any thread. Latency injection routes through the owner loop's timers
  // (FIFO for equal deadlines), preserving per-conn response order.
  void send_packet(Conn* c, std::string pkt) {
    int lat = latency_ms.load();
    if (lat > 0) {
      uint64_t cid = c->id;
      loops[c->io_idx]->schedule_from_any(lat, [this, cid, pkt = std::move(pkt)]() mutable {
        ConnPtr cp = lookup(cid);
        if (cp) deliver(cp.get(), std::move(pkt));
      });
      return;
    }
    if (loops[c->io_idx]->on_loop_thread()) {
      // reply path: queue only; flushed once per event drain (on_conn_event)
      // with a same-iteration 0 ms timer as the safety net for sends outside
      // the event path (sweep-timer watch deliveries)
      {
        std::lock_guard<std::mutex> og(c->out_mu);
        if (c->dead.load()) return;
        c->outbuf += pkt;
      }
      ensure_flush_scheduled(c);
      return;
    }
    deliver(c, std::move(pkt));
  }

  // owner loop thread only
  void ensure_flush_scheduled(Conn* c) {
    if (c->flush_scheduled) return;
    c->flush_scheduled = true;
    uint64_t cid = c->id;
    loops[c->io_idx]->schedule(0, [this, cid] {
      ConnPtr cp = lookup(cid);
      if (!cp) return;
      cp->flush_scheduled = false;
      if (!cp->dead.load()) flush_out(cp.get());
    });
  }

  // Cross-thread delivery: direct write when the buffer is clear, else
  // buffer + arm EPOLLOUT on the owner loop.
  void deliver(Conn* c, std::string pkt) {
    std::lock_guard<std::mutex> og(c->out_mu);
    if (c->dead.load()) return;
    if (c->outbuf.empty()) {
      size_t off = 0;
      while (off < pkt.size()) {
        ssize_t n = send(c->fd, pkt.data() + off, pkt.size() - off, MSG_NOSIGNAL);
        if (n > 0) {
          off += static_cast<size_t>(n);
        } else if (n < 0 && (errno == EAGAIN || errno == EWOULDBLOCK)) {
          break;
        } else if (n < 0 && errno == EINTR) {
          continue;
        } else {
          mark_dead(c);
          return;
        }
      }
      if (off < pkt.size()) {
        c->outbuf = pkt.substr(off);
        arm_epollout(c);
      } else if (c->closing) {
        mark_dead(c);
      }
      return;
    }
    c->outbuf += pkt;
  }

  // out_mu held
  void arm_epollout(Conn* c) {
    if (!c->epollout_armed) {
      c->epollout_armed = true;
      loops[c->io_idx]->mod_fd(c->fd, EPOLLIN | EPOLLOUT);
    }
  }

  // Marks the conn for closure; the owner loop performs the actual close.
  void mark_dead(Conn* c) {
    if (c->dead.exchange(true)) return;
    uint64_t cid = c->id;
    loops[c->io_idx]->post([this, cid] {
      ConnPtr cp = lookup(cid);
      if (cp) close_conn(cp.get());
    });
  }

  void flush_out(Conn* c) {
    bool close_now = false;
    {
      std::lock_guard<std::mutex> og(c->out_mu);
      size_t off = 0;
      while (off < c->outbuf.size()) {
        ssize_t n = send(c->fd, c->outbuf.data() + off, c->outbuf.size() - off, MSG_NOSIGNAL);
        if (n > 0) {
          off += static_cast<size_t>(n);
        } else if (n < 0 && (errno == EAGAIN || errno == EWOULDBLOCK)) {
          break;
        } else if (n < 0 && errno == EINTR) {
          continue;
        } else {
          close_now = true;
          break;
        }
      }
      if (!close_now) {
        c->outbuf.erase(0, off);
        if (c->outbuf.empty()) {
          if (c->epollout_armed) {
            c->epollout_armed = false;
            loops[c->io_idx]->mod_fd(c->fd, EPOLLIN);
          }
          if (c->closing) close_now = true;
        } else {
          arm_epollout(c);
        }
      }
    }
    if (close_now) close_conn(c);
  }

  // MUST run on the owner loop thread.
  void close_conn(Conn* c) {
    c->dead.store(true);
    {
      std::lock_guard<std::mutex> g(conns_mu);
      auto it = conns.find(c->id);
      if (it == conns.end()) return;  // already closed
      conns.erase(it);
    }
    if (c->session) {
      uint64_t expected = c->id;
      c->session->conn_id.compare_exchange_strong(expected, 0);
    }
    {
      // out_mu serializes against a cross-thread deliver() mid-write: the fd
      // must not be closed (and possibly reused) under a concurrent write
      std::lock_guard<std::mutex> og(c->out_mu);
      loops[c->io_idx]->del_fd(c->fd);
      ::close(c->fd);
      c->fd = -1;
    }
  }

  // queue a close on the conn's owner loop
  void post_close(const ConnPtr& c) {
    uint64_t cid = c->id;
    size_t srv = c->io_idx;
    loops[srv]->post([this, cid] {
      ConnPtr cp = lookup(cid);
      if (cp) close_conn(cp.get());
    });
  }

  // ---------------- control (any thread) ----------------

  void kill_server(size_t idx) {
    {
      std::lock_guard<std::mutex> g(admin_mu);
      if (idx >= servers.size()) return;
    }
    size_t lidx = idx % loops.size();
    std::promise<void> done;
    loops[lidx]->post([this, idx, lidx, &done] {
      {
        std::lock_guard<std::mutex> g(admin_mu);
        if (idx < servers.size() && servers[idx].up) {
          Server& s = servers[idx];
          loops[lidx]->del_fd(s.listen_fd);
          ::close(s.listen_fd);
          s.listen_fd = -1;
          s.up = false;
        }
      }
      done.set_value();
    });
    done.get_future().wait();
    // retire this server's connections on their own IO loops
    std::vector<ConnPtr> victims;
    {
      std::lock_guard<std::mutex> g(conns_mu);
      for (auto& kv : conns)
        if (kv.second->server_idx == idx) victims.push_back(kv.second);
    }
    std::vector<std::promise<void>> vdone(victims.size());
    for (size_t i = 0; i < victims.size(); i++) {
      ConnPtr c = victims[i];
      c->dead.store(true);
      loops[c->io_idx]->post([this, c, &vdone, i] {
        close_conn(c.get());
        vdone[i].set_value();
      });
    }
    for (auto& d : vdone) d.get_future().wait();
    log.info("server killed", {{"server", Json(static_cast<int64_t>(idx))}});
  }

  void restart_server(size_t idx) {
    {
      std::lock_guard<std::mutex> g(admin_mu);
      if (idx >= servers.size()) return;
    }
    std::promise<void> done;
    loops[idx % loops.size()]->post([this, idx, &done] {
      {
        std::lock_guard<std::mutex> g(admin_mu);
        if (idx < servers.size() && !servers[idx].up) {
          open_listener(idx, servers[idx].port);
          log.info("server restarted", {{"server", Json(static_cast<int64_t>(idx))}});
        }
      }
      done.set_value();
    });
    done.get_future().wait();
  }

  // admin_mu held
  void maybe_elect_leader_locked() {
    if (leader_idx < servers.size() && servers[leader_idx].up) return;
    for (size_t i = 0; i < servers.size(); i++) {
      if (servers[i].up) {
        leader_idx = i;
        log.info("leader elected", {{"leader", Json(static_cast<int64_t>(i))}});
        return;
      }
    }
  }

  size_t kill_leader() {
    size_t victim;
    {
      std::lock_guard<std::mutex> g(admin_mu);
      victim = leader_idx;
    }
    kill_server(victim);
    std::lock_guard<std::mutex> g(admin_mu);
    if (cfg.election_ms > 0) election_until = now_ms() + cfg.election_ms;
    maybe_elect_leader_locked();
    return victim;
  }
};

// ---------------- public API ----------------

Ensemble::Ensemble(EnsembleConfig cfg) : impl_(std::make_unique<Impl>(std::move(cfg))) {}

Ensemble::~Ensemble() {
  try {
    stop();
  } catch (...) {
  }
}

void Ensemble::start() { impl_->start(); }
void Ensemble::stop() { impl_->stop(); }

std::vector<int> Ensemble::ports() const {
  std::lock_guard<std::mutex> g(impl_->admin_mu);
  std::vector<int> out;
  for (const auto& s : impl_->servers) out.push_back(s.port);
  return out;
}

std::string Ensemble::connect_string() const {
  std::string out;
  for (int p : ports()) {
    if (!out.empty()) out += ',';
    out += impl_->cfg.bind_host + ":" + std::to_string(p);
  }
  return out;
}

void Ensemble::kill_server(size_t idx) { impl_->kill_server(idx); }
void Ensemble::restart_server(size_t idx) { impl_->restart_server(idx); }

bool Ensemble::server_up(size_t idx) const {
  std::lock_guard<std::mutex> g(impl_->admin_mu);
  return idx < impl_->servers.size() && impl_->servers[idx].up;
}

size_t Ensemble::leader() const {
  std::lock_guard<std::mutex> g(impl_->admin_mu);
  return impl_->leader_idx;
}

size_t Ensemble::kill_leader() { return impl_->kill_leader(); }

void Ensemble::expire_session(int64_t session_id) { impl_->kill_session(session_id, /*close_conn_too=*/true); }

void Ensemble::set_latency_ms(int ms) { impl_->latency_ms.store(ms); }

NodeInfo Ensemble::get(const std::string& path) const {
  auto& sh = impl_->shard_of(path);
  std::lock_guard<std::mutex> g(sh.mu);
  NodeInfo info;
  auto it = sh.nodes.find(path);
  if (it != sh.nodes.end()) {
    info.exists = true;
    info.data = it->second.data;
    info.stat = it->second.stat;
  }
  return info;
}

std::vector<std::string> Ensemble::children(const std::string& path) const {
  std::vector<std::string> out;
  {
    auto& sh = impl_->shard_of(path);
    std::lock_guard<std::mutex> g(sh.mu);
    auto it = sh.nodes.find(path);
    if (it == sh.nodes.end()) return out;
    out.assign(it->second.children.begin(), it->second.children.end());
  }
  std::sort(out.begin(), out.end());
  return out;
}

size_t Ensemble::node_count() const {
  size_t n = 0;
  for (auto& sh : impl_->shards) {
    std::lock_guard<std::mutex> g(sh.mu);
    n += sh.nodes.size();
  }
  return n - 1;  // exclude root
}

size_t Ensemble::ephemeral_count() const {
  std::vector<Impl::SessionPtr> snap;
  {
    std::lock_guard<std::mutex> g(impl_->session_mu);
    for (const auto& kv : impl_->sessions) snap.push_back(kv.second);
  }
  size_t n = 0;
  for (const auto& s : snap) {
    std::lock_guard<std::mutex> eg(s->eph_mu);
    n += s->ephemerals.size();
  }
  return n;
}

std::vector<int64_t> Ensemble::session_ids() const {
  std::lock_guard<std::mutex> g(impl_->session_mu);
  std::vector<int64_t> out;
  for (const auto& kv : impl_->sessions) out.push_back(kv.first);
  return out;
}

int64_t Ensemble::zxid() const { return impl_->zxid_counter.load(); }

std::map<std::string, uint64_t> Ensemble::counters() const {
  std::map<std::string, uint64_t> out;
  const auto& o = impl_->ops;
  out["connect"] = o.connect.load();
  out["create"] = o.create.load();
  out["delete"] = o.del.load();
  out["exists"] = o.exists.load();
  out["getData"] = o.get_data.load();
  out["setData"] = o.set_data.load();
  out["getChildren"] = o.get_children.load();
  out["ping"] = o.ping.load();
  out["closeSession"] = o.close_session.load();
  out["unknown"] = o.unknown.load();
  for (auto it = out.begin(); it != out.end();) {
    if (it->second == 0)
      it = out.erase(it);
    else
      ++it;
  }
  return out;
}

}  // namespace zk
}  // namespace registrar
