// ensemble.cpp — synthetic in-process ZooKeeper ensemble (see ensemble.hpp).
//
// Threading model (the scaling core): one epoll loop THREAD PER SERVER, so N
// registrar processes spread across the ensemble's servers get parallel
// socket I/O and frame processing. The shared data tree / session table /
// watch maps live under one state mutex (`mu`) whose critical sections are
// memory-only (no syscalls); per-connection output buffers have their own
// mutex so reply writes and cross-thread watch deliveries never serialize on
// the global lock. Lock order: mu → conn->out_mu (never the reverse).
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
  struct ZNode {
    std::string data;
    Stat stat;
    std::set<std::string> children;
  };

  struct Session {
    int64_t id = 0;
    std::string passwd;
    int timeout_ms = 30000;
    int64_t last_touch = 0;  // monotonic ms
    uint64_t conn_id = 0;    // 0 = detached
  };

  struct Conn {
    uint64_t id = 0;
    int fd = -1;
    size_t server_idx = 0;
    // owner-loop-thread only:
    std::string inbuf;
    size_t inpos = 0;
    bool handshaken = false;
    int64_t session_id = 0;
    // shared output state:
    std::mutex out_mu;
    std::string outbuf;
    bool epollout_armed = false;
    bool closing = false;  // close once outbuf drains
    std::atomic<bool> dead{false};
    bool flush_scheduled = false;  // owner loop thread only
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
  std::vector<std::unique_ptr<EventLoop>> loops;  // one per server slot
  std::vector<std::thread> threads;
  std::atomic<bool> started{false};
  std::atomic<int> latency_ms{0};
  OpCounters ops;

  // ---- state under mu ----
  mutable std::mutex mu;
  std::vector<Server> servers;
  std::unordered_map<uint64_t, ConnPtr> conns;
  uint64_t next_conn_id = 1;
  std::unordered_map<std::string, ZNode> nodes;
  std::unordered_map<int64_t, Session> sessions;
  std::unordered_set<int64_t> dead_sessions;  // expired or closed: reconnect ⇒ expired
  std::unordered_map<int64_t, std::set<std::string>> ephemerals;
  std::unordered_map<std::string, std::set<int64_t>> data_watches;
  std::unordered_map<std::string, std::set<int64_t>> child_watches;
  int64_t zxid_counter = 0;
  int64_t next_session = 0x100000;
  size_t leader_idx = 0;
  int64_t election_until = 0;  // monotonic ms; connects refused until then

  explicit Impl(EnsembleConfig c) : cfg(std::move(c)), log(Logger("zk-ensemble").child("ensemble")) {
    log.set_level(cfg.log_level);
    latency_ms.store(cfg.latency_ms);
    nodes["/"] = ZNode{};
  }

  // ---------------- lifecycle ----------------

  void start() {
    if (started.exchange(true)) return;
    size_t n = cfg.ports.size();
    loops.clear();
    for (size_t i = 0; i < n; i++) loops.push_back(std::make_unique<EventLoop>());
    {
      std::lock_guard<std::mutex> g(mu);
      servers.resize(n);
      for (size_t i = 0; i < n; i++) open_listener(i, cfg.ports[i]);
    }
    for (size_t i = 0; i < n; i++) {
      threads.emplace_back([this, i] {
        if (i == 0) schedule_sweep();
        loops[i]->run();
      });
    }
  }

  void stop() {
    if (!started.load()) return;
    // close everything from each owner loop, then stop the loops
    std::vector<std::promise<void>> done(loops.size());
    for (size_t i = 0; i < loops.size(); i++) {
      loops[i]->post([this, i, &done] {
        std::vector<ConnPtr> victims;
        {
          std::lock_guard<std::mutex> g(mu);
          for (auto& kv : conns)
            if (kv.second->server_idx == i) victims.push_back(kv.second);
        }
        for (auto& c : victims) close_conn(c.get());
        {
          std::lock_guard<std::mutex> g(mu);
          if (i < servers.size() && servers[i].listen_fd >= 0) {
            loops[i]->del_fd(servers[i].listen_fd);
            ::close(servers[i].listen_fd);
            servers[i].listen_fd = -1;
            servers[i].up = false;
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

  // mu held; listener registration goes to loops[idx]
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
    auto install = [this, fd, srv] {
      loops[srv]->add_fd(fd, EPOLLIN, [this, fd, srv](uint32_t) { on_accept(fd, srv); });
    };
    if (loops[srv]->on_loop_thread()) {
      install();
    } else if (started.load() && loops[srv]->running()) {
      loops[srv]->post(install);
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
        std::lock_guard<std::mutex> g(mu);
        if (now_ms() < election_until) {
          // mid-election: nobody serves (BASELINE config 4 storm realism)
          ::close(fd);
          continue;
        }
        maybe_elect_leader_locked();
        set_nodelay(fd);
        auto conn = std::make_shared<Conn>();
        conn->id = next_conn_id++;
        conn->fd = fd;
        conn->server_idx = server_idx;
        conns[conn->id] = conn;
        uint64_t cid = conn->id;
        loops[server_idx]->add_fd(fd, EPOLLIN, [this, cid](uint32_t ev) { on_conn_event(cid, ev); });
      }
    }
  }

  ConnPtr lookup(uint64_t cid) {
    std::lock_guard<std::mutex> g(mu);
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

  // returns false if the conn was closed (dead conns are closed by caller)
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
      switch (hdr.type) {
        case kOpPing: {
          ops.ping++;
          int64_t z;
          {
            std::lock_guard<std::mutex> g(mu);
            touch_session_locked(c->session_id);
            z = zxid_counter;
          }
          send_reply(c, kXidPing, z, kZOk, nullptr);
          break;
        }
        case kOpCreate:
          handle_create(c, hdr.xid, r);
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
        case kOpCloseSession:
          handle_close_session(c, hdr.xid);
          break;
        default: {
          ops.unknown++;
          int64_t z;
          {
            std::lock_guard<std::mutex> g(mu);
            touch_session_locked(c->session_id);
            z = zxid_counter;
          }
          send_reply(c, hdr.xid, z, kZSystemError, nullptr);
          break;
        }
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
    ops.connect++;
    ConnectResponse resp;
    resp.has_read_only = req.has_read_only;
    bool expired_handshake = false;
    {
      std::lock_guard<std::mutex> g(mu);
      if (req.session_id != 0) {
        auto sit = sessions.find(req.session_id);
        if (sit == sessions.end() || sit->second.passwd != req.passwd) {
          // unknown/expired/bad-passwd session ⇒ the canonical "expired"
          // ConnectResponse: sessionId=0, timeOut=0
          resp.session_id = 0;
          resp.time_out_ms = 0;
          expired_handshake = true;
          log.info("connect: session expired/unknown", {{"session", Json(req.session_id)}});
        } else {
          Session& s = sit->second;
          if (s.conn_id != 0) {
            auto old = conns.find(s.conn_id);
            if (old != conns.end()) {
              // session moved: retire the old connection without detaching
              old->second->dead.store(true);
              post_close_locked(old->second);
            }
          }
          s.conn_id = c->id;
          s.last_touch = now_ms();
          c->session_id = s.id;
          resp.session_id = s.id;
          resp.time_out_ms = s.timeout_ms;
          resp.passwd = s.passwd;
          log.info("connect: session re-attached", {{"session", Json(s.id)}});
        }
      } else {
        Session s;
        s.id = next_session++;
        s.passwd.resize(16);
        uint64_t seed = static_cast<uint64_t>(s.id) * 0x9E3779B97F4A7C15ull + 0xD1B54A32D192ED03ull;
        for (int i = 0; i < 16; i++) {
          seed ^= seed >> 27;
          seed *= 0x94D049BB133111EBull;
          s.passwd[i] = static_cast<char>(seed >> (8 * (i % 8)));
        }
        int req_to = req.time_out_ms > 0 ? req.time_out_ms : 30000;
        s.timeout_ms = std::max(cfg.min_session_timeout_ms, std::min(cfg.max_session_timeout_ms, req_to));
        s.last_touch = now_ms();
        s.conn_id = c->id;
        c->session_id = s.id;
        resp.session_id = s.id;
        resp.time_out_ms = s.timeout_ms;
        resp.passwd = s.passwd;
        sessions[s.id] = s;
        log.info("connect: new session",
                 {{"session", Json(s.id)}, {"timeout_ms", Json(static_cast<int64_t>(s.timeout_ms))}});
      }
    }
    c->handshaken = true;
    std::string pkt;
    begin_packet(&pkt);
    JuteWriter w(&pkt);
    resp.serialize(w);
    frame_packet(&pkt);
    if (expired_handshake) {
      // expired handshake: server notifies then closes
      {
        std::lock_guard<std::mutex> og(c->out_mu);
        c->closing = true;
      }
      send_packet(c, std::move(pkt));
    } else {
      send_packet(c, std::move(pkt));
    }
  }

  // --- ops ---

  void handle_create(Conn* c, int32_t xid, JuteReader& r) {
    ops.create++;
    CreateRequest req;
    req.deserialize(r);
    int32_t err = kZOk;
    int64_t z;
    CreateResponse resp;
    {
      std::lock_guard<std::mutex> g(mu);
      touch_session_locked(c->session_id);
      if (!valid_path(req.path) || req.path == "/") {
        err = kZMarshallingError;
      } else {
        std::string parent = parent_path(req.path);
        auto pit = nodes.find(parent);
        if (pit == nodes.end()) {
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
          if (nodes.count(path)) {
            err = kZNodeExists;
          } else {
            int64_t zz = ++zxid_counter;
            ZNode n;
            n.data = req.data;
            n.stat.czxid = zz;
            n.stat.mzxid = zz;
            n.stat.ctime = wall_ms();
            n.stat.mtime = n.stat.ctime;
            n.stat.data_length = static_cast<int32_t>(req.data.size());
            if (req.flags & kEphemeral) {
              n.stat.ephemeral_owner = c->session_id;
              ephemerals[c->session_id].insert(path);
            }
            nodes[path] = std::move(n);
            ZNode& par = nodes[parent];
            par.children.insert(basename_of(path));
            par.stat.cversion++;
            par.stat.pzxid = zz;
            par.stat.num_children = static_cast<int32_t>(par.children.size());
            fire_data_watches_locked(path, kEventNodeCreated);
            fire_child_watches_locked(parent);
            resp.path = path;
          }
        }
      }
      z = zxid_counter;
    }
    if (err == kZOk)
      send_reply(c, xid, z, kZOk, [&](JuteWriter& w) { resp.serialize(w); });
    else
      send_reply(c, xid, z, err, nullptr);
  }

  void handle_delete(Conn* c, int32_t xid, JuteReader& r) {
    ops.del++;
    DeleteRequest req;
    req.deserialize(r);
    int32_t err = kZOk;
    int64_t z;
    {
      std::lock_guard<std::mutex> g(mu);
      touch_session_locked(c->session_id);
      auto it = nodes.find(req.path);
      if (it == nodes.end() || req.path == "/") {
        err = kZNoNode;
      } else if (!it->second.children.empty()) {
        err = kZNotEmpty;
      } else if (req.version != -1 && req.version != it->second.stat.version) {
        err = kZBadVersion;
      } else {
        delete_node_locked(req.path);
      }
      z = zxid_counter;
    }
    send_reply(c, xid, z, err, nullptr);
  }

  // mu held; fires watches, updates parent
  void delete_node_locked(const std::string& path) {
    auto it = nodes.find(path);
    if (it == nodes.end()) return;
    int64_t z = ++zxid_counter;
    if (it->second.stat.ephemeral_owner != 0) {
      auto eit = ephemerals.find(it->second.stat.ephemeral_owner);
      if (eit != ephemerals.end()) eit->second.erase(path);
    }
    nodes.erase(it);
    std::string parent = parent_path(path);
    auto pit = nodes.find(parent);
    if (pit != nodes.end()) {
      pit->second.children.erase(basename_of(path));
      pit->second.stat.cversion++;
      pit->second.stat.pzxid = z;
      pit->second.stat.num_children = static_cast<int32_t>(pit->second.children.size());
      fire_child_watches_locked(parent);
    }
    fire_data_watches_locked(path, kEventNodeDeleted);
  }

  void handle_exists(Conn* c, int32_t xid, JuteReader& r) {
    ops.exists++;
    ExistsRequest req;
    req.deserialize(r);
    int32_t err = kZOk;
    int64_t z;
    ExistsResponse resp;
    {
      std::lock_guard<std::mutex> g(mu);
      touch_session_locked(c->session_id);
      if (req.watch) data_watches[req.path].insert(c->session_id);
      auto it = nodes.find(req.path);
      if (it == nodes.end())
        err = kZNoNode;
      else
        resp.stat = it->second.stat;
      z = zxid_counter;
    }
    if (err == kZOk)
      send_reply(c, xid, z, kZOk, [&](JuteWriter& w) { resp.serialize(w); });
    else
      send_reply(c, xid, z, err, nullptr);
  }

  void handle_get_data(Conn* c, int32_t xid, JuteReader& r) {
    ops.get_data++;
    GetDataRequest req;
    req.deserialize(r);
    int32_t err = kZOk;
    int64_t z;
    GetDataResponse resp;
    {
      std::lock_guard<std::mutex> g(mu);
      touch_session_locked(c->session_id);
      auto it = nodes.find(req.path);
      if (it == nodes.end()) {
        err = kZNoNode;
      } else {
        if (req.watch) data_watches[req.path].insert(c->session_id);
        resp.data = it->second.data;
        resp.stat = it->second.stat;
      }
      z = zxid_counter;
    }
    if (err == kZOk)
      send_reply(c, xid, z, kZOk, [&](JuteWriter& w) { resp.serialize(w); });
    else
      send_reply(c, xid, z, err, nullptr);
  }

  void handle_set_data(Conn* c, int32_t xid, JuteReader& r) {
    ops.set_data++;
    SetDataRequest req;
    req.deserialize(r);
    int32_t err = kZOk;
    int64_t z;
    SetDataResponse resp;
    {
      std::lock_guard<std::mutex> g(mu);
      touch_session_locked(c->session_id);
      auto it = nodes.find(req.path);
      if (it == nodes.end()) {
        err = kZNoNode;
      } else if (req.version != -1 && req.version != it->second.stat.version) {
        err = kZBadVersion;
      } else {
        int64_t zz = ++zxid_counter;
        it->second.data = req.data;
        it->second.stat.mzxid = zz;
        it->second.stat.mtime = wall_ms();
        it->second.stat.version++;
        it->second.stat.data_length = static_cast<int32_t>(req.data.size());
        fire_data_watches_locked(req.path, kEventNodeDataChanged);
        resp.stat = it->second.stat;
      }
      z = zxid_counter;
    }
    if (err == kZOk)
      send_reply(c, xid, z, kZOk, [&](JuteWriter& w) { resp.serialize(w); });
    else
      send_reply(c, xid, z, err, nullptr);
  }

  void handle_get_children(Conn* c, int32_t xid, JuteReader& r) {
    ops.get_children++;
    GetChildrenRequest req;
    req.deserialize(r);
    int32_t err = kZOk;
    int64_t z;
    GetChildrenResponse resp;
    {
      std::lock_guard<std::mutex> g(mu);
      touch_session_locked(c->session_id);
      auto it = nodes.find(req.path);
      if (it == nodes.end()) {
        err = kZNoNode;
      } else {
        if (req.watch) child_watches[req.path].insert(c->session_id);
        resp.children.assign(it->second.children.begin(), it->second.children.end());
      }
      z = zxid_counter;
    }
    if (err == kZOk)
      send_reply(c, xid, z, kZOk, [&](JuteWriter& w) { resp.serialize(w); });
    else
      send_reply(c, xid, z, err, nullptr);
  }

  void handle_close_session(Conn* c, int32_t xid) {
    ops.close_session++;
    int64_t sid = c->session_id;
    int64_t z;
    {
      std::lock_guard<std::mutex> g(mu);
      z = zxid_counter;
    }
    {
      std::lock_guard<std::mutex> og(c->out_mu);
      c->closing = true;  // close after the reply drains
    }
    send_reply(c, xid, z, kZOk, nullptr);
    if (sid != 0) {
      std::lock_guard<std::mutex> g(mu);
      kill_session_locked(sid, /*close_conn=*/false);
    }
  }

  // ---------------- session lifecycle ----------------

  // mu held
  void touch_session_locked(int64_t sid) {
    auto it = sessions.find(sid);
    if (it != sessions.end()) it->second.last_touch = now_ms();
  }

  // mu held. Expire/close a session: remove ephemerals (firing watches),
  // tombstone it, optionally retire its connection.
  void kill_session_locked(int64_t sid, bool close_conn_too) {
    auto it = sessions.find(sid);
    if (it == sessions.end()) return;
    uint64_t cid = it->second.conn_id;
    auto eit = ephemerals.find(sid);
    if (eit != ephemerals.end()) {
      std::vector<std::string> paths(eit->second.begin(), eit->second.end());
      for (const auto& p : paths) delete_node_locked(p);
      ephemerals.erase(sid);
    }
    for (auto& kv : data_watches) kv.second.erase(sid);
    for (auto& kv : child_watches) kv.second.erase(sid);
    sessions.erase(sid);
    dead_sessions.insert(sid);
    if (close_conn_too && cid != 0) {
      auto cit = conns.find(cid);
      if (cit != conns.end()) {
        cit->second->dead.store(true);
        post_close_locked(cit->second);
      }
    }
  }

  void schedule_sweep() {
    loops[0]->schedule(cfg.tick_ms, [this] {
      {
        std::lock_guard<std::mutex> g(mu);
        int64_t now = now_ms();
        std::vector<int64_t> expired;
        for (const auto& kv : sessions)
          if (now - kv.second.last_touch > kv.second.timeout_ms) expired.push_back(kv.first);
        for (int64_t sid : expired) {
          log.info("session expired", {{"session", Json(sid)}});
          kill_session_locked(sid, /*close_conn=*/true);
        }
      }
      schedule_sweep();
    });
  }

  // ---------------- watches ----------------

  // mu held for all three
  void fire_data_watches_locked(const std::string& path, int32_t event_type) {
    auto it = data_watches.find(path);
    if (it == data_watches.end()) return;
    std::set<int64_t> watchers = std::move(it->second);
    data_watches.erase(it);
    deliver_watch_locked(watchers, path, event_type);
  }

  void fire_child_watches_locked(const std::string& path) {
    auto it = child_watches.find(path);
    if (it == child_watches.end()) return;
    std::set<int64_t> watchers = std::move(it->second);
    child_watches.erase(it);
    deliver_watch_locked(watchers, path, kEventNodeChildrenChanged);
  }

  void deliver_watch_locked(const std::set<int64_t>& watchers, const std::string& path, int32_t event_type) {
    for (int64_t sid : watchers) {
      auto sit = sessions.find(sid);
      if (sit == sessions.end() || sit->second.conn_id == 0) continue;
      auto cit = conns.find(sit->second.conn_id);
      if (cit == conns.end()) continue;
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
      send_packet(cit->second.get(), std::move(pkt));
    }
  }

  // ---------------- response sending ----------------

  template <typename BodyFn>
  void send_reply(Conn* c, int32_t xid, int64_t zxid, int32_t err, BodyFn body) {
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

  // Any thread. Latency injection routes through the owner loop's timers
  // (FIFO for equal deadlines), preserving per-conn response order.
  void send_packet(Conn* c, std::string pkt) {
    int lat = latency_ms.load();
    if (lat > 0) {
      uint64_t cid = c->id;
      loops[c->server_idx]->schedule_from_any(lat, [this, cid, pkt = std::move(pkt)]() mutable {
        ConnPtr cp = lookup(cid);
        if (cp) deliver(cp.get(), std::move(pkt));
      });
      return;
    }
    if (loops[c->server_idx]->on_loop_thread()) {
      // reply path: queue only; flushed once per event drain (on_conn_event)
      // with a same-iteration 0 ms timer as the safety net for sends outside
      // the event path (sweep-timer watch deliveries)
      {
        std::lock_guard<std::mutex> og(c->out_mu);
        if (c->dead.load()) return;
        c->outbuf += pkt;
      }
      if (!c->flush_scheduled) {
        c->flush_scheduled = true;
        uint64_t cid = c->id;
        loops[c->server_idx]->schedule(0, [this, cid] {
          ConnPtr cp = lookup(cid);
          if (!cp) return;
          cp->flush_scheduled = false;
          if (!cp->dead.load()) flush_out(cp.get());
        });
      }
      return;
    }
    deliver(c, std::move(pkt));
  }

  // Any thread. Direct write when the buffer is clear, else buffer + arm
  // EPOLLOUT on the owner loop.
  void deliver(Conn* c, std::string pkt) {
    std::lock_guard<std::mutex> og(c->out_mu);
    if (c->dead.load()) return;
    if (c->outbuf.empty()) {
      size_t off = 0;
      while (off < pkt.size()) {
        ssize_t n = write(c->fd, pkt.data() + off, pkt.size() - off);
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
      loops[c->server_idx]->mod_fd(c->fd, EPOLLIN | EPOLLOUT);
    }
  }

  // out_mu held (or unambiguous owner). Marks the conn for closure; the owner
  // loop performs the actual close.
  void mark_dead(Conn* c) {
    if (c->dead.exchange(true)) return;
    uint64_t cid = c->id;
    loops[c->server_idx]->post([this, cid] {
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
        ssize_t n = write(c->fd, c->outbuf.data() + off, c->outbuf.size() - off);
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
          c->epollout_armed = false;
          loops[c->server_idx]->mod_fd(c->fd, EPOLLIN);
          if (c->closing) close_now = true;
        }
      }
    }
    if (close_now) close_conn(c);
  }

  // MUST run on the owner loop thread.
  void close_conn(Conn* c) {
    c->dead.store(true);
    std::lock_guard<std::mutex> g(mu);
    auto it = conns.find(c->id);
    if (it == conns.end()) return;  // already closed
    if (c->session_id != 0) {
      auto sit = sessions.find(c->session_id);
      if (sit != sessions.end() && sit->second.conn_id == c->id) sit->second.conn_id = 0;
    }
    loops[c->server_idx]->del_fd(c->fd);
    ::close(c->fd);
    conns.erase(it);
  }

  // mu held: queue a close on the conn's owner loop
  void post_close_locked(const ConnPtr& c) {
    uint64_t cid = c->id;
    size_t srv = c->server_idx;
    loops[srv]->post([this, cid] {
      ConnPtr cp = lookup(cid);
      if (cp) close_conn(cp.get());
    });
  }

  // ---------------- control (any thread) ----------------

  void kill_server(size_t idx) {
    if (idx >= loops.size()) return;
    std::promise<void> done;
    loops[idx]->post([this, idx, &done] {
      std::vector<ConnPtr> victims;
      {
        std::lock_guard<std::mutex> g(mu);
        if (idx < servers.size() && servers[idx].up) {
          Server& s = servers[idx];
          loops[idx]->del_fd(s.listen_fd);
          ::close(s.listen_fd);
          s.listen_fd = -1;
          s.up = false;
          for (auto& kv : conns)
            if (kv.second->server_idx == idx) victims.push_back(kv.second);
        }
      }
      for (auto& c : victims) close_conn(c.get());
      log.info("server killed", {{"server", Json(static_cast<int64_t>(idx))}});
      done.set_value();
    });
    done.get_future().wait();
  }

  void restart_server(size_t idx) {
    if (idx >= loops.size()) return;
    std::promise<void> done;
    loops[idx]->post([this, idx, &done] {
      {
        std::lock_guard<std::mutex> g(mu);
        if (idx < servers.size() && !servers[idx].up) {
          open_listener(idx, servers[idx].port);
          log.info("server restarted", {{"server", Json(static_cast<int64_t>(idx))}});
        }
      }
      done.set_value();
    });
    done.get_future().wait();
  }

  // mu held
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
      std::lock_guard<std::mutex> g(mu);
      victim = leader_idx;
    }
    kill_server(victim);
    std::lock_guard<std::mutex> g(mu);
    if (cfg.election_ms > 0) election_until = now_ms() + cfg.election_ms;
    maybe_elect_leader_locked();
    return victim;
  }

  void expire_session(int64_t session_id) {
    std::lock_guard<std::mutex> g(mu);
    kill_session_locked(session_id, /*close_conn=*/true);
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
  std::lock_guard<std::mutex> g(impl_->mu);
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
  std::lock_guard<std::mutex> g(impl_->mu);
  return idx < impl_->servers.size() && impl_->servers[idx].up;
}

size_t Ensemble::leader() const {
  std::lock_guard<std::mutex> g(impl_->mu);
  return impl_->leader_idx;
}

size_t Ensemble::kill_leader() { return impl_->kill_leader(); }

void Ensemble::expire_session(int64_t session_id) { impl_->expire_session(session_id); }

void Ensemble::set_latency_ms(int ms) { impl_->latency_ms.store(ms); }

NodeInfo Ensemble::get(const std::string& path) const {
  std::lock_guard<std::mutex> g(impl_->mu);
  NodeInfo info;
  auto it = impl_->nodes.find(path);
  if (it != impl_->nodes.end()) {
    info.exists = true;
    info.data = it->second.data;
    info.stat = it->second.stat;
  }
  return info;
}

std::vector<std::string> Ensemble::children(const std::string& path) const {
  std::lock_guard<std::mutex> g(impl_->mu);
  auto it = impl_->nodes.find(path);
  if (it == impl_->nodes.end()) return {};
  return std::vector<std::string>(it->second.children.begin(), it->second.children.end());
}

size_t Ensemble::node_count() const {
  std::lock_guard<std::mutex> g(impl_->mu);
  return impl_->nodes.size() - 1;  // exclude root
}

size_t Ensemble::ephemeral_count() const {
  std::lock_guard<std::mutex> g(impl_->mu);
  size_t n = 0;
  for (const auto& kv : impl_->ephemerals) n += kv.second.size();
  return n;
}

std::vector<int64_t> Ensemble::session_ids() const {
  std::lock_guard<std::mutex> g(impl_->mu);
  std::vector<int64_t> out;
  for (const auto& kv : impl_->sessions) out.push_back(kv.first);
  return out;
}

int64_t Ensemble::zxid() const {
  std::lock_guard<std::mutex> g(impl_->mu);
  return impl_->zxid_counter;
}

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
  // drop zero counters so tests can assert presence meaningfully
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
