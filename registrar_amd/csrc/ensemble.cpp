// ensemble.cpp — synthetic in-process ZooKeeper ensemble (see ensemble.hpp).
#include "ensemble.hpp"

#include <arpa/inet.h>
#include <fcntl.h>
#include <netinet/in.h>
#include <netinet/tcp.h>
#include <sys/socket.h>
#include <unistd.h>

#include <algorithm>
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
    std::string inbuf;
    size_t inpos = 0;
    std::string outbuf;
    bool handshaken = false;
    bool closing = false;  // close once outbuf drains
    int64_t session_id = 0;
  };

  struct Server {
    int listen_fd = -1;
    int port = 0;
    bool up = false;
  };

  EnsembleConfig cfg;
  Logger log;
  EventLoop loop;
  std::thread thread;
  std::atomic<bool> started{false};

  // All mutable ensemble state below is guarded by mu (request handlers run
  // on the loop thread but introspection comes from arbitrary threads).
  mutable std::mutex mu;
  std::vector<Server> servers;
  std::unordered_map<uint64_t, std::unique_ptr<Conn>> conns;
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
  std::atomic<int> latency_ms{0};
  std::map<std::string, uint64_t> op_counters;

  explicit Impl(EnsembleConfig c) : cfg(std::move(c)), log(Logger("zk-ensemble").child("ensemble")) {
    log.set_level(cfg.log_level);
    latency_ms.store(cfg.latency_ms);
    nodes["/"] = ZNode{};
  }

  // ---------------- lifecycle ----------------

  void start() {
    if (started.exchange(true)) return;
    {
      std::lock_guard<std::mutex> g(mu);
      servers.resize(cfg.ports.size());
      for (size_t i = 0; i < cfg.ports.size(); i++) open_listener(i, cfg.ports[i]);
    }
    thread = std::thread([this] {
      schedule_sweep();
      loop.run();
    });
  }

  void stop() {
    if (!started.load()) return;
    std::promise<void> done;
    loop.post([this, &done] {
      std::lock_guard<std::mutex> g(mu);
      for (auto& kv : conns) {
        loop.del_fd(kv.second->fd);
        close(kv.second->fd);
      }
      conns.clear();
      for (auto& s : servers) {
        if (s.listen_fd >= 0) {
          loop.del_fd(s.listen_fd);
          close(s.listen_fd);
          s.listen_fd = -1;
          s.up = false;
        }
      }
      done.set_value();
    });
    done.get_future().wait();
    loop.stop();
    if (thread.joinable()) thread.join();
    started.store(false);
  }

  // mu held
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
      close(fd);
      throw std::runtime_error("ensemble: bind failed on port " + std::to_string(port));
    }
    if (listen(fd, 512) < 0) {
      close(fd);
      throw std::runtime_error("ensemble: listen failed");
    }
    socklen_t alen = sizeof(addr);
    getsockname(fd, reinterpret_cast<struct sockaddr*>(&addr), &alen);
    set_nonblock(fd);
    servers[idx].listen_fd = fd;
    servers[idx].port = ntohs(addr.sin_port);
    servers[idx].up = true;
    size_t srv = idx;
    // add_fd must run on the loop thread once it is running; during start()
    // the loop thread hasn't started yet, so direct add is safe. After
    // restart_server we always go through loop.post.
    auto install = [this, fd, srv] { loop.add_fd(fd, EPOLLIN, [this, fd, srv](uint32_t) { on_accept(fd, srv); }); };
    if (started.load() && !loop.on_loop_thread()) {
      loop.post(install);
    } else {
      install();
    }
  }

  // ---------------- socket handling (loop thread) ----------------

  void on_accept(int listen_fd, size_t server_idx) {
    while (true) {
      int fd = accept4(listen_fd, nullptr, nullptr, SOCK_NONBLOCK | SOCK_CLOEXEC);
      if (fd < 0) break;
      std::lock_guard<std::mutex> g(mu);
      if (now_ms() < election_until) {
        // mid-election: nobody serves (BASELINE config 4 storm realism)
        close(fd);
        continue;
      }
      maybe_elect_leader_locked();
      set_nodelay(fd);
      auto conn = std::make_unique<Conn>();
      conn->id = next_conn_id++;
      conn->fd = fd;
      conn->server_idx = server_idx;
      uint64_t cid = conn->id;
      conns[cid] = std::move(conn);
      loop.add_fd(fd, EPOLLIN, [this, cid](uint32_t ev) { on_conn_event(cid, ev); });
    }
  }

  void on_conn_event(uint64_t cid, uint32_t ev) {
    std::unique_lock<std::mutex> g(mu);
    auto it = conns.find(cid);
    if (it == conns.end()) return;
    Conn* c = it->second.get();
    if (ev & (EPOLLHUP | EPOLLERR)) {
      close_conn_locked(c);
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
      uint64_t cid = c->id;
      if (!process_frames_locked(c)) return;  // conn closed
      if (eof) {
        auto it2 = conns.find(cid);
        if (it2 != conns.end()) close_conn_locked(it2->second.get());
        return;
      }
    }
    if (ev & EPOLLOUT) flush_out_locked(c);
  }

  // returns false if the conn was closed. Handlers may close this conn (write
  // error, closeSession, expired handshake) or any other conn (watch delivery,
  // session steal), so the conn is re-resolved by id after every frame.
  bool process_frames_locked(Conn* c) {
    uint64_t cid = c->id;
    while (true) {
      size_t avail = c->inbuf.size() - c->inpos;
      if (avail < 4) break;
      const unsigned char* p = reinterpret_cast<const unsigned char*>(c->inbuf.data() + c->inpos);
      uint32_t len = (static_cast<uint32_t>(p[0]) << 24) | (static_cast<uint32_t>(p[1]) << 16) |
                     (static_cast<uint32_t>(p[2]) << 8) | static_cast<uint32_t>(p[3]);
      if (len > 4 * 1024 * 1024) {  // jute.maxbuffer-ish sanity cap
        close_conn_locked(c);
        return false;
      }
      if (avail < 4 + len) break;
      // frame body lives in c->inbuf; handlers must not touch inbuf (they
      // don't — they only parse `body` and enqueue output)
      const char* body = c->inbuf.data() + c->inpos + 4;
      c->inpos += 4 + len;
      handle_frame_locked(c, body, len);
      auto it = conns.find(cid);
      if (it == conns.end()) return false;
      c = it->second.get();
    }
    if (c->inpos > 0) {
      c->inbuf.erase(0, c->inpos);
      c->inpos = 0;
    }
    return true;
  }

  void handle_frame_locked(Conn* c, const char* body, size_t len) {
    try {
      JuteReader r(body, len);
      if (!c->handshaken) {
        handle_connect_locked(c, r);
        return;
      }
      RequestHeader hdr;
      hdr.deserialize(r);
      touch_session_locked(c->session_id);
      switch (hdr.type) {
        case kOpPing:
          count_op("ping");
          send_reply_locked(c, kXidPing, kZOk, nullptr);
          break;
        case kOpCreate:
          handle_create_locked(c, hdr.xid, r);
          break;
        case kOpDelete:
          handle_delete_locked(c, hdr.xid, r);
          break;
        case kOpExists:
          handle_exists_locked(c, hdr.xid, r);
          break;
        case kOpGetData:
          handle_get_data_locked(c, hdr.xid, r);
          break;
        case kOpSetData:
          handle_set_data_locked(c, hdr.xid, r);
          break;
        case kOpGetChildren:
          handle_get_children_locked(c, hdr.xid, r);
          break;
        case kOpCloseSession:
          handle_close_session_locked(c, hdr.xid);
          return;
        default:
          count_op("unknown");
          send_reply_locked(c, hdr.xid, kZSystemError, nullptr);
          break;
      }
    } catch (const std::exception& e) {
      log.warn("ensemble: malformed frame, closing conn", {{"err", Json(e.what())}});
      if (conns.count(c->id)) close_conn_locked(c);
    }
  }

  void handle_connect_locked(Conn* c, JuteReader& r) {
    ConnectRequest req;
    req.deserialize(r);
    count_op("connect");
    ConnectResponse resp;
    resp.has_read_only = req.has_read_only;
    if (req.session_id != 0) {
      auto sit = sessions.find(req.session_id);
      if (sit == sessions.end() || sit->second.passwd != req.passwd) {
        // unknown/expired/bad-passwd session ⇒ the canonical "expired"
        // ConnectResponse: sessionId=0, timeOut=0
        resp.session_id = 0;
        resp.time_out_ms = 0;
        log.info("connect: session expired/unknown", {{"session", Json(req.session_id)}});
      } else {
        Session& s = sit->second;
        if (s.conn_id != 0) {
          auto old = conns.find(s.conn_id);
          if (old != conns.end()) close_conn_locked(old->second.get(), /*detach_session=*/false);
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
      log.info("connect: new session", {{"session", Json(s.id)}, {"timeout_ms", Json(static_cast<int64_t>(s.timeout_ms))}});
    }
    c->handshaken = true;
    std::string pkt;
    begin_packet(&pkt);
    JuteWriter w(&pkt);
    resp.serialize(w);
    frame_packet(&pkt);
    send_raw_locked(c, std::move(pkt));
    if (resp.session_id == 0) {
      // expired handshake: server closes after notifying
      shutdown_after_flush_locked(c);
    }
  }

  // --- ops ---

  void handle_create_locked(Conn* c, int32_t xid, JuteReader& r) {
    count_op("create");
    CreateRequest req;
    req.deserialize(r);
    if (!valid_path(req.path) || req.path == "/") {
      send_reply_locked(c, xid, kZMarshallingError, nullptr);
      return;
    }
    std::string parent = parent_path(req.path);
    auto pit = nodes.find(parent);
    if (pit == nodes.end()) {
      send_reply_locked(c, xid, kZNoNode, nullptr);
      return;
    }
    if (pit->second.stat.ephemeral_owner != 0) {
      send_reply_locked(c, xid, kZNoChildrenForEphemerals, nullptr);
      return;
    }
    std::string path = req.path;
    if (req.flags & kSequence) {
      char suffix[16];
      snprintf(suffix, sizeof(suffix), "%010d", pit->second.stat.cversion);
      path += suffix;
    }
    if (nodes.count(path)) {
      send_reply_locked(c, xid, kZNodeExists, nullptr);
      return;
    }
    int64_t z = ++zxid_counter;
    ZNode n;
    n.data = req.data;
    n.stat.czxid = z;
    n.stat.mzxid = z;
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
    par.stat.pzxid = z;
    par.stat.num_children = static_cast<int32_t>(par.children.size());
    fire_data_watches_locked(path, kEventNodeCreated);
    fire_child_watches_locked(parent);
    CreateResponse resp;
    resp.path = path;
    send_reply_locked(c, xid, kZOk, [&](JuteWriter& w) { resp.serialize(w); });
  }

  void handle_delete_locked(Conn* c, int32_t xid, JuteReader& r) {
    count_op("delete");
    DeleteRequest req;
    req.deserialize(r);
    auto it = nodes.find(req.path);
    if (it == nodes.end() || req.path == "/") {
      send_reply_locked(c, xid, kZNoNode, nullptr);
      return;
    }
    if (!it->second.children.empty()) {
      send_reply_locked(c, xid, kZNotEmpty, nullptr);
      return;
    }
    if (req.version != -1 && req.version != it->second.stat.version) {
      send_reply_locked(c, xid, kZBadVersion, nullptr);
      return;
    }
    delete_node_locked(req.path);
    send_reply_locked(c, xid, kZOk, nullptr);
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

  void handle_exists_locked(Conn* c, int32_t xid, JuteReader& r) {
    count_op("exists");
    ExistsRequest req;
    req.deserialize(r);
    auto it = nodes.find(req.path);
    if (req.watch) data_watches[req.path].insert(c->session_id);
    if (it == nodes.end()) {
      send_reply_locked(c, xid, kZNoNode, nullptr);
      return;
    }
    ExistsResponse resp;
    resp.stat = it->second.stat;
    send_reply_locked(c, xid, kZOk, [&](JuteWriter& w) { resp.serialize(w); });
  }

  void handle_get_data_locked(Conn* c, int32_t xid, JuteReader& r) {
    count_op("getData");
    GetDataRequest req;
    req.deserialize(r);
    auto it = nodes.find(req.path);
    if (it == nodes.end()) {
      send_reply_locked(c, xid, kZNoNode, nullptr);
      return;
    }
    if (req.watch) data_watches[req.path].insert(c->session_id);
    GetDataResponse resp;
    resp.data = it->second.data;
    resp.stat = it->second.stat;
    send_reply_locked(c, xid, kZOk, [&](JuteWriter& w) { resp.serialize(w); });
  }

  void handle_set_data_locked(Conn* c, int32_t xid, JuteReader& r) {
    count_op("setData");
    SetDataRequest req;
    req.deserialize(r);
    auto it = nodes.find(req.path);
    if (it == nodes.end()) {
      send_reply_locked(c, xid, kZNoNode, nullptr);
      return;
    }
    if (req.version != -1 && req.version != it->second.stat.version) {
      send_reply_locked(c, xid, kZBadVersion, nullptr);
      return;
    }
    int64_t z = ++zxid_counter;
    it->second.data = req.data;
    it->second.stat.mzxid = z;
    it->second.stat.mtime = wall_ms();
    it->second.stat.version++;
    it->second.stat.data_length = static_cast<int32_t>(req.data.size());
    fire_data_watches_locked(req.path, kEventNodeDataChanged);
    SetDataResponse resp;
    resp.stat = it->second.stat;
    send_reply_locked(c, xid, kZOk, [&](JuteWriter& w) { resp.serialize(w); });
  }

  void handle_get_children_locked(Conn* c, int32_t xid, JuteReader& r) {
    count_op("getChildren");
    GetChildrenRequest req;
    req.deserialize(r);
    auto it = nodes.find(req.path);
    if (it == nodes.end()) {
      send_reply_locked(c, xid, kZNoNode, nullptr);
      return;
    }
    if (req.watch) child_watches[req.path].insert(c->session_id);
    GetChildrenResponse resp;
    resp.children.assign(it->second.children.begin(), it->second.children.end());
    send_reply_locked(c, xid, kZOk, [&](JuteWriter& w) { resp.serialize(w); });
  }

  void handle_close_session_locked(Conn* c, int32_t xid) {
    count_op("closeSession");
    send_reply_locked(c, xid, kZOk, nullptr);
    int64_t sid = c->session_id;
    shutdown_after_flush_locked(c);
    if (sid != 0) kill_session_locked(sid, /*notify_conn=*/false);
  }

  // ---------------- session lifecycle ----------------

  void touch_session_locked(int64_t sid) {
    auto it = sessions.find(sid);
    if (it != sessions.end()) it->second.last_touch = now_ms();
  }

  // Expire/close a session: remove ephemerals (firing watches), tombstone it.
  void kill_session_locked(int64_t sid, bool notify_conn) {
    auto it = sessions.find(sid);
    if (it == sessions.end()) return;
    uint64_t cid = it->second.conn_id;
    auto eit = ephemerals.find(sid);
    if (eit != ephemerals.end()) {
      std::vector<std::string> paths(eit->second.begin(), eit->second.end());
      for (const auto& p : paths) delete_node_locked(p);
      ephemerals.erase(sid);
    }
    // drop its watches
    for (auto& kv : data_watches) kv.second.erase(sid);
    for (auto& kv : child_watches) kv.second.erase(sid);
    sessions.erase(sid);
    dead_sessions.insert(sid);
    if (notify_conn && cid != 0) {
      auto cit = conns.find(cid);
      if (cit != conns.end()) close_conn_locked(cit->second.get(), /*detach_session=*/false);
    }
  }

  void schedule_sweep() {
    loop.schedule(cfg.tick_ms, [this] {
      {
        std::lock_guard<std::mutex> g(mu);
        int64_t now = now_ms();
        std::vector<int64_t> expired;
        for (const auto& kv : sessions)
          if (now - kv.second.last_touch > kv.second.timeout_ms) expired.push_back(kv.first);
        for (int64_t sid : expired) {
          log.info("session expired", {{"session", Json(sid)}});
          kill_session_locked(sid, /*notify_conn=*/true);
        }
      }
      schedule_sweep();
    });
  }

  // ---------------- watches ----------------

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
      send_raw_locked(cit->second.get(), std::move(pkt));
    }
  }

  // ---------------- response sending ----------------

  void count_op(const char* name) { op_counters[name]++; }

  template <typename BodyFn>
  void send_reply_locked(Conn* c, int32_t xid, int32_t err, BodyFn body) {
    std::string pkt;
    begin_packet(&pkt);
    JuteWriter w(&pkt);
    ReplyHeader hdr;
    hdr.xid = xid;
    hdr.zxid = zxid_counter;
    hdr.err = err;
    hdr.serialize(w);
    if constexpr (!std::is_same_v<BodyFn, std::nullptr_t>) {
      if (err == kZOk) body(w);
    }
    frame_packet(&pkt);
    send_raw_locked(c, std::move(pkt));
  }

  void send_reply_locked(Conn* c, int32_t xid, int32_t err, std::nullptr_t) {
    send_reply_locked<std::nullptr_t>(c, xid, err, nullptr);
  }

  void send_raw_locked(Conn* c, std::string pkt) {
    int lat = latency_ms.load();
    if (lat > 0) {
      uint64_t cid = c->id;
      loop.schedule(lat, [this, cid, pkt = std::move(pkt)]() mutable {
        std::lock_guard<std::mutex> g(mu);
        auto it = conns.find(cid);
        if (it == conns.end()) return;
        enqueue_locked(it->second.get(), std::move(pkt));
      });
      return;
    }
    enqueue_locked(c, std::move(pkt));
  }

  void enqueue_locked(Conn* c, std::string pkt) {
    if (c->outbuf.empty()) {
      // fast path: try a direct write before buffering
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
          close_conn_locked(c);
          return;
        }
      }
      if (off < pkt.size()) {
        c->outbuf = pkt.substr(off);
        loop.mod_fd(c->fd, EPOLLIN | EPOLLOUT);
      } else if (c->closing) {
        close_conn_locked(c);
      }
      return;
    }
    c->outbuf += pkt;
  }

  void flush_out_locked(Conn* c) {
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
        close_conn_locked(c);
        return;
      }
    }
    c->outbuf.erase(0, off);
    if (c->outbuf.empty()) {
      if (c->closing) {
        close_conn_locked(c);
        return;
      }
      loop.mod_fd(c->fd, EPOLLIN);
    }
  }

  void shutdown_after_flush_locked(Conn* c) {
    if (c->outbuf.empty()) {
      close_conn_locked(c);
    } else {
      c->closing = true;
    }
  }

  void close_conn_locked(Conn* c, bool detach_session = true) {
    if (detach_session && c->session_id != 0) {
      auto sit = sessions.find(c->session_id);
      if (sit != sessions.end() && sit->second.conn_id == c->id) sit->second.conn_id = 0;
    }
    loop.del_fd(c->fd);
    close(c->fd);
    conns.erase(c->id);
  }

  // ---------------- control (any thread) ----------------

  void run_on_loop(std::function<void()> fn) {
    if (loop.on_loop_thread()) {
      fn();
      return;
    }
    std::promise<void> done;
    loop.post([&] {
      fn();
      done.set_value();
    });
    done.get_future().wait();
  }

  void kill_server(size_t idx) {
    run_on_loop([this, idx] {
      std::lock_guard<std::mutex> g(mu);
      if (idx >= servers.size() || !servers[idx].up) return;
      Server& s = servers[idx];
      loop.del_fd(s.listen_fd);
      close(s.listen_fd);
      s.listen_fd = -1;
      s.up = false;
      std::vector<Conn*> victims;
      for (auto& kv : conns)
        if (kv.second->server_idx == idx) victims.push_back(kv.second.get());
      for (Conn* c : victims) close_conn_locked(c);
      log.info("server killed", {{"server", Json(static_cast<int64_t>(idx))}});
    });
  }

  void restart_server(size_t idx) {
    run_on_loop([this, idx] {
      std::lock_guard<std::mutex> g(mu);
      if (idx >= servers.size() || servers[idx].up) return;
      open_listener(idx, servers[idx].port);
      log.info("server restarted", {{"server", Json(static_cast<int64_t>(idx))}});
    });
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

void Ensemble::expire_session(int64_t session_id) {
  impl_->run_on_loop([this, session_id] {
    std::lock_guard<std::mutex> g(impl_->mu);
    impl_->kill_session_locked(session_id, /*notify_conn=*/true);
  });
}

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
  std::lock_guard<std::mutex> g(impl_->mu);
  return impl_->op_counters;
}

}  // namespace zk
}  // namespace registrar
