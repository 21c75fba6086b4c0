// jute.hpp — ZooKeeper jute wire-format codec + protocol records.
//
// From-scratch implementation of the subset of the ZooKeeper client/server
// wire protocol that registrar needs (the reference
// reaches this protocol through the zkplus → ZooKeeper C client chain; here
// it is implemented natively). All integers are big-endian; strings and
// buffers are length-prefixed with a 4-byte length (-1 encodes null). Every
// packet on the wire is itself length-prefixed with a 4-byte frame length.
//
// Records implemented: ConnectRequest/Response, RequestHeader/ReplyHeader,
// Stat, Create/Delete/Exists/GetData/SetData/GetChildren requests+responses,
// MultiHeader, SetWatchesRequest, WatcherEvent. Ops covered:
// create(±EPHEMERAL|SEQUENCE), delete, exists, getData, setData,
// getChildren(+2), sync, multi, setWatches, ping, closeSession — the exact
// client verb set the reference uses (stat/put/create/mkdirp/unlink/close,
// SURVEY §2.4) plus watches and transactions so the synthetic ensemble can
// serve Binder-style readers and the atomic-swap registration mode
// (docs/protocol.md is the wire reference).
#pragma once

#include <cstdint>
#include <cstring>
#include <stdexcept>
#include <string>
#include <vector>

namespace registrar {
namespace zk {

// --- opcodes (RequestHeader.type) ---
enum OpCode : int32_t {
  kOpNotification = 0,
  kOpCreate = 1,
  kOpDelete = 2,
  kOpExists = 3,
  kOpGetData = 4,
  kOpSetData = 5,
  kOpGetACL = 6,
  kOpSetACL = 7,
  kOpGetChildren = 8,
  kOpSync = 9,
  kOpPing = 11,
  kOpGetChildren2 = 12,
  kOpMulti = 14,
  kOpCreate2 = 15,
  kOpCloseSession = -11,
  kOpSetWatches = 101,
  kOpError = -1,
};

// --- well-known xids ---
enum Xid : int32_t {
  kXidWatcherEvent = -1,
  kXidPing = -2,
  kXidAuth = -4,
  kXidSetWatches = -8,
};

// --- server error codes (ReplyHeader.err) ---
enum ErrorCode : int32_t {
  kZOk = 0,
  kZSystemError = -1,
  kZRuntimeInconsistency = -2,
  kZConnectionLoss = -4,    // client-side
  kZMarshallingError = -5,
  kZOperationTimeout = -7,  // client-side
  kZNoNode = -101,
  kZNoAuth = -102,
  kZBadVersion = -103,
  kZNoChildrenForEphemerals = -108,
  kZNodeExists = -110,
  kZNotEmpty = -111,
  kZSessionExpired = -112,
  kZInvalidACL = -114,
  kZAuthFailed = -115,
  kZSessionMoved = -118,
};

// zkplus-compatible error names: cleanupPreviousEntries tolerates err.name ==
// 'NO_NODE' (reference: lib/register.js:88-93); we preserve those names in
// errors surfaced to callers and logs.
inline const char* error_name(int32_t code) {
  switch (code) {
    case kZOk:
      return "OK";
    case kZNoNode:
      return "NO_NODE";
    case kZNodeExists:
      return "NODE_EXISTS";
    case kZNotEmpty:
      return "NOT_EMPTY";
    case kZBadVersion:
      return "BAD_VERSION";
    case kZNoChildrenForEphemerals:
      return "NO_CHILDREN_FOR_EPHEMERALS";
    case kZSessionExpired:
      return "SESSION_EXPIRED";
    case kZConnectionLoss:
      return "CONNECTION_LOSS";
    case kZOperationTimeout:
      return "OPERATION_TIMEOUT";
    case kZInvalidACL:
      return "INVALID_ACL";
    case kZAuthFailed:
      return "AUTH_FAILED";
    case kZNoAuth:
      return "NO_AUTH";
    case kZSessionMoved:
      return "SESSION_MOVED";
    case kZMarshallingError:
      return "MARSHALLING_ERROR";
    default:
      return "SYSTEM_ERROR";
  }
}

// --- create flags ---
enum CreateFlags : int32_t {
  kEphemeral = 1,
  kSequence = 2,
};

// --- watcher event types / keeper states ---
enum EventType : int32_t {
  kEventNone = -1,
  kEventNodeCreated = 1,
  kEventNodeDeleted = 2,
  kEventNodeDataChanged = 3,
  kEventNodeChildrenChanged = 4,
};
enum KeeperState : int32_t {
  kStateDisconnected = 0,
  kStateSyncConnected = 3,
  kStateExpired = -112,
};

// ---------------------------------------------------------------------------
// codec

class JuteWriter {
 public:
  explicit JuteWriter(std::string* out) : out_(out) {}

  void write_int(int32_t v) {
    uint32_t u = static_cast<uint32_t>(v);
    char b[4] = {static_cast<char>(u >> 24), static_cast<char>(u >> 16), static_cast<char>(u >> 8),
                 static_cast<char>(u)};
    out_->append(b, 4);
  }

  void write_long(int64_t v) {
    uint64_t u = static_cast<uint64_t>(v);
    char b[8];
    for (int i = 0; i < 8; i++) b[i] = static_cast<char>(u >> (56 - 8 * i));
    out_->append(b, 8);
  }

  void write_bool(bool v) { out_->push_back(v ? 1 : 0); }

  void write_string(const std::string& s) {
    write_int(static_cast<int32_t>(s.size()));
    out_->append(s);
  }

  void write_buffer(const std::string& s) { write_string(s); }

  void write_null_buffer() { write_int(-1); }

 private:
  std::string* out_;
};

class JuteReader {
 public:
  JuteReader(const char* data, size_t len) : data_(data), len_(len) {}
  explicit JuteReader(const std::string& s) : data_(s.data()), len_(s.size()) {}

  size_t remaining() const { return len_ - pos_; }
  size_t pos() const { return pos_; }

  int32_t read_int() {
    need(4);
    uint32_t v = 0;
    for (int i = 0; i < 4; i++) v = (v << 8) | static_cast<uint8_t>(data_[pos_ + i]);
    pos_ += 4;
    return static_cast<int32_t>(v);
  }

  int64_t read_long() {
    need(8);
    uint64_t v = 0;
    for (int i = 0; i < 8; i++) v = (v << 8) | static_cast<uint8_t>(data_[pos_ + i]);
    pos_ += 8;
    return static_cast<int64_t>(v);
  }

  bool read_bool() {
    need(1);
    return data_[pos_++] != 0;
  }

  std::string read_string() {
    int32_t n = read_int();
    if (n < 0) return std::string();  // null string/buffer
    need(static_cast<size_t>(n));
    std::string s(data_ + pos_, static_cast<size_t>(n));
    pos_ += static_cast<size_t>(n);
    return s;
  }

  std::string read_buffer() { return read_string(); }

  // Validate-and-skip a length-prefixed string without materializing it
  // (hot-path requests carry ACL scheme/id strings the server discards).
  void skip_string() {
    int32_t n = read_int();
    if (n < 0) return;  // null string/buffer
    need(static_cast<size_t>(n));
    pos_ += static_cast<size_t>(n);
  }

  // Count prefix of a jute vector, bounded by what the frame can actually
  // hold (each element is ≥ min_elem bytes): a hostile count like 2^29 must
  // throw immediately instead of allocating/constructing for minutes and
  // wedging the IO loop (found by tests/test_fuzz.py hostile-frame cases).
  int32_t read_vec_count(size_t min_elem) {
    int32_t n = read_int();
    if (n < 0) return 0;  // null vector
    if (min_elem == 0) min_elem = 1;
    if (static_cast<size_t>(n) > remaining() / min_elem)
      throw std::runtime_error("jute: vector count exceeds frame");
    return n;
  }

 private:
  void need(size_t n) const {
    if (pos_ + n > len_) throw std::runtime_error("jute: short read");
  }

  const char* data_;
  size_t len_;
  size_t pos_ = 0;
};

// ---------------------------------------------------------------------------
// records

struct Stat {
  int64_t czxid = 0;
  int64_t mzxid = 0;
  int64_t ctime = 0;
  int64_t mtime = 0;
  int32_t version = 0;
  int32_t cversion = 0;
  int32_t aversion = 0;
  int64_t ephemeral_owner = 0;
  int32_t data_length = 0;
  int32_t num_children = 0;
  int64_t pzxid = 0;

  void serialize(JuteWriter& w) const {
    w.write_long(czxid);
    w.write_long(mzxid);
    w.write_long(ctime);
    w.write_long(mtime);
    w.write_int(version);
    w.write_int(cversion);
    w.write_int(aversion);
    w.write_long(ephemeral_owner);
    w.write_int(data_length);
    w.write_int(num_children);
    w.write_long(pzxid);
  }

  void deserialize(JuteReader& r) {
    czxid = r.read_long();
    mzxid = r.read_long();
    ctime = r.read_long();
    mtime = r.read_long();
    version = r.read_int();
    cversion = r.read_int();
    aversion = r.read_int();
    ephemeral_owner = r.read_long();
    data_length = r.read_int();
    num_children = r.read_int();
    pzxid = r.read_long();
  }
};

struct ConnectRequest {
  int32_t protocol_version = 0;
  int64_t last_zxid_seen = 0;
  int32_t time_out_ms = 30000;
  int64_t session_id = 0;
  std::string passwd;  // 16 bytes
  bool read_only = false;
  bool has_read_only = false;  // 3.4+ clients append it; tolerate both

  void serialize(JuteWriter& w) const {
    w.write_int(protocol_version);
    w.write_long(last_zxid_seen);
    w.write_int(time_out_ms);
    w.write_long(session_id);
    w.write_buffer(passwd);
    if (has_read_only) w.write_bool(read_only);
  }

  void deserialize(JuteReader& r) {
    protocol_version = r.read_int();
    last_zxid_seen = r.read_long();
    time_out_ms = r.read_int();
    session_id = r.read_long();
    passwd = r.read_buffer();
    has_read_only = r.remaining() >= 1;
    if (has_read_only) read_only = r.read_bool();
  }
};

struct ConnectResponse {
  int32_t protocol_version = 0;
  int32_t time_out_ms = 0;  // 0 ⇒ session expired / rejected
  int64_t session_id = 0;
  std::string passwd;
  bool read_only = false;
  bool has_read_only = false;

  void serialize(JuteWriter& w) const {
    w.write_int(protocol_version);
    w.write_int(time_out_ms);
    w.write_long(session_id);
    w.write_buffer(passwd);
    if (has_read_only) w.write_bool(read_only);
  }

  void deserialize(JuteReader& r) {
    protocol_version = r.read_int();
    time_out_ms = r.read_int();
    session_id = r.read_long();
    passwd = r.read_buffer();
    has_read_only = r.remaining() >= 1;
    if (has_read_only) read_only = r.read_bool();
  }
};

struct RequestHeader {
  int32_t xid = 0;
  int32_t type = 0;

  void serialize(JuteWriter& w) const {
    w.write_int(xid);
    w.write_int(type);
  }
  void deserialize(JuteReader& r) {
    xid = r.read_int();
    type = r.read_int();
  }
};

struct ReplyHeader {
  int32_t xid = 0;
  int64_t zxid = 0;
  int32_t err = 0;

  void serialize(JuteWriter& w) const {
    w.write_int(xid);
    w.write_long(zxid);
    w.write_int(err);
  }
  void deserialize(JuteReader& r) {
    xid = r.read_int();
    zxid = r.read_long();
    err = r.read_int();
  }
};

struct ACL {
  int32_t perms = 31;  // ZOO_PERM_ALL
  std::string scheme = "world";
  std::string id = "anyone";

  void serialize(JuteWriter& w) const {
    w.write_int(perms);
    w.write_string(scheme);
    w.write_string(id);
  }
  void deserialize(JuteReader& r) {
    perms = r.read_int();
    scheme = r.read_string();
    id = r.read_string();
  }
};

inline void write_acl_vector(JuteWriter& w, const std::vector<ACL>& acls) {
  w.write_int(static_cast<int32_t>(acls.size()));
  for (const auto& a : acls) a.serialize(w);
}

inline std::vector<ACL> read_acl_vector(JuteReader& r) {
  // each ACL is ≥ 12 bytes (perms + two length-prefixed strings)
  int32_t n = r.read_vec_count(12);
  std::vector<ACL> acls;
  if (n > 0) {
    acls.resize(static_cast<size_t>(n));
    for (auto& a : acls) a.deserialize(r);
  }
  return acls;
}

struct CreateRequest {
  std::string path;
  std::string data;
  std::vector<ACL> acls{ACL{}};
  int32_t flags = 0;

  void serialize(JuteWriter& w) const {
    w.write_string(path);
    w.write_buffer(data);
    write_acl_vector(w, acls);
    w.write_int(flags);
  }
  void deserialize(JuteReader& r) {
    // validate-and-skip the ACL vector instead of materializing it: the
    // open-ACL server discards it, and a 1k-create burst would otherwise
    // allocate 2 strings per op just to free them (hot path)
    path = r.read_string();
    data = r.read_buffer();
    int32_t n = r.read_vec_count(12);
    for (int32_t i = 0; i < n; i++) {
      r.read_int();    // perms
      r.skip_string();  // scheme
      r.skip_string();  // id
    }
    flags = r.read_int();
  }
};

struct CreateResponse {
  std::string path;
  void serialize(JuteWriter& w) const { w.write_string(path); }
  void deserialize(JuteReader& r) { path = r.read_string(); }
};

struct DeleteRequest {
  std::string path;
  int32_t version = -1;

  void serialize(JuteWriter& w) const {
    w.write_string(path);
    w.write_int(version);
  }
  void deserialize(JuteReader& r) {
    path = r.read_string();
    version = r.read_int();
  }
};

struct ExistsRequest {
  std::string path;
  bool watch = false;

  void serialize(JuteWriter& w) const {
    w.write_string(path);
    w.write_bool(watch);
  }
  void deserialize(JuteReader& r) {
    path = r.read_string();
    watch = r.read_bool();
  }
};

struct ExistsResponse {
  Stat stat;
  void serialize(JuteWriter& w) const { stat.serialize(w); }
  void deserialize(JuteReader& r) { stat.deserialize(r); }
};

struct GetDataRequest {
  std::string path;
  bool watch = false;

  void serialize(JuteWriter& w) const {
    w.write_string(path);
    w.write_bool(watch);
  }
  void deserialize(JuteReader& r) {
    path = r.read_string();
    watch = r.read_bool();
  }
};

struct GetDataResponse {
  std::string data;
  Stat stat;

  void serialize(JuteWriter& w) const {
    w.write_buffer(data);
    stat.serialize(w);
  }
  void deserialize(JuteReader& r) {
    data = r.read_buffer();
    stat.deserialize(r);
  }
};

struct SetDataRequest {
  std::string path;
  std::string data;
  int32_t version = -1;

  void serialize(JuteWriter& w) const {
    w.write_string(path);
    w.write_buffer(data);
    w.write_int(version);
  }
  void deserialize(JuteReader& r) {
    path = r.read_string();
    data = r.read_buffer();
    version = r.read_int();
  }
};

struct SetDataResponse {
  Stat stat;
  void serialize(JuteWriter& w) const { stat.serialize(w); }
  void deserialize(JuteReader& r) { stat.deserialize(r); }
};

struct GetChildrenRequest {
  std::string path;
  bool watch = false;

  void serialize(JuteWriter& w) const {
    w.write_string(path);
    w.write_bool(watch);
  }
  void deserialize(JuteReader& r) {
    path = r.read_string();
    watch = r.read_bool();
  }
};

struct GetChildrenResponse {
  std::vector<std::string> children;

  void serialize(JuteWriter& w) const {
    w.write_int(static_cast<int32_t>(children.size()));
    for (const auto& c : children) w.write_string(c);
  }
  void deserialize(JuteReader& r) {
    int32_t n = r.read_vec_count(4);  // each child name: 4-byte length prefix
    children.clear();
    children.reserve(static_cast<size_t>(n));
    for (int32_t i = 0; i < n; i++) children.push_back(r.read_string());
  }
};

// multi (op 14): a transaction of create/delete/setData/check ops applied
// atomically. Wire format: repeated {MultiHeader{type, done=false, err} +
// op record}, terminated by MultiHeader{-1, true, -1}. The response uses
// the same shape with result records (create → path; delete/check →
// nothing; error result type -1 → int error code).
struct MultiHeader {
  int32_t type = -1;
  bool done = true;
  int32_t err = -1;

  void serialize(JuteWriter& w) const {
    w.write_int(type);
    w.write_bool(done);
    w.write_int(err);
  }
  void deserialize(JuteReader& r) {
    type = r.read_int();
    done = r.read_bool();
    err = r.read_int();
  }
};

// setWatches (op 101, xid -8): re-arm watches after a same-session
// reconnect; the server fires synthetic events for changes that happened
// after relative_zxid while the client was disconnected.
struct SetWatchesRequest {
  int64_t relative_zxid = 0;
  std::vector<std::string> data_watches;
  std::vector<std::string> exist_watches;
  std::vector<std::string> child_watches;

  static void write_vec(JuteWriter& w, const std::vector<std::string>& v) {
    w.write_int(static_cast<int32_t>(v.size()));
    for (const auto& s : v) w.write_string(s);
  }
  static std::vector<std::string> read_vec(JuteReader& r) {
    int32_t n = r.read_vec_count(4);  // each path: 4-byte length prefix
    std::vector<std::string> v;
    v.reserve(static_cast<size_t>(n));
    for (int32_t i = 0; i < n; i++) v.push_back(r.read_string());
    return v;
  }

  void serialize(JuteWriter& w) const {
    w.write_long(relative_zxid);
    write_vec(w, data_watches);
    write_vec(w, exist_watches);
    write_vec(w, child_watches);
  }
  void deserialize(JuteReader& r) {
    relative_zxid = r.read_long();
    data_watches = read_vec(r);
    exist_watches = read_vec(r);
    child_watches = read_vec(r);
  }
};

struct WatcherEvent {
  int32_t type = kEventNone;
  int32_t state = kStateSyncConnected;
  std::string path;

  void serialize(JuteWriter& w) const {
    w.write_int(type);
    w.write_int(state);
    w.write_string(path);
  }
  void deserialize(JuteReader& r) {
    type = r.read_int();
    state = r.read_int();
    path = r.read_string();
  }
};

// Frame helper: prepend the 4-byte big-endian length to a serialized body.
inline void frame_packet(std::string* body_with_4byte_hole) {
  std::string& s = *body_with_4byte_hole;
  uint32_t n = static_cast<uint32_t>(s.size() - 4);
  s[0] = static_cast<char>(n >> 24);
  s[1] = static_cast<char>(n >> 16);
  s[2] = static_cast<char>(n >> 8);
  s[3] = static_cast<char>(n);
}

// Start a framed packet: reserves the length hole; call frame_packet() after
// serializing the body.
inline void begin_packet(std::string* out) { out->assign(4, '\0'); }

}  // namespace zk
}  // namespace registrar
