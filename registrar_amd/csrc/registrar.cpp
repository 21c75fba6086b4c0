// registrar.cpp — registration engine implementation (see registrar.hpp).
#include "registrar.hpp"

#include <ifaddrs.h>
#include <net/if.h>
#include <netinet/in.h>
#include <sys/socket.h>
#include <unistd.h>

#include <algorithm>
#include <arpa/inet.h>
#include <chrono>
#include <cstring>
#include <set>
#include <thread>

namespace registrar {

std::string domain_to_path(const std::string& domain) {
  std::string lower;
  lower.reserve(domain.size());
  for (char c : domain) lower += static_cast<char>(tolower(static_cast<unsigned char>(c)));
  std::vector<std::string> labels;
  size_t start = 0;
  while (true) {
    size_t dot = lower.find('.', start);
    if (dot == std::string::npos) {
      labels.push_back(lower.substr(start));
      break;
    }
    labels.push_back(lower.substr(start, dot - start));
    start = dot + 1;
  }
  std::string path;
  for (auto it = labels.rbegin(); it != labels.rend(); ++it) {
    path += '/';
    path += *it;
  }
  return path;
}

std::string self_hostname() {
  char buf[256] = {0};
  if (gethostname(buf, sizeof(buf) - 1) != 0) return "localhost";
  return buf;
}

std::string self_address(const std::string& admin_ip) {
  if (!admin_ip.empty()) return admin_ip;
  struct ifaddrs* ifs = nullptr;
  if (getifaddrs(&ifs) != 0) return "127.0.0.1";
  std::string addr = "127.0.0.1";
  for (struct ifaddrs* i = ifs; i; i = i->ifa_next) {
    if (!i->ifa_addr || i->ifa_addr->sa_family != AF_INET) continue;
    if (i->ifa_flags & IFF_LOOPBACK) continue;
    char buf[INET_ADDRSTRLEN];
    auto* sin = reinterpret_cast<struct sockaddr_in*>(i->ifa_addr);
    if (inet_ntop(AF_INET, &sin->sin_addr, buf, sizeof(buf))) {
      addr = buf;
      break;
    }
  }
  freeifaddrs(ifs);
  return addr;
}

RegistrationConfig parse_registration(const Json& j) {
  if (!j.is_object()) throw std::runtime_error("registration: must be an object");
  RegistrationConfig cfg;
  const Json* domain = j.find("domain");
  if (!domain || !domain->is_string()) throw std::runtime_error("registration.domain: string required");
  cfg.domain = domain->as_string();
  const Json* type = j.find("type");
  if (!type || !type->is_string()) throw std::runtime_error("registration.type: string required");
  cfg.type = type->as_string();
  if (const Json* ttl = j.find("ttl")) {
    if (!ttl->is_number()) throw std::runtime_error("registration.ttl: number required");
    cfg.ttl = ttl->as_int();
  }
  if (const Json* ports = j.find("ports")) {
    if (!ports->is_array()) throw std::runtime_error("registration.ports: array of numbers required");
    for (const auto& p : ports->items()) {
      if (!p.is_number()) throw std::runtime_error("registration.ports: array of numbers required");
      cfg.ports.push_back(p.as_int());
    }
  }
  if (const Json* aliases = j.find("aliases")) {
    if (!aliases->is_array()) throw std::runtime_error("registration.aliases: array of strings required");
    for (const auto& a : aliases->items()) {
      if (!a.is_string()) throw std::runtime_error("registration.aliases: array of strings required");
      cfg.aliases.push_back(a.as_string());
    }
  }
  if (const Json* svc = j.find("service")) {
    // {type:'service', service:{srvce, proto, port, ttl?}} (lib/register.js:187-199)
    if (!svc->is_object()) throw std::runtime_error("registration.service: object required");
    if (svc->get_string("type", "") != "service")
      throw std::runtime_error("registration.service.type: must be 'service'");
    const Json* inner = svc->find("service");
    if (!inner || !inner->is_object()) throw std::runtime_error("registration.service.service: object required");
    ServiceConfig sc;
    const Json* srvce = inner->find("srvce");
    if (!srvce || !srvce->is_string()) throw std::runtime_error("registration.service.service.srvce: string required");
    sc.srvce = srvce->as_string();
    const Json* proto = inner->find("proto");
    if (!proto || !proto->is_string()) throw std::runtime_error("registration.service.service.proto: string required");
    sc.proto = proto->as_string();
    const Json* port = inner->find("port");
    if (!port || !port->is_number()) throw std::runtime_error("registration.service.service.port: number required");
    sc.port = port->as_int();
    if (const Json* ttl = inner->find("ttl")) {
      if (!ttl->is_number()) throw std::runtime_error("registration.service.service.ttl: number required");
      sc.ttl = ttl->as_int();
    }  // else defaulted to 60 (lib/register.js:197)
    cfg.service = sc;
  }
  cfg.admin_ip = j.get_string("adminIp", "");
  cfg.hostname = j.get_string("hostname", "");
  if (const Json* gpu = j.find("gpu")) {
    if (gpu->is_object()) {
      GpuInfo gi;
      gi.index = static_cast<int>(gpu->get_int("index", -1));
      gi.xgmi_rank = static_cast<int>(gpu->get_int("xgmiRank", -1));
      gi.uuid = gpu->get_string("uuid", "");
      cfg.gpu = gi;
    }
  }
  if (const Json* settle = j.find("settleMs")) {
    if (settle->is_number()) cfg.settle_ms = settle->as_int();
  }
  cfg.atomic_swap = j.get_bool("atomicSwap", false);
  return cfg;
}

Json build_host_record(const RegistrationConfig& cfg) {
  // {type, address, ttl?, [type]: {address, ports?, gpu?}}
  // (lib/register.js:140-159; expected shapes test/register.test.js:122-153)
  std::string address = self_address(cfg.admin_ip);
  Json rec = Json::object();
  rec.set("type", Json(cfg.type));
  rec.set("address", Json(address));
  if (cfg.ttl) rec.set("ttl", Json(*cfg.ttl));
  Json inner = Json::object();
  inner.set("address", Json(address));
  // ports: registration.ports, else [service.service.port], else absent
  // (lib/register.js:146-151)
  if (!cfg.ports.empty()) {
    Json ports = Json::array();
    for (int64_t p : cfg.ports) ports.push_back(Json(p));
    inner.set("ports", std::move(ports));
  } else if (cfg.service) {
    Json ports = Json::array();
    ports.push_back(Json(cfg.service->port));
    inner.set("ports", std::move(ports));
  }
  if (cfg.gpu) {
    Json gpu = Json::object();
    gpu.set("index", Json(static_cast<int64_t>(cfg.gpu->index)));
    gpu.set("xgmiRank", Json(static_cast<int64_t>(cfg.gpu->xgmi_rank)));
    if (!cfg.gpu->uuid.empty()) gpu.set("uuid", Json(cfg.gpu->uuid));
    inner.set("gpu", std::move(gpu));
  }
  rec.set(cfg.type, std::move(inner));
  return rec;
}

Json build_service_record(const RegistrationConfig& cfg) {
  // {type:'service', service:{type:'service', service:{srvce,proto,ttl,port}}}
  // — the registration.service block verbatim with ttl defaulted
  // (lib/register.js:45-75; shape test/register.test.js:176-182)
  Json svc = Json::object();
  svc.set("srvce", Json(cfg.service->srvce));
  svc.set("proto", Json(cfg.service->proto));
  svc.set("ttl", Json(cfg.service->ttl));
  svc.set("port", Json(cfg.service->port));
  Json typed = Json::object();
  typed.set("type", Json("service"));
  typed.set("service", std::move(svc));
  Json rec = Json::object();
  rec.set("type", Json("service"));
  rec.set("service", std::move(typed));
  return rec;
}

std::vector<std::string> build_node_list(const RegistrationConfig& cfg) {
  std::string p = domain_to_path(cfg.domain);
  std::string host = cfg.hostname.empty() ? self_hostname() : cfg.hostname;
  std::vector<std::string> nodes;
  nodes.push_back(p + "/" + host);
  // each alias is a full domain, independently reversed (lib/register.js:217-227)
  for (const auto& a : cfg.aliases) nodes.push_back(domain_to_path(a));
  return nodes;
}

namespace {
std::string dirname_of(const std::string& p) {
  size_t pos = p.rfind('/');
  if (pos == std::string::npos || pos == 0) return "/";
  return p.substr(0, pos);
}
}  // namespace

PreparedRegistration prepare_registration(const RegistrationConfig& cfg) {
  PreparedRegistration prep;
  prep.cfg = cfg;
  prep.path = domain_to_path(cfg.domain);
  prep.nodes = build_node_list(cfg);
  std::set<std::string> prefixes;
  for (const auto& n : prep.nodes) {
    std::string dir = dirname_of(n);
    size_t pos = 0;
    while ((pos = dir.find('/', pos + 1)) != std::string::npos) prefixes.insert(dir.substr(0, pos));
    if (dir != "/") prefixes.insert(dir);
  }
  // lexicographic order puts every parent before its children
  prep.dirs.assign(prefixes.begin(), prefixes.end());
  prep.host_payload = build_host_record(cfg).dump();
  if (cfg.service) prep.service_payload = build_service_record(cfg).dump();
  prep.wire_ops.reserve(prep.nodes.size() * 2 + prep.dirs.size());
  for (const auto& n : prep.nodes) {
    zk::ZkClient::MixedOp op;
    op.op = zk::kOpDelete;
    op.path = n;
    prep.wire_ops.push_back(std::move(op));
  }
  for (const auto& d : prep.dirs) {
    zk::ZkClient::MixedOp op;
    op.op = zk::kOpCreate;
    op.path = d;
    prep.wire_ops.push_back(std::move(op));
  }
  for (const auto& n : prep.nodes) {
    zk::ZkClient::MixedOp op;
    op.op = zk::kOpCreate;
    op.path = n;
    op.data = prep.host_payload;
    op.flags = zk::kEphemeral;
    prep.wire_ops.push_back(std::move(op));
  }
  prep.register_tpl = zk::ZkClient::make_template(prep.wire_ops);
  prep.rest_tpl = zk::ZkClient::make_template(
      std::vector<zk::ZkClient::MixedOp>(prep.wire_ops.begin() + static_cast<long>(prep.nodes.size()),
                                         prep.wire_ops.end()));
  prep.heartbeat_nodes = prep.nodes;
  if (cfg.service) prep.heartbeat_nodes.push_back(prep.path);
  prep.heartbeat_tpl = zk::ZkClient::make_exists_template(prep.heartbeat_nodes);
  return prep;
}

int heartbeat_prepared(zk::ZkClient& client, PreparedRegistration& prep, const zk::RetryPolicy& retry,
                       int64_t* rtt_us) {
  return client.heartbeat_template(prep.heartbeat_tpl, retry, rtt_us);
}

RegisterResult register_prepared(zk::ZkClient& client, PreparedRegistration& prep, const Logger& log) {
  RegisterResult result;
  const RegistrationConfig& cfg = prep.cfg;
  Logger rlog =
      log.child({{"component", Json("register")}, {"domain", Json(cfg.domain)}, {"path", Json(prep.path)}});
  rlog.debug("register: entered");

  size_t n_nodes = prep.nodes.size();
  size_t n_dirs = prep.dirs.size();

  auto check_cleanup = [&](const std::vector<int>& rcs, size_t base) -> bool {
    // cleanupPreviousEntries: NO_NODE tolerated (lib/register.js:78-105)
    for (size_t i = 0; i < n_nodes; i++) {
      int rc = rcs[base + i];
      if (rc != zk::kZOk && rc != zk::kZNoNode) {
        result.rc = rc;
        result.error =
            std::string("cleanupPreviousEntries: unlink ") + prep.nodes[i] + " failed: " + zk::error_name(rc);
        return false;
      }
    }
    return true;
  };
  auto check_dirs = [&](const std::vector<int>& rcs, size_t base) -> bool {
    // setupDirectories: NODE_EXISTS tolerated (lib/register.js:108-129)
    for (size_t i = 0; i < n_dirs; i++) {
      int rc = rcs[base + i];
      if (rc != zk::kZOk && rc != zk::kZNodeExists) {
        result.rc = rc;
        result.error = std::string("setupDirectories: mkdirp ") + prep.dirs[i] + " failed: " + zk::error_name(rc);
        return false;
      }
    }
    return true;
  };
  auto check_creates = [&](const std::vector<int>& rcs, size_t base) -> bool {
    // registerEntries: every ephemeral create must succeed
    for (size_t i = 0; i < n_nodes; i++) {
      int rc = rcs[base + i];
      if (rc != zk::kZOk) {
        result.rc = rc;
        result.error =
            std::string("registerEntries: create ") + prep.nodes[i] + " failed: " + zk::error_name(rc);
        return false;
      }
    }
    return true;
  };

  if (cfg.atomic_swap) {
    // ---- atomic swap (beyond the reference) ----
    // 1) parents (idempotent, persistent, usually a no-op)
    {
      std::vector<std::string> datas(prep.dirs.size());
      std::vector<int> rcs = client.create_many(prep.dirs, datas, 0);
      if (!check_dirs(rcs, 0)) {
        rlog.debug("setupDirectories: failed", {{"err", Json(result.error)}});
        return result;
      }
    }
    // 2) read current state, 3) swap in one transaction; retry on races
    for (int attempt = 0;; attempt++) {
      std::vector<int> ex = client.exists_many(prep.nodes, nullptr);
      std::vector<zk::ZkClient::MixedOp> ops;
      ops.reserve(prep.nodes.size() * 2);
      for (size_t i = 0; i < prep.nodes.size(); i++) {
        if (ex[i] == zk::kZOk) {
          zk::ZkClient::MixedOp del;
          del.op = zk::kOpDelete;
          del.path = prep.nodes[i];
          ops.push_back(std::move(del));
        } else if (ex[i] != zk::kZNoNode) {
          result.rc = ex[i];
          result.error = std::string("atomicSwap: exists ") + prep.nodes[i] + " failed: " +
                         zk::error_name(ex[i]);
          return result;
        }
      }
      for (size_t i = 0; i < prep.nodes.size(); i++) {
        zk::ZkClient::MixedOp cr;
        cr.op = zk::kOpCreate;
        cr.path = prep.nodes[i];
        cr.data = prep.host_payload;
        cr.flags = zk::kEphemeral;
        ops.push_back(std::move(cr));
      }
      std::vector<int> per_op;
      int rc = client.multi(ops, &per_op);
      if (rc == zk::kZOk) break;
      bool race = (rc == zk::kZNodeExists || rc == zk::kZNoNode);
      if (!race || attempt >= 3) {
        result.rc = rc;
        result.error = std::string("atomicSwap: multi failed: ") + zk::error_name(rc);
        rlog.debug("atomicSwap: failed", {{"err", Json(result.error)}});
        return result;
      }
      // another session created/removed one of our nodes between the read
      // and the transaction — re-read and retry
    }
  } else if (cfg.settle_ms > 0) {
    // settle configured: a real barrier after cleanup (reference semantics,
    // fixed 1000 ms at lib/register.js:232-235), then the remaining stages
    // in one pipelined round trip
    std::vector<int> rcs = client.delete_many(prep.nodes);
    if (!check_cleanup(rcs, 0)) {
      rlog.debug("cleanupPreviousEntries: failed", {{"err", Json(result.error)}});
      return result;
    }
    std::this_thread::sleep_for(std::chrono::milliseconds(cfg.settle_ms));
    std::vector<int> rcs2 = client.submit_template(prep.rest_tpl);
    if (!check_dirs(rcs2, 0) || !check_creates(rcs2, n_dirs)) {
      rlog.debug("register: failed", {{"err", Json(result.error)}});
      return result;
    }
  } else {
    // no settle: the whole cleanup → mkdirp → create sequence is one
    // pipelined submission — ZooKeeper's per-session in-order processing
    // guarantees the same final state as the reference's staged barriers
    std::vector<int> rcs = client.submit_template(prep.register_tpl);
    if (!check_cleanup(rcs, 0) || !check_dirs(rcs, n_nodes) || !check_creates(rcs, n_nodes + n_dirs)) {
      rlog.debug("register: failed", {{"err", Json(result.error)}});
      return result;
    }
  }

  result.znodes = prep.nodes;

  // 5) registerService: persistent put of the service record at $path itself,
  //    appended to the heartbeat node list (lib/register.js:45-75)
  if (cfg.service) {
    int rc = client.put(prep.path, prep.service_payload);
    if (rc != zk::kZOk) {
      result.rc = rc;
      result.error = std::string("registerService: put ") + prep.path + " failed: " + zk::error_name(rc);
      rlog.error("registerService: put failed", {{"err", Json(result.error)}});
      result.znodes.clear();
      return result;
    }
    if (std::find(result.znodes.begin(), result.znodes.end(), prep.path) == result.znodes.end())
      result.znodes.push_back(prep.path);
    rlog.debug("registerService: done");
  }

  result.rc = zk::kZOk;
  {
    Json zn = Json::array();
    for (const auto& n : result.znodes) zn.push_back(Json(n));
    rlog.debug("register: done", {{"znodes", std::move(zn)}});
  }
  return result;
}

RegisterResult register_node(zk::ZkClient& client, const RegistrationConfig& cfg, const Logger& log) {
  PreparedRegistration prep = prepare_registration(cfg);
  return register_prepared(client, prep, log);
}

int unregister_node(zk::ZkClient& client, const std::vector<std::string>& znodes, const Logger& log) {
  Logger ulog = log.child("unregister");
  ulog.debug("unregister: entered");
  // ALL nodes are deleted (pipelined) — fixing the reference bug where the
  // caller's callback fired after the first unlink and the remaining znodes
  // were never processed (lib/register.js:271-284).
  // NO_NODE is tolerated so unregister is idempotent.
  std::vector<int> rcs = client.delete_many(znodes);
  int rc = zk::kZOk;
  for (size_t i = 0; i < rcs.size(); i++) {
    if (rcs[i] != zk::kZOk && rcs[i] != zk::kZNoNode) {
      ulog.debug("unregister: failed to delete node",
                 {{"node", Json(znodes[i])}, {"err", Json(zk::error_name(rcs[i]))}});
      if (rc == zk::kZOk) rc = rcs[i];
    }
  }
  ulog.debug("unregister: done");
  return rc;
}

}  // namespace registrar
