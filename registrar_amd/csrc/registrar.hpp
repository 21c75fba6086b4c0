// registrar.hpp — registration engine: domain→path mapping, payload builders,
// the 5-step register pipeline, and unregister.
//
// Re-implements the reference's lib/register.js contract:
//   - domainToPath: reverse dot-labels, join with '/' (lib/register.js:34-39)
//   - node list = $path/$(hostname) plus one node per alias, each alias being
//     a full domain independently reversed (lib/register.js:217-227)
//   - pipeline: cleanupPreviousEntries → settle wait (1000 ms default,
//     configurable here) → setupDirectories (mkdirp) → registerEntries
//     (ephemeral creates) → registerService (persistent put), order preserved
//     (lib/register.js:228-239)
//   - host-record payload {type, address, ttl?, [type]: {address, ports?}}
//     with zkplus JSON semantics: undefined keys dropped
//     (lib/register.js:140-159; exact shapes test/register.test.js:122-153)
//   - service-record payload {type:'service', service: <verbatim>} with
//     service.service.ttl defaulted to 60 (lib/register.js:45-75, 197)
//   - unregister deletes ALL znodes (fixing the reference bug where the
//     outer callback fired after the first unlink, lib/register.js:271-284,
//     SURVEY.md §2.2.1) and is idempotent (NO_NODE tolerated).
//
// MI355X extension (north star): when a GpuInfo is attached, the host record
// carries a "gpu" object {index, xgmiRank} so consumers can prefer
// xGMI-local peers.
#pragma once

#include <optional>
#include <string>
#include <vector>

#include "json.hpp"
#include "log.hpp"
#include "zkclient.hpp"

namespace registrar {

// DNS SRV service block: {type:'service', service:{srvce, proto, port, ttl?}}
struct ServiceConfig {
  std::string srvce;  // e.g. "_http"
  std::string proto;  // e.g. "_tcp"
  int64_t port = 0;
  int64_t ttl = 60;  // defaulted during validation (lib/register.js:197)
};

struct GpuInfo {
  int index = -1;      // HIP device index this registrar fronts
  int xgmi_rank = -1;  // xGMI-local rank within the node (topology order)
  std::string uuid;    // GPU UUID when discoverable
};

struct RegistrationConfig {
  std::string domain;                  // required
  std::string type;                    // required (host / load_balancer / ...)
  std::optional<int64_t> ttl;          // optional
  std::vector<int64_t> ports;          // optional
  std::vector<std::string> aliases;    // optional, full domains
  std::optional<ServiceConfig> service;
  std::string admin_ip;                // empty ⇒ first non-internal interface
  std::string hostname;                // empty ⇒ gethostname()
  std::optional<GpuInfo> gpu;          // MI355X extension
  int64_t settle_ms = 1000;            // post-cleanup watcher settle delay
                                       // (fixed 1000 ms in the reference,
                                       // lib/register.js:232-235)
  // atomic swap (beyond the reference): replace stale znodes and create the
  // new set in ONE ZooKeeper multi transaction — consumers never observe a
  // partially-registered domain (the reference's cleanup→create gap is what
  // its 1 s settle delay papered over). settle_ms is moot in this mode.
  bool atomic_swap = false;
};

// Parse + validate a `registration` JSON block (schema: SURVEY.md §2.5).
// Throws std::runtime_error naming the offending field.
RegistrationConfig parse_registration(const Json& j);

// 1.moray.us-east.joyent.com → /com/joyent/us-east/moray/1
std::string domain_to_path(const std::string& domain);

// adminIp if set, else the first address of the first non-internal interface
// (reference lib/register.js:22-31; README recommends adminIp — §2.2.6).
std::string self_address(const std::string& admin_ip);

std::string self_hostname();

// The ephemeral host-record JSON (lib/register.js:140-159 + gpu extension).
Json build_host_record(const RegistrationConfig& cfg);

// The persistent service-record JSON (lib/register.js:45-75).
Json build_service_record(const RegistrationConfig& cfg);

// The znode list register() will create: $path/$hostname + reversed aliases.
std::vector<std::string> build_node_list(const RegistrationConfig& cfg);

struct RegisterResult {
  int rc = 0;                        // zk::kZOk on success
  std::string error;                 // human-readable failure
  std::vector<std::string> znodes;   // nodes to heartbeat (incl. service path)
};

// All the derived state a register() needs — node list, deduped mkdirp
// prefix chain, serialized payloads — computed once so repeated
// registrations (expiry storms, health recoveries, benchmarks) spend their
// time on the wire, not re-deriving strings.
struct PreparedRegistration {
  RegistrationConfig cfg;
  std::string path;                  // domainToPath(domain)
  std::vector<std::string> nodes;    // ephemeral znodes to create
  std::vector<std::string> dirs;     // sorted unique prefixes (parents first)
  std::string host_payload;          // serialized host record
  std::string service_payload;       // serialized service record ("" if none)
  // the full wire sequence (cleanup deletes, dir creates, ephemeral
  // creates) built once; with settle_ms == 0 it ships as a single pipelined
  // round trip (see zkclient submit_mixed)
  std::vector<zk::ZkClient::MixedOp> wire_ops;
  // pre-serialized request streams (xid-patched per submission): the whole
  // register sequence and the heartbeat exists-sweep
  zk::ZkClient::BatchTemplate register_tpl;   // == wire_ops serialized
  zk::ZkClient::BatchTemplate rest_tpl;       // wire_ops minus cleanup (settle path)
  zk::ZkClient::BatchTemplate heartbeat_tpl;  // exists over heartbeat_nodes
  std::vector<std::string> heartbeat_nodes;   // nodes (+ service path)
};

PreparedRegistration prepare_registration(const RegistrationConfig& cfg);

// Full 5-step pipeline against a connected client.
RegisterResult register_prepared(zk::ZkClient& client, PreparedRegistration& prep, const Logger& log);
// App-level heartbeat over the prepared node set (pre-serialized sweep).
int heartbeat_prepared(zk::ZkClient& client, PreparedRegistration& prep,
                       const zk::RetryPolicy& retry, int64_t* rtt_us);
RegisterResult register_node(zk::ZkClient& client, const RegistrationConfig& cfg, const Logger& log);

// Delete every znode; idempotent. Returns zk::kZOk or the first hard error.
int unregister_node(zk::ZkClient& client, const std::vector<std::string>& znodes, const Logger& log);

}  // namespace registrar
