// bindings.cpp — pybind11 module `registrar_amd._core`.
//
// Exposes the native ensemble, client, registration engine, health checker,
// orchestrator and GPU topology probes to Python for tests and the bench
// harness. Design rules:
//   - every blocking native call releases the GIL,
//   - events flow through thread-safe poll queues only (no Python callbacks
//     from native threads), so native threads never need the GIL.
#include <pybind11/pybind11.h>
#include <pybind11/stl.h>

#include "ensemble.hpp"
#include "gpu.hpp"
#include "health.hpp"
#include "json.hpp"
#include "log.hpp"
#include "orchestrator.hpp"
#include "registrar.hpp"
#include "zkclient.hpp"

namespace py = pybind11;
using namespace registrar;

namespace {

LogLevel level_from(const std::string& name) {
  LogLevel l = LogLevel::Warn;
  log_level_from_name(name, &l);
  return l;
}

Logger make_logger(const std::string& name, const std::string& level) {
  Logger log(name);
  log.set_level(level_from(level));
  return log;
}

py::dict stat_to_dict(const zk::Stat& st) {
  py::dict d;
  d["czxid"] = st.czxid;
  d["mzxid"] = st.mzxid;
  d["ctime"] = st.ctime;
  d["mtime"] = st.mtime;
  d["version"] = st.version;
  d["cversion"] = st.cversion;
  d["ephemeralOwner"] = st.ephemeral_owner;
  d["dataLength"] = st.data_length;
  d["numChildren"] = st.num_children;
  d["pzxid"] = st.pzxid;
  return d;
}

py::dict session_event_to_dict(const zk::SessionEvent& ev) {
  py::dict d;
  d["type"] = zk::session_event_name(ev.type);
  d["session_id"] = ev.session_id;
  d["attempt"] = ev.attempt;
  d["delay_ms"] = ev.delay_ms;
  return d;
}

py::dict orch_event_to_dict(const OrchEvent& ev) {
  py::dict d;
  d["type"] = orch_event_name(ev.type);
  d["detail"] = ev.detail;
  d["znodes"] = ev.znodes;
  d["rtt_us"] = ev.rtt_us;
  return d;
}

py::dict health_record_to_dict(const HealthRecord& rec) {
  py::dict d;
  d["type"] = rec.ok ? "ok" : "fail";
  d["command"] = rec.command;
  d["error"] = rec.error;
  d["failures"] = rec.failures;
  d["isDown"] = rec.is_down;
  d["threshold"] = rec.threshold;
  d["exit_status"] = rec.exit_status;
  d["stdout"] = rec.stdout_tail;
  d["stderr"] = rec.stderr_tail;
  return d;
}

zk::ZkClientConfig client_config_from_args(const std::vector<std::pair<std::string, int>>& servers,
                                           int session_timeout_ms, int connect_timeout_ms,
                                           int64_t connect_initial_delay_ms, int64_t connect_max_delay_ms,
                                           int64_t connect_max_attempts, bool randomize_start,
                                           const std::string& log_level) {
  zk::ZkClientConfig cfg;
  for (const auto& s : servers) cfg.servers.push_back({s.first, s.second});
  cfg.session_timeout_ms = session_timeout_ms;
  cfg.connect_timeout_ms = connect_timeout_ms;
  cfg.connect_initial_delay_ms = connect_initial_delay_ms;
  cfg.connect_max_delay_ms = connect_max_delay_ms;
  cfg.connect_max_attempts = connect_max_attempts;
  cfg.randomize_start = randomize_start;
  cfg.log_level = level_from(log_level);
  return cfg;
}

zk::RetryPolicy retry_from_dict(const py::dict& d) {
  zk::RetryPolicy rp;
  if (d.contains("maxAttempts")) rp.max_attempts = d["maxAttempts"].cast<int64_t>();
  if (d.contains("initialDelay")) rp.initial_delay_ms = d["initialDelay"].cast<int64_t>();
  if (d.contains("maxDelay")) rp.max_delay_ms = d["maxDelay"].cast<int64_t>();
  return rp;
}

}  // namespace

PYBIND11_MODULE(_core, m) {
  m.doc() = "registrar_amd native core: ZK jute client, synthetic ensemble, registration engine";

  // ---- error codes ----
  m.attr("ZOK") = static_cast<int>(zk::kZOk);
  m.attr("ZNONODE") = static_cast<int>(zk::kZNoNode);
  m.attr("ZNODEEXISTS") = static_cast<int>(zk::kZNodeExists);
  m.attr("ZNOTEMPTY") = static_cast<int>(zk::kZNotEmpty);
  m.attr("ZBADVERSION") = static_cast<int>(zk::kZBadVersion);
  m.attr("ZSESSIONEXPIRED") = static_cast<int>(zk::kZSessionExpired);
  m.attr("ZCONNECTIONLOSS") = static_cast<int>(zk::kZConnectionLoss);
  m.attr("ZNOCHILDRENFOREPHEMERALS") = static_cast<int>(zk::kZNoChildrenForEphemerals);
  m.def("error_name", [](int rc) { return std::string(zk::error_name(rc)); });

  // ---- JSON helpers (round-trip sanity from Python) ----
  m.def("json_roundtrip", [](const std::string& text) { return Json::parse(text).dump(); });
  m.def("json_equal",
        [](const std::string& a, const std::string& b) { return Json::parse(a) == Json::parse(b); });

  // ---- Ensemble ----
  py::class_<zk::Ensemble>(m, "Ensemble")
      .def(py::init([](size_t servers, int tick_ms, int min_session_timeout_ms, int max_session_timeout_ms,
                       int latency_ms, int election_ms, const std::string& log_level,
                       const std::vector<int>& ports, const std::string& bind_host, int io_threads) {
             zk::EnsembleConfig cfg;
             if (!ports.empty()) {
               cfg.ports = ports;
             } else {
               cfg.ports.assign(servers, 0);
             }
             cfg.bind_host = bind_host;
             cfg.io_threads = io_threads;
             cfg.tick_ms = tick_ms;
             cfg.min_session_timeout_ms = min_session_timeout_ms;
             cfg.max_session_timeout_ms = max_session_timeout_ms;
             cfg.latency_ms = latency_ms;
             cfg.election_ms = election_ms;
             cfg.log_level = level_from(log_level);
             return std::make_unique<zk::Ensemble>(std::move(cfg));
           }),
           py::arg("servers") = 1, py::arg("tick_ms") = 100, py::arg("min_session_timeout_ms") = 400,
           py::arg("max_session_timeout_ms") = 60000, py::arg("latency_ms") = 0, py::arg("election_ms") = 0,
           py::arg("log_level") = "warn", py::arg("ports") = std::vector<int>{},
           py::arg("bind_host") = "127.0.0.1", py::arg("io_threads") = 0)
      .def("start", &zk::Ensemble::start, py::call_guard<py::gil_scoped_release>())
      .def("stop", &zk::Ensemble::stop, py::call_guard<py::gil_scoped_release>())
      .def("ports", &zk::Ensemble::ports)
      .def("connect_string", &zk::Ensemble::connect_string)
      .def("kill_server", &zk::Ensemble::kill_server, py::call_guard<py::gil_scoped_release>())
      .def("restart_server", &zk::Ensemble::restart_server, py::call_guard<py::gil_scoped_release>())
      .def("server_up", &zk::Ensemble::server_up)
      .def("leader", &zk::Ensemble::leader)
      .def("kill_leader", &zk::Ensemble::kill_leader, py::call_guard<py::gil_scoped_release>())
      .def("expire_session", &zk::Ensemble::expire_session, py::call_guard<py::gil_scoped_release>())
      .def("set_latency_ms", &zk::Ensemble::set_latency_ms)
      .def("get",
           [](const zk::Ensemble& e, const std::string& path) {
             zk::NodeInfo info = e.get(path);
             py::dict d;
             d["exists"] = info.exists;
             d["data"] = py::bytes(info.data);
             d["stat"] = stat_to_dict(info.stat);
             return d;
           })
      .def("children", &zk::Ensemble::children)
      .def("node_count", &zk::Ensemble::node_count)
      .def("ephemeral_count", &zk::Ensemble::ephemeral_count)
      .def("session_ids", &zk::Ensemble::session_ids)
      .def("zxid", &zk::Ensemble::zxid)
      .def("counters", &zk::Ensemble::counters);

  // ---- ZkClient ----
  py::class_<zk::ZkClient>(m, "ZkClient")
      .def(py::init([](const std::vector<std::pair<std::string, int>>& servers, int session_timeout_ms,
                       int connect_timeout_ms, int64_t connect_initial_delay_ms, int64_t connect_max_delay_ms,
                       int64_t connect_max_attempts, bool randomize_start, const std::string& log_level) {
             return std::make_unique<zk::ZkClient>(
                 client_config_from_args(servers, session_timeout_ms, connect_timeout_ms, connect_initial_delay_ms,
                                         connect_max_delay_ms, connect_max_attempts, randomize_start, log_level),
                 make_logger("zkclient", log_level));
           }),
           py::arg("servers"), py::arg("session_timeout_ms") = 30000, py::arg("connect_timeout_ms") = 4000,
           py::arg("connect_initial_delay_ms") = 1000, py::arg("connect_max_delay_ms") = 90000,
           py::arg("connect_max_attempts") = -1, py::arg("randomize_start") = true,
           py::arg("log_level") = "warn")
      .def("start", &zk::ZkClient::start, py::call_guard<py::gil_scoped_release>())
      .def("wait_connected", &zk::ZkClient::wait_connected, py::arg("timeout_ms") = -1,
           py::call_guard<py::gil_scoped_release>())
      .def("abort_connect", &zk::ZkClient::abort_connect, py::call_guard<py::gil_scoped_release>())
      .def("close", &zk::ZkClient::close, py::call_guard<py::gil_scoped_release>())
      .def("state",
           [](const zk::ZkClient& c) {
             switch (c.state()) {
               case zk::SessionState::Connecting:
                 return "connecting";
               case zk::SessionState::Connected:
                 return "connected";
               case zk::SessionState::Expired:
                 return "expired";
               case zk::SessionState::Closed:
                 return "closed";
             }
             return "?";
           })
      .def("session_id", &zk::ZkClient::session_id)
      .def("session_timeout_ms", &zk::ZkClient::session_timeout_ms)
      .def("to_string", &zk::ZkClient::to_string)
      .def("poll_events",
           [](zk::ZkClient& c) {
             py::list out;
             // poll_events itself is quick; no GIL release needed
             for (const auto& ev : c.poll_events()) out.append(session_event_to_dict(ev));
             return out;
           })
      .def("poll_watches",
           [](zk::ZkClient& c) {
             py::list out;
             for (const auto& ev : c.poll_watches()) {
               py::dict d;
               const char* type = "none";
               switch (ev.type) {
                 case zk::kEventNodeCreated: type = "created"; break;
                 case zk::kEventNodeDeleted: type = "deleted"; break;
                 case zk::kEventNodeDataChanged: type = "changed"; break;
                 case zk::kEventNodeChildrenChanged: type = "child"; break;
                 default: break;
               }
               d["type"] = type;
               d["path"] = ev.path;
               out.append(d);
             }
             return out;
           })
      .def("create",
           [](zk::ZkClient& c, const std::string& path, const py::bytes& data, bool ephemeral) {
             std::string d = data;
             std::string created;
             int rc;
             {
               py::gil_scoped_release rel;
               rc = c.create(path, d, ephemeral ? zk::kEphemeral : 0, &created);
             }
             return py::make_tuple(rc, created);
           },
           py::arg("path"), py::arg("data") = py::bytes(""), py::arg("ephemeral") = false)
      .def("delete_", &zk::ZkClient::del, py::arg("path"), py::arg("version") = -1,
           py::call_guard<py::gil_scoped_release>())
      .def("exists",
           [](zk::ZkClient& c, const std::string& path, bool watch) {
             zk::Stat st;
             int rc;
             {
               py::gil_scoped_release rel;
               rc = c.exists(path, &st, watch);
             }
             return py::make_tuple(rc, stat_to_dict(st));
           },
           py::arg("path"), py::arg("watch") = false)
      .def("get",
           [](zk::ZkClient& c, const std::string& path, bool watch) {
             std::string data;
             zk::Stat st;
             int rc;
             {
               py::gil_scoped_release rel;
               rc = c.get(path, &data, &st, watch);
             }
             return py::make_tuple(rc, py::bytes(data), stat_to_dict(st));
           },
           py::arg("path"), py::arg("watch") = false)
      .def("set",
           [](zk::ZkClient& c, const std::string& path, const py::bytes& data, int version) {
             std::string d = data;
             py::gil_scoped_release rel;
             return c.set(path, d, version, nullptr);
           },
           py::arg("path"), py::arg("data"), py::arg("version") = -1)
      .def("get_children",
           [](zk::ZkClient& c, const std::string& path, bool watch) {
             std::vector<std::string> ch;
             int rc;
             {
               py::gil_scoped_release rel;
               rc = c.get_children(path, &ch, watch);
             }
             return py::make_tuple(rc, ch);
           },
           py::arg("path"), py::arg("watch") = false)
      .def("put",
           [](zk::ZkClient& c, const std::string& path, const py::bytes& data) {
             std::string d = data;
             py::gil_scoped_release rel;
             return c.put(path, d);
           })
      .def("mkdirp", &zk::ZkClient::mkdirp, py::call_guard<py::gil_scoped_release>())
      .def("unlink", &zk::ZkClient::unlink, py::call_guard<py::gil_scoped_release>())
      .def("create_many",
           [](zk::ZkClient& c, const std::vector<std::string>& paths, const py::bytes& data, bool ephemeral) {
             std::string d = data;
             std::vector<std::string> datas(paths.size(), d);
             py::gil_scoped_release rel;
             return c.create_many(paths, datas, ephemeral ? zk::kEphemeral : 0);
           },
           py::arg("paths"), py::arg("data") = py::bytes(""), py::arg("ephemeral") = false)
      .def("delete_many", &zk::ZkClient::delete_many, py::call_guard<py::gil_scoped_release>())
      .def("exists_many",
           [](zk::ZkClient& c, const std::vector<std::string>& paths) {
             py::gil_scoped_release rel;
             return c.exists_many(paths, nullptr);
           })
      .def("multi",
           [](zk::ZkClient& c, const py::list& ops) {
             std::vector<zk::ZkClient::MixedOp> mops;
             for (auto item : ops) {
               py::tuple t = item.cast<py::tuple>();
               zk::ZkClient::MixedOp mo;
               std::string kind = t[0].cast<std::string>();
               mo.path = t[1].cast<std::string>();
               if (kind == "delete") {
                 mo.op = zk::kOpDelete;
               } else if (kind == "create") {
                 mo.op = zk::kOpCreate;
                 if (t.size() > 2) mo.data = t[2].cast<std::string>();
                 if (t.size() > 3 && t[3].cast<bool>()) mo.flags = zk::kEphemeral;
               } else {
                 throw std::runtime_error("multi op kind must be 'create' or 'delete'");
               }
               mops.push_back(std::move(mo));
             }
             std::vector<int> per_op;
             int rc;
             {
               py::gil_scoped_release rel;
               rc = c.multi(mops, &per_op);
             }
             return py::make_tuple(rc, per_op);
           },
           py::arg("ops"))
      .def("heartbeat",
           [](zk::ZkClient& c, const std::vector<std::string>& nodes, const py::dict& retry) {
             zk::RetryPolicy rp = retry_from_dict(retry);
             int64_t rtt = 0;
             int rc;
             {
               py::gil_scoped_release rel;
               rc = c.heartbeat(nodes, rp, &rtt);
             }
             return py::make_tuple(rc, rtt);
           },
           py::arg("nodes"), py::arg("retry") = py::dict());

  // ---- registration engine ----
  m.def("domain_to_path", &domain_to_path);
  m.def("self_hostname", &self_hostname);
  m.def("self_address", &self_address, py::arg("admin_ip") = "");
  m.def("build_host_record",
        [](const std::string& registration_json) {
          return build_host_record(parse_registration(Json::parse(registration_json))).dump();
        });
  m.def("build_service_record",
        [](const std::string& registration_json) {
          return build_service_record(parse_registration(Json::parse(registration_json))).dump();
        });
  m.def("build_node_list",
        [](const std::string& registration_json) {
          return build_node_list(parse_registration(Json::parse(registration_json)));
        });
  // Pre-parsed registration: node list / mkdirp chain / payloads derived
  // once, so repeated registrations (storms, benches) are pure wire work.
  py::class_<PreparedRegistration>(m, "PreparedRegistration")
      .def(py::init([](const std::string& registration_json) {
             return std::make_unique<PreparedRegistration>(
                 prepare_registration(parse_registration(Json::parse(registration_json))));
           }),
           py::arg("registration_json"))
      .def_property_readonly("path", [](const PreparedRegistration& p) { return p.path; })
      .def_property_readonly("nodes", [](const PreparedRegistration& p) { return p.nodes; })
      .def_property_readonly("host_payload",
                             [](const PreparedRegistration& p) { return py::bytes(p.host_payload); })
      .def("register_",
           [](PreparedRegistration& prep, zk::ZkClient& c, const std::string& log_level) {
             Logger log = make_logger("registrar", log_level);
             RegisterResult res;
             {
               py::gil_scoped_release rel;
               res = register_prepared(c, prep, log);
             }
             return py::make_tuple(res.rc, res.error, res.znodes);
           },
           py::arg("client"), py::arg("log_level") = "warn")
      .def("heartbeat",
           [](PreparedRegistration& prep, zk::ZkClient& c, const py::dict& retry) {
             zk::RetryPolicy rp = retry_from_dict(retry);
             int64_t rtt = 0;
             int rc;
             {
               py::gil_scoped_release rel;
               rc = heartbeat_prepared(c, prep, rp, &rtt);
             }
             return py::make_tuple(rc, rtt);
           },
           py::arg("client"), py::arg("retry") = py::dict());

  m.def("register_node",
        [](zk::ZkClient& c, const std::string& registration_json, const std::string& log_level) {
          RegistrationConfig cfg = parse_registration(Json::parse(registration_json));
          Logger log = make_logger("registrar", log_level);
          RegisterResult res;
          {
            py::gil_scoped_release rel;
            res = register_node(c, cfg, log);
          }
          return py::make_tuple(res.rc, res.error, res.znodes);
        },
        py::arg("client"), py::arg("registration_json"), py::arg("log_level") = "warn");
  m.def("unregister_node",
        [](zk::ZkClient& c, const std::vector<std::string>& znodes, const std::string& log_level) {
          Logger log = make_logger("registrar", log_level);
          py::gil_scoped_release rel;
          return unregister_node(c, znodes, log);
        },
        py::arg("client"), py::arg("znodes"), py::arg("log_level") = "warn");

  // ---- health checker ----
  m.def("exec_with_timeout",
        [](const std::string& command, int64_t timeout_ms) {
          ExecResult res;
          {
            py::gil_scoped_release rel;
            res = exec_with_timeout(command, timeout_ms);
          }
          py::dict d;
          d["exit_status"] = res.exit_status;
          d["timed_out"] = res.timed_out;
          d["stdout"] = py::bytes(res.out);
          d["stderr"] = py::bytes(res.err);
          return d;
        },
        py::arg("command"), py::arg("timeout_ms") = 1000);

  py::class_<HealthCheck>(m, "HealthCheck")
      .def(py::init([](const std::string& config_json, const std::string& log_level) {
             return std::make_unique<HealthCheck>(parse_health_check(Json::parse(config_json)),
                                                  make_logger("health", log_level));
           }),
           py::arg("config_json"), py::arg("log_level") = "warn")
      .def("start", &HealthCheck::start, py::call_guard<py::gil_scoped_release>())
      .def("stop", &HealthCheck::stop, py::call_guard<py::gil_scoped_release>())
      .def("check_once",
           [](HealthCheck& h) {
             HealthRecord rec;
             {
               py::gil_scoped_release rel;
               rec = h.check_once();
             }
             return health_record_to_dict(rec);
           })
      .def("poll_records",
           [](HealthCheck& h) {
             py::list out;
             for (const auto& r : h.poll_records()) out.append(health_record_to_dict(r));
             return out;
           })
      .def("is_down", &HealthCheck::is_down);

  // ---- orchestrator ----
  py::class_<Orchestrator>(m, "Orchestrator")
      .def(py::init([](const std::string& config_json, const std::string& log_level) {
             OrchestratorConfig cfg = parse_config(Json::parse(config_json));
             std::string lvl = !cfg.log_level.empty() ? cfg.log_level : log_level;
             return std::make_unique<Orchestrator>(std::move(cfg), make_logger("registrar", lvl));
           }),
           py::arg("config_json"), py::arg("log_level") = "warn")
      .def("start", &Orchestrator::start, py::call_guard<py::gil_scoped_release>())
      .def("wait_registered", &Orchestrator::wait_registered, py::arg("timeout_ms") = -1,
           py::call_guard<py::gil_scoped_release>())
      .def("stop", &Orchestrator::stop, py::call_guard<py::gil_scoped_release>())
      .def("heartbeat_now",
           [](Orchestrator& o) {
             int64_t rtt = 0;
             int rc;
             {
               py::gil_scoped_release rel;
               rc = o.heartbeat_now(&rtt);
             }
             return py::make_tuple(rc, rtt);
           })
      .def("znodes", &Orchestrator::znodes)
      .def("poll_events",
           [](Orchestrator& o) {
             py::list out;
             for (const auto& ev : o.poll_events()) out.append(orch_event_to_dict(ev));
             return out;
           })
      .def("metrics",
           [](const Orchestrator& o) {
             OrchMetrics mtx = o.metrics();
             py::dict d;
             d["registers"] = mtx.registers;
             d["unregisters"] = mtx.unregisters;
             d["heartbeats"] = mtx.heartbeats;
             d["heartbeat_failures"] = mtx.heartbeat_failures;
             d["session_expiries"] = mtx.session_expiries;
             d["errors"] = mtx.errors;
             d["recent_heartbeat_rtt_us"] = mtx.recent_heartbeat_rtt_us;
             return d;
           })
      .def("session_id", &Orchestrator::session_id)
      .def("expired", &Orchestrator::expired);

  // ---- GPU topology ----
  m.def("gpu_count", &gpu::gpu_count);
  m.def("xgmi_local_rank", &gpu::xgmi_local_rank);
  m.def("gpu_health_command", &gpu::gpu_health_command);
  m.def("gpu_alive", &gpu::gpu_alive);
  m.def("discover_gpus",
        [](const std::string& root) {
          py::list out;
          for (const auto& g : gpu::discover_gpus(root.empty() ? "/sys/class/kfd/kfd/topology/nodes" : root)) {
            py::dict d;
            d["kfd_node"] = g.kfd_node;
            d["device_index"] = g.device_index;
            d["hive_id"] = g.hive_id;
            d["xgmi_rank"] = g.xgmi_rank;
            d["location_id"] = g.location_id;
            d["name"] = g.name;
            d["uuid"] = g.uuid;
            out.append(d);
          }
          return out;
        },
        py::arg("root") = "");
}
