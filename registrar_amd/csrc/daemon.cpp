// daemon.cpp — registrard: the MI355X-host-native registrar daemon.
//
// CLI-compatible with the reference's main.js: `registrard -f CONFIG [-v]...`
// (-v repeats lower the log level by one step each, main.js:66-76). Loads and
// validates the JSON config (schema SURVEY.md §2.5), runs the orchestrator,
// and logs every lifecycle event as bunyan JSON lines (main.js:160-198),
// including the heartbeat is_down latch so a failing heartbeat logs once
// until recovery (main.js:187-198).
//
// Session-expiry policy: by default the daemon re-registers in-process; with
// --exit-on-expiry (or config exitOnExpiry:true) it logs fatal and exits 1,
// matching the reference's crash-and-restart contract (main.js:141-144) for
// use under systemd/SMF-style supervisors.
#include <getopt.h>
#include <signal.h>
#include <unistd.h>

#include <algorithm>
#include <atomic>
#include <chrono>
#include <cstdio>
#include <fstream>
#include <sstream>
#include <thread>

#include "json.hpp"
#include "log.hpp"
#include "orchestrator.hpp"

using namespace registrar;

#ifndef REGISTRAR_VERSION
#define REGISTRAR_VERSION "0.1.0"
#endif

namespace {

std::atomic<int> g_signal{0};
std::atomic<bool> g_dump_metrics{false};
void on_signal(int sig) { g_signal.store(sig); }
void on_usr1(int) { g_dump_metrics.store(true); }

void usage(const char* argv0, const char* msg) {
  if (msg) fprintf(stderr, "%s\n", msg);
  fprintf(stdout,
          "usage: %s [OPTIONS]\n"
          "options:\n"
          "    -f FILE, --file=FILE      JSON config file (required)\n"
          "    -v, --verbose             lower log level (repeatable)\n"
          "    -e, --exit-on-expiry      exit(1) on ZK session expiry instead of\n"
          "                              re-registering in-process\n"
          "    -h, --help                print this help and exit\n",
          argv0);
  exit(msg ? 1 : 0);
}

LogLevel lower_level(LogLevel l) {
  int v = static_cast<int>(l) - 10;
  if (v < static_cast<int>(LogLevel::Trace)) v = static_cast<int>(LogLevel::Trace);
  return static_cast<LogLevel>(v);
}

}  // namespace

int main(int argc, char** argv) {
  std::string config_file;
  int verbose = 0;
  bool exit_on_expiry_flag = false;

  static struct option long_opts[] = {{"file", required_argument, nullptr, 'f'},
                                      {"verbose", no_argument, nullptr, 'v'},
                                      {"exit-on-expiry", no_argument, nullptr, 'e'},
                                      {"help", no_argument, nullptr, 'h'},
                                      {"version", no_argument, nullptr, 'V'},
                                      {nullptr, 0, nullptr, 0}};
  int c;
  while ((c = getopt_long(argc, argv, "f:vehV", long_opts, nullptr)) != -1) {
    switch (c) {
      case 'V':
        printf("registrard %s\n", REGISTRAR_VERSION);
        return 0;
      case 'f':
        config_file = optarg;
        break;
      case 'v':
        verbose++;
        break;
      case 'e':
        exit_on_expiry_flag = true;
        break;
      case 'h':
        usage(argv[0], nullptr);
        break;
      default:
        usage(argv[0], "bad option");
    }
  }
  if (config_file.empty()) usage(argv[0], "file is required");

  Logger log("registrar");
  if (const char* env = getenv("LOG_LEVEL")) {
    LogLevel l;
    if (log_level_from_name(env, &l)) log.set_level(l);
  }

  Json cfg;
  try {
    std::ifstream f(config_file);
    if (!f) throw std::runtime_error("cannot open " + config_file);
    std::stringstream ss;
    ss << f.rdbuf();
    cfg = Json::parse(ss.str());
  } catch (const std::exception& e) {
    log.fatal("unable to read configuration", {{"file", Json(config_file)}, {"err", Json(e.what())}});
    return 1;
  }

  // level: config logLevel, then -v repeats lower by one step (main.js:66-76)
  std::string lvl = cfg.get_string("logLevel", "");
  if (!lvl.empty()) {
    LogLevel l;
    if (log_level_from_name(lvl, &l)) log.set_level(l);
  }
  for (int i = 0; i < verbose; i++) log.set_level(lower_level(log.level()));

  log.info("configuration loaded", {{"file", Json(config_file)}, {"config", cfg}});

  OrchestratorConfig ocfg;
  try {
    ocfg = parse_config(cfg);
  } catch (const std::exception& e) {
    log.fatal("invalid configuration", {{"err", Json(e.what())}});
    return 1;
  }
  if (exit_on_expiry_flag) ocfg.exit_on_expiry = true;
  ocfg.zk.log_level = log.level();

  signal(SIGINT, on_signal);
  signal(SIGTERM, on_signal);
  signal(SIGUSR1, on_usr1);  // operational metrics dump
  signal(SIGPIPE, SIG_IGN);

  Orchestrator orch(ocfg, log);
  bool is_down = false;  // heartbeat-failure log latch (main.js:149,187-198)
  bool exit_on_expiry = ocfg.exit_on_expiry;
  std::atomic<int> exit_code{0};
  std::atomic<bool> want_exit{false};

  orch.set_event_callback([&](const OrchEvent& ev) {
    switch (ev.type) {
      case OrchEvent::Type::Register: {
        Json zn = Json::array();
        for (const auto& n : ev.znodes) zn.push_back(Json(n));
        log.info("registrar: registered", {{"znodes", std::move(zn)}});
        break;
      }
      case OrchEvent::Type::Unregister: {
        Json zn = Json::array();
        for (const auto& n : ev.znodes) zn.push_back(Json(n));
        log.warn("registrar: unregistered", {{"err", Json(ev.detail)}, {"znodes", std::move(zn)}});
        break;
      }
      case OrchEvent::Type::Ok:
        log.info("registrar: healthcheck ok (was down)");
        break;
      case OrchEvent::Type::Fail:
        log.error("registrar: healthcheck failed", {{"err", Json(ev.detail)}});
        break;
      case OrchEvent::Type::Error:
        log.error("registrar: unexpected error", {{"err", Json(ev.detail)}});
        break;
      case OrchEvent::Type::Heartbeat:
        if (is_down) log.info("zookeeper heartbeat ok");
        is_down = false;
        break;
      case OrchEvent::Type::HeartbeatFailure:
        if (!is_down) log.error("zookeeper: heartbeat failed", {{"err", Json(ev.detail)}});
        is_down = true;
        break;
      case OrchEvent::Type::SessionExpired:
        if (exit_on_expiry) {
          log.fatal("Zookeeper session_expired event; exiting");
          exit_code.store(1);
          want_exit.store(true);
        } else {
          log.warn("zookeeper: session expired; re-registering in-process");
        }
        break;
      case OrchEvent::Type::Stopped:
        break;
    }
  });

  orch.start();

  while (g_signal.load() == 0 && !want_exit.load()) {
    if (orch.expired()) {
      exit_code.store(1);
      break;
    }
    if (g_dump_metrics.exchange(false)) {
      OrchMetrics mtx = orch.metrics();
      int64_t p50 = 0;
      if (!mtx.recent_heartbeat_rtt_us.empty()) {
        std::vector<int64_t> v = mtx.recent_heartbeat_rtt_us;
        std::nth_element(v.begin(), v.begin() + static_cast<long>(v.size() / 2), v.end());
        p50 = v[v.size() / 2];
      }
      log.info("registrar: metrics",
               {{"registers", Json(static_cast<int64_t>(mtx.registers))},
                {"unregisters", Json(static_cast<int64_t>(mtx.unregisters))},
                {"heartbeats", Json(static_cast<int64_t>(mtx.heartbeats))},
                {"heartbeatFailures", Json(static_cast<int64_t>(mtx.heartbeat_failures))},
                {"sessionExpiries", Json(static_cast<int64_t>(mtx.session_expiries))},
                {"errors", Json(static_cast<int64_t>(mtx.errors))},
                {"p50HeartbeatRttUs", Json(p50)}});
    }
    std::this_thread::sleep_for(std::chrono::milliseconds(100));
  }

  int sig = g_signal.load();
  if (sig != 0) log.info("shutting down on signal", {{"signal", Json(static_cast<int64_t>(sig))}});
  if (want_exit.load() && exit_code.load() != 0) {
    // expiry fast-exit: skip graceful teardown; ephemerals are already gone
    // server-side (ZK semantics) and the supervisor restart re-registers
    return exit_code.load();
  }
  orch.stop();
  return exit_code.load();
}
