// orchestrator.cpp — register_plus implementation (see orchestrator.hpp).
#include "orchestrator.hpp"

#include <algorithm>
#include <chrono>

#include "gpu.hpp"

namespace registrar {

const char* orch_event_name(OrchEvent::Type t) {
  switch (t) {
    case OrchEvent::Type::Register:
      return "register";
    case OrchEvent::Type::Unregister:
      return "unregister";
    case OrchEvent::Type::Ok:
      return "ok";
    case OrchEvent::Type::Fail:
      return "fail";
    case OrchEvent::Type::Error:
      return "error";
    case OrchEvent::Type::Heartbeat:
      return "heartbeat";
    case OrchEvent::Type::HeartbeatFailure:
      return "heartbeatFailure";
    case OrchEvent::Type::SessionExpired:
      return "sessionExpired";
    case OrchEvent::Type::Stopped:
      return "stopped";
  }
  return "?";
}

OrchestratorConfig parse_config(const Json& cfg) {
  if (!cfg.is_object()) throw std::runtime_error("config: must be a JSON object");
  OrchestratorConfig out;

  // zookeeper block (required: main.js:78 assert)
  const Json* zkj = cfg.find("zookeeper");
  if (!zkj || !zkj->is_object()) throw std::runtime_error("config.zookeeper: object required");
  const Json* servers = zkj->find("servers");
  if (!servers || !servers->is_array() || servers->size() == 0)
    throw std::runtime_error("config.zookeeper.servers: non-empty array required");
  for (const auto& s : servers->items()) {
    if (!s.is_object()) throw std::runtime_error("config.zookeeper.servers: objects required");
    const Json* host = s.find("host");
    const Json* port = s.find("port");
    if (!host || !host->is_string()) throw std::runtime_error("servers.host: string required");
    if (!port || !port->is_number()) throw std::runtime_error("servers.port: number required");
    out.zk.servers.push_back({host->as_string(), static_cast<int>(port->as_int())});
  }
  out.zk.session_timeout_ms = static_cast<int>(zkj->get_int("timeout", out.zk.session_timeout_ms));
  out.zk.connect_timeout_ms = static_cast<int>(zkj->get_int("connectTimeout", out.zk.connect_timeout_ms));

  // registration block + adminIp hoist (main.js:147)
  const Json* regj = cfg.find("registration");
  if (!regj || !regj->is_object()) throw std::runtime_error("config.registration: object required");
  out.registration = parse_registration(*regj);
  if (out.registration.admin_ip.empty()) out.registration.admin_ip = cfg.get_string("adminIp", "");

  // MI355X: gpuIndex attaches GPU identity + xGMI rank to the payload
  if (const Json* gi = cfg.find("gpuIndex")) {
    if (gi->is_number() && !out.registration.gpu) {
      GpuInfo info;
      info.index = static_cast<int>(gi->as_int());
      auto gpus = gpu::discover_gpus();
      if (info.index >= 0 && info.index < static_cast<int>(gpus.size())) {
        info.xgmi_rank = gpus[static_cast<size_t>(info.index)].xgmi_rank;
        info.uuid = gpus[static_cast<size_t>(info.index)].uuid;
      }
      out.registration.gpu = info;
    }
  }

  if (const Json* hc = cfg.find("healthCheck")) {
    if (!hc->is_object()) throw std::runtime_error("config.healthCheck: object required");
    HealthCheckConfig hcfg = parse_health_check(*hc);
    // "gpu-liveness" preset: substitute the SMI/sysfs probe for this
    // process's GPU (BASELINE config 3)
    if (hcfg.command == "gpu-liveness") {
      int idx = out.registration.gpu ? out.registration.gpu->index : 0;
      hcfg.command = gpu::gpu_health_command(idx);
    }
    out.health = std::move(hcfg);
  }

  out.heartbeat_interval_ms = cfg.get_int("heartbeatInterval", out.heartbeat_interval_ms);
  out.heartbeat_failure_floor_ms =
      cfg.get_int("heartbeatFailureFloor", out.heartbeat_failure_floor_ms);
  // heartbeat retry policy — read-but-unused in the reference (§2.2.5), plumbed here
  if (const Json* hb = cfg.find("heartbeat")) {
    if (hb->is_object()) {
      const Json* retry = hb->find("retry");
      if (retry && retry->is_object()) {
        out.heartbeat_retry.max_attempts = retry->get_int("maxAttempts", out.heartbeat_retry.max_attempts);
        out.heartbeat_retry.initial_delay_ms = retry->get_int("initialDelay", out.heartbeat_retry.initial_delay_ms);
        out.heartbeat_retry.max_delay_ms = retry->get_int("maxDelay", out.heartbeat_retry.max_delay_ms);
      }
    }
  }
  out.exit_on_expiry = cfg.get_bool("exitOnExpiry", false);
  out.log_level = cfg.get_string("logLevel", "");
  return out;
}

Orchestrator::Orchestrator(OrchestratorConfig cfg, Logger log)
    : cfg_(std::move(cfg)), log_(log.child("registrar")) {}

Orchestrator::~Orchestrator() {
  try {
    stop();
  } catch (...) {
  }
}

void Orchestrator::set_event_callback(EventCallback cb) { cb_ = std::move(cb); }

void Orchestrator::emit(OrchEvent ev) {
  {
    std::lock_guard<std::mutex> g(metrics_mu_);
    switch (ev.type) {
      case OrchEvent::Type::Register:
        metrics_.registers++;
        break;
      case OrchEvent::Type::Unregister:
        metrics_.unregisters++;
        break;
      case OrchEvent::Type::Heartbeat:
        metrics_.heartbeats++;
        if (metrics_.recent_heartbeat_rtt_us.size() >= 1024)
          metrics_.recent_heartbeat_rtt_us.erase(metrics_.recent_heartbeat_rtt_us.begin());
        metrics_.recent_heartbeat_rtt_us.push_back(ev.rtt_us);
        break;
      case OrchEvent::Type::HeartbeatFailure:
        metrics_.heartbeat_failures++;
        break;
      case OrchEvent::Type::SessionExpired:
        metrics_.session_expiries++;
        break;
      case OrchEvent::Type::Error:
        metrics_.errors++;
        break;
      default:
        break;
    }
  }
  {
    std::lock_guard<std::mutex> g(ev_mu_);
    ev_queue_.push_back(ev);
    if (ev.type == OrchEvent::Type::Register) registered_once_ = true;
    if (ev.type == OrchEvent::Type::Error || ev.type == OrchEvent::Type::Stopped) failed_ = true;
  }
  ev_cv_.notify_all();
  if (cb_) cb_(ev);
}

void Orchestrator::start() {
  if (running_.exchange(true)) return;
  control_ = std::thread([this] { control_loop(); });
}

bool Orchestrator::wait_registered(int64_t timeout_ms) {
  std::unique_lock<std::mutex> g(ev_mu_);
  auto pred = [this] { return registered_once_ || failed_; };
  if (timeout_ms < 0) {
    ev_cv_.wait(g, pred);
  } else if (!ev_cv_.wait_for(g, std::chrono::milliseconds(timeout_ms), pred)) {
    return false;
  }
  return registered_once_;
}

void Orchestrator::stop() {
  if (!running_.exchange(false)) return;
  wake_cv_.notify_all();
  {
    // unblock a control thread stuck in the initial-connect retry loop
    std::lock_guard<std::mutex> g(mu_);
    if (client_) client_->abort_connect();
  }
  if (health_) health_->stop();
  if (control_.joinable()) control_.join();
  std::shared_ptr<zk::ZkClient> client;
  {
    std::lock_guard<std::mutex> g(mu_);
    client = client_;
    client_.reset();
  }
  if (client) client->close();
  emit({OrchEvent::Type::Stopped, "", {}, 0});
}

// Create a client, connect, run the full register pipeline. Returns false on
// unrecoverable failure (emits 'error').
bool Orchestrator::connect_and_register(bool initial) {
  auto client = std::make_shared<zk::ZkClient>(cfg_.zk, log_);
  client->set_event_callback([this](const zk::SessionEvent& ev) {
    if (ev.type == zk::SessionEvent::Type::Expired) {
      {
        std::lock_guard<std::mutex> g(wake_mu_);
        expiry_signal_ = true;
      }
      wake_cv_.notify_all();
    }
  });
  {
    // published before wait so stop() can abort a connect stuck in its
    // (infinite, reference-parity) retry loop
    std::lock_guard<std::mutex> g(mu_);
    client_ = client;
  }
  client->start();
  if (!client->wait_connected(-1)) {
    {
      std::lock_guard<std::mutex> g(mu_);
      if (client_ == client) client_.reset();
    }
    client->close();
    if (running_.load())
      emit({OrchEvent::Type::Error, "createZKClient: unable to create ZK client", {}, 0});
    return false;
  }
  // expiry recovery while the health checker holds us DOWN: reconnect the
  // session but do NOT advertise an unhealthy node — the next ok health
  // record re-registers via on_health_record (the old code reset
  // down_=false unconditionally, briefly putting a sick node back in DNS)
  if (!initial && health_ && health_->is_down()) {
    {
      std::lock_guard<std::mutex> g(mu_);
      znodes_.clear();
      down_ = true;
    }
    log_.warn("expiry recovery: health is down; session restored, register deferred");
    return true;
  }
  RegisterResult res = register_node(*client, cfg_.registration, log_);
  if (res.rc != zk::kZOk) {
    // reference would crash here on an undefined variable (§2.2.3); we emit
    // 'error' with the real failure instead
    emit({OrchEvent::Type::Error, "registration failed: " + res.error, {}, 0});
    return false;
  }
  {
    std::lock_guard<std::mutex> g(mu_);
    znodes_ = res.znodes;
    down_ = false;
  }
  emit({OrchEvent::Type::Register, initial ? "" : "re-register", res.znodes, 0});
  return true;
}

void Orchestrator::control_loop() {
  if (!connect_and_register(true)) return;

  // health checker starts after the first successful register
  // (lib/index.js:161-162)
  if (cfg_.health) {
    health_ = std::make_unique<HealthCheck>(*cfg_.health, log_);
    health_->set_callback([this](const HealthRecord& rec) { on_health_record(rec); });
    health_->start();
  }

  heartbeat_loop();
}

int Orchestrator::heartbeat_now(int64_t* rtt_us) {
  std::shared_ptr<zk::ZkClient> client;
  std::vector<std::string> nodes;
  {
    std::lock_guard<std::mutex> g(mu_);
    client = client_;
    nodes = znodes_;
  }
  if (!client) return zk::kZConnectionLoss;
  return client->heartbeat(nodes, cfg_.heartbeat_retry, rtt_us);
}

void Orchestrator::heartbeat_loop() {
  // loop cadence: interval on success, max(interval, 60 s) after a failure
  // (lib/index.js:131-159)
  while (running_.load()) {
    bool expired_now = false;
    {
      std::lock_guard<std::mutex> g(wake_mu_);
      expired_now = expiry_signal_;
      expiry_signal_ = false;
    }
    if (expired_now) {
      emit({OrchEvent::Type::SessionExpired, "zookeeper session expired", {}, 0});
      if (cfg_.exit_on_expiry) {
        // supervisor-parity policy: surface and stop (daemon exits 1,
        // reference main.js:141-144)
        expired_flag_.store(true);
        return;
      }
      log_.warn("session expired; re-registering in-process");
      std::shared_ptr<zk::ZkClient> old;
      {
        std::lock_guard<std::mutex> g(mu_);
        old = client_;
        client_.reset();
      }
      if (old) old->close();
      // bounded retries: a transient failure (storm chaos, mid-election)
      // must not permanently kill registration (the reference gave up on
      // any register error — lib/index.js:46-51)
      bool recovered = false;
      for (int attempt = 0; attempt < 5 && running_.load(); attempt++) {
        if (connect_and_register(false)) {
          recovered = true;
          break;
        }
        std::unique_lock<std::mutex> g(wake_mu_);
        wake_cv_.wait_for(g, std::chrono::milliseconds(500LL << attempt),
                          [this] { return !running_.load(); });
      }
      if (!recovered) {
        if (running_.load()) expired_flag_.store(true);
        return;
      }
    }

    bool skip = false;
    {
      std::lock_guard<std::mutex> g(mu_);
      skip = down_;  // health-down: nothing registered to heartbeat
    }

    int64_t next_wait = cfg_.heartbeat_interval_ms;
    if (!skip) {
      int64_t rtt = 0;
      int rc = heartbeat_now(&rtt);
      std::vector<std::string> nodes;
      {
        std::lock_guard<std::mutex> g(mu_);
        nodes = znodes_;
      }
      if (rc == zk::kZOk) {
        emit({OrchEvent::Type::Heartbeat, "", nodes, rtt});
      } else {
        emit({OrchEvent::Type::HeartbeatFailure, zk::error_name(rc), nodes, 0});
        next_wait = std::max<int64_t>(cfg_.heartbeat_interval_ms, cfg_.heartbeat_failure_floor_ms);
      }
    }

    std::unique_lock<std::mutex> g(wake_mu_);
    wake_cv_.wait_for(g, std::chrono::milliseconds(next_wait),
                      [this] { return !running_.load() || expiry_signal_; });
  }
}

void Orchestrator::on_health_record(const HealthRecord& rec) {
  // the health→registration glue (lib/index.js:55-129)
  if (!running_.load()) return;
  if (!rec.ok) {
    bool transition = false;
    {
      std::lock_guard<std::mutex> g(mu_);
      if (rec.is_down && !down_) {
        down_ = true;
        transition = true;
      }
    }
    if (!transition) return;
    emit({OrchEvent::Type::Fail, rec.error, {}, 0});
    std::shared_ptr<zk::ZkClient> client;
    std::vector<std::string> nodes;
    {
      std::lock_guard<std::mutex> g(mu_);
      client = client_;
      nodes = znodes_;
    }
    if (client) {
      int rc = unregister_node(*client, nodes, log_);
      if (rc != zk::kZOk) {
        emit({OrchEvent::Type::Error, std::string("healthcheck: unregister failed: ") + zk::error_name(rc), {}, 0});
      } else {
        emit({OrchEvent::Type::Unregister, rec.error, nodes, 0});
      }
    }
    return;
  }

  // ok record: if we were down, re-register (lib/index.js:60-77)
  bool was_down;
  {
    std::lock_guard<std::mutex> g(mu_);
    was_down = down_;
  }
  if (!was_down) return;
  emit({OrchEvent::Type::Ok, "", {}, 0});
  std::shared_ptr<zk::ZkClient> client;
  {
    std::lock_guard<std::mutex> g(mu_);
    client = client_;
  }
  if (!client) return;
  RegisterResult res = register_node(*client, cfg_.registration, log_);
  if (res.rc != zk::kZOk) {
    emit({OrchEvent::Type::Error, "re-register failed: " + res.error, {}, 0});
    return;
  }
  {
    std::lock_guard<std::mutex> g(mu_);
    znodes_ = res.znodes;
    down_ = false;
  }
  emit({OrchEvent::Type::Register, "health-recovery", res.znodes, 0});
}

std::vector<std::string> Orchestrator::znodes() const {
  std::lock_guard<std::mutex> g(mu_);
  return znodes_;
}

std::vector<OrchEvent> Orchestrator::poll_events() {
  std::lock_guard<std::mutex> g(ev_mu_);
  std::vector<OrchEvent> out;
  out.swap(ev_queue_);
  return out;
}

OrchMetrics Orchestrator::metrics() const {
  std::lock_guard<std::mutex> g(metrics_mu_);
  return metrics_;
}

int64_t Orchestrator::session_id() const {
  std::lock_guard<std::mutex> g(mu_);
  return client_ ? client_->session_id() : 0;
}

}  // namespace registrar
