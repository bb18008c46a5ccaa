// Lighthouse: job-wide quorum/membership service.
// Reference semantics: /root/reference/src/lighthouse.rs (Rust/tokio/tonic);
// re-implemented here as a blocking-thread C++ TCP server (see coord.h).
#include "coord.h"

#include <algorithm>
#include <cstdio>
#include <sstream>

namespace ftcoord {

static bool quorum_changed(const std::vector<QuorumMember>& a, const std::vector<QuorumMember>& b) {
  if (a.size() != b.size()) return true;
  for (size_t i = 0; i < a.size(); i++)
    if (a[i].replica_id != b[i].replica_id) return true;
  return false;
}

std::pair<std::optional<std::vector<QuorumMember>>, std::string> quorum_compute(
    TimePoint now, const LighthouseState& state, const LighthouseOptions& opt) {
  // Health filter: replicas whose last heartbeat is younger than the timeout.
  std::set<std::string> healthy_replicas;
  for (auto& [rid, last] : state.heartbeats) {
    if (now - last < Millis(opt.heartbeat_timeout_ms)) healthy_replicas.insert(rid);
  }

  std::map<std::string, const MemberDetails*> healthy_participants;
  for (auto& [rid, details] : state.participants) {
    if (healthy_replicas.count(rid)) healthy_participants[rid] = &details;
  }

  // std::map iteration is already sorted by replica_id — the deterministic
  // ordering the protocol requires.
  std::vector<QuorumMember> candidates;
  candidates.reserve(healthy_participants.size());
  bool shrink_only = false;
  for (auto& [rid, details] : healthy_participants) {
    candidates.push_back(details->member);
    if (details->member.shrink_only) shrink_only = true;
  }

  char metabuf[160];
  snprintf(metabuf, sizeof(metabuf), "[%zu/%zu participants healthy][%zu heartbeating][shrink_only=%d]",
           healthy_participants.size(), state.participants.size(), healthy_replicas.size(),
           (int)shrink_only);
  std::string metadata(metabuf);

  if (state.prev_quorum.has_value()) {
    const auto& prev = *state.prev_quorum;
    std::set<std::string> prev_ids;
    for (auto& p : prev.participants) prev_ids.insert(p.replica_id);

    if (shrink_only) {
      // A shrink-only quorum may not admit newcomers.
      std::vector<QuorumMember> filtered;
      for (auto& c : candidates)
        if (prev_ids.count(c.replica_id)) filtered.push_back(c);
      candidates = std::move(filtered);
    }

    // Fast quorum: every previous member is still healthy and participating.
    bool is_fast = true;
    for (auto& p : prev.participants) {
      if (!healthy_participants.count(p.replica_id)) {
        is_fast = false;
        break;
      }
    }
    if (is_fast) return {candidates, "Fast quorum found! " + metadata};
  }

  if (healthy_participants.size() < opt.min_replicas) {
    return {std::nullopt, "New quorum not ready, only have " +
                              std::to_string(healthy_participants.size()) +
                              " participants, need min_replicas " +
                              std::to_string(opt.min_replicas) + " " + metadata};
  }

  // Split-brain guard: require a strict majority of currently-alive replicas.
  if (healthy_participants.size() <= healthy_replicas.size() / 2) {
    return {std::nullopt, "New quorum not ready, only have " +
                              std::to_string(healthy_participants.size()) +
                              " participants, need at least half of " +
                              std::to_string(healthy_replicas.size()) + " healthy workers " +
                              metadata};
  }

  bool all_healthy_joined = healthy_participants.size() == healthy_replicas.size();
  TimePoint first_joined = now;
  for (auto& [rid, details] : healthy_participants)
    first_joined = std::min(first_joined, details->joined);

  if (!all_healthy_joined && now - first_joined < Millis(opt.join_timeout_ms)) {
    return {std::nullopt,
            "Valid quorum with " + std::to_string(healthy_participants.size()) +
                " participants, waiting for " +
                std::to_string(healthy_replicas.size() - healthy_participants.size()) +
                " healthy but not participating stragglers due to join timeout " + metadata};
  }

  return {candidates, "Valid quorum found " + metadata};
}

// ------------------------------------------------------------- server

Lighthouse::Lighthouse(const std::string& bind, LighthouseOptions opt) : opt_(opt) {
  listen_fd_ = tcp_listen(bind, port_);
  const char* env = ::getenv("TORCHFT_AMD_HOSTNAME");
  hostname_ = env ? env : my_hostname_or_loopback();
  accept_thread_ = std::thread([this] { accept_loop(); });
  tick_thread_ = std::thread([this] { tick_loop(); });
}

Lighthouse::~Lighthouse() { shutdown(); }

std::string Lighthouse::address() const {
  return "http://" + hostname_ + ":" + std::to_string(port_);
}

int64_t Lighthouse::quorum_id() {
  std::unique_lock<std::mutex> lk(mu_);
  return state_.quorum_id;
}

std::string Lighthouse::status_reason() {
  std::unique_lock<std::mutex> lk(mu_);
  auto [_, reason] = quorum_compute(Clock::now(), state_, opt_);
  return reason;
}

void Lighthouse::shutdown() {
  bool expected = false;
  if (!stop_.compare_exchange_strong(expected, true)) return;
  if (listen_fd_ >= 0) {
    ::shutdown(listen_fd_, SHUT_RDWR);
    ::close(listen_fd_);
  }
  cv_.notify_all();
  {
    std::lock_guard<std::mutex> g(conns_mu_);
    for (int fd : conns_) ::shutdown(fd, SHUT_RDWR);
  }
  if (accept_thread_.joinable()) accept_thread_.join();
  if (tick_thread_.joinable()) tick_thread_.join();
  // detached connection threads observe stop_/closed fds and exit; wait
  // them out (bounded) so no thread touches this object after destruction
  for (int i = 0; i < 600 && active_conns_.load() > 0; i++) {
    std::this_thread::sleep_for(Millis(10));
  }
}

void Lighthouse::tick_loop() {
  while (!stop_.load()) {
    {
      std::unique_lock<std::mutex> lk(mu_);
      tick_locked(lk);
    }
    std::this_thread::sleep_for(Millis(opt_.quorum_tick_ms));
  }
}

void Lighthouse::tick_locked(std::unique_lock<std::mutex>& lk) {
  // prune long-dead heartbeat entries so a job with many replica
  // restarts (each a fresh uuid) doesn't grow state without bound
  const auto now = Clock::now();
  const auto prune_age = Millis(opt_.heartbeat_timeout_ms * 10);
  for (auto it = state_.heartbeats.begin(); it != state_.heartbeats.end();) {
    if (now - it->second > prune_age && !state_.participants.count(it->first)) {
      it = state_.heartbeats.erase(it);
    } else {
      ++it;
    }
  }
  auto [met, reason] = quorum_compute(now, state_, opt_);
  if (reason != last_reason_) {
    last_reason_ = reason;  // change-logged status, mirrors ChangeLogger
  }
  if (!met.has_value()) return;
  auto participants = std::move(*met);

  bool commit_failed = false;
  for (auto& p : participants)
    if (p.commit_failures > 0) commit_failed = true;

  // quorum_id bumps only on membership change or commit failures.
  if (!state_.prev_quorum.has_value() ||
      quorum_changed(participants, state_.prev_quorum->participants) || commit_failed) {
    state_.quorum_id += 1;
  }

  Quorum q;
  q.quorum_id = state_.quorum_id;
  q.participants = std::move(participants);
  auto now_sys = std::chrono::system_clock::now().time_since_epoch();
  q.created_sec = std::chrono::duration_cast<std::chrono::seconds>(now_sys).count();
  q.created_nanos =
      (std::chrono::duration_cast<std::chrono::nanoseconds>(now_sys).count()) % 1000000000;

  state_.prev_quorum = q;
  state_.participants.clear();
  latest_ = std::move(q);
  quorum_seq_ += 1;
  cv_.notify_all();
}

void Lighthouse::accept_loop() {
  while (!stop_.load()) {
    struct pollfd pfd = {listen_fd_, POLLIN, 0};
    int rc = ::poll(&pfd, 1, 250);
    if (stop_.load()) break;
    if (rc <= 0) continue;
    int fd = ::accept(listen_fd_, nullptr, nullptr);
    if (fd < 0) continue;
    set_nodelay(fd);
    {
      std::lock_guard<std::mutex> g(conns_mu_);
      conns_.insert(fd);
    }
    active_conns_.fetch_add(1);
    std::thread([this, fd] {
      handle_conn(fd);
      {
        std::lock_guard<std::mutex> g2(conns_mu_);
        conns_.erase(fd);
      }
      ::close(fd);
      active_conns_.fetch_sub(1);
    }).detach();
  }
}

void Lighthouse::handle_conn(int fd) {
  // Dual protocol on one port (reference: accept_http1 in lighthouse.rs):
  // HTTP requests start with ASCII method names; FTP/1 frames start with a
  // little-endian length whose low byte is almost never ASCII 'G'/'P'. Peek.
  char peek[8];
  ssize_t k = -1;
  while (!stop_.load()) {
    k = ::recv(fd, peek, sizeof(peek), MSG_PEEK | MSG_DONTWAIT);
    if (k > 0) break;
    if (k == 0) return;
    if (errno != EAGAIN && errno != EWOULDBLOCK) return;
    struct pollfd pfd = {fd, POLLIN, 0};
    ::poll(&pfd, 1, 250);
  }
  if (k >= 4 && (memcmp(peek, "GET ", 4) == 0 || memcmp(peek, "POST", 4) == 0)) {
    std::string req;
    char buf[4096];
    auto deadline = Clock::now() + Millis(5000);
    // read until end of headers
    while (req.find("\r\n\r\n") == std::string::npos && req.size() < 65536) {
      ssize_t got = ::recv(fd, buf, sizeof(buf), MSG_DONTWAIT);
      if (got > 0) {
        req.append(buf, got);
        continue;
      }
      if (got == 0) break;
      if (errno != EAGAIN && errno != EWOULDBLOCK) break;
      try {
        poll_wait(fd, POLLIN, deadline, "http read");
      } catch (...) {
        break;
      }
    }
    if (!req.empty()) handle_http(fd, req);
    return;
  }

  // FTP/1 request loop (persistent connection).
  while (!stop_.load()) {
    // idle-wait for the next frame
    struct pollfd pfd = {fd, POLLIN, 0};
    int rc = ::poll(&pfd, 1, 250);
    if (stop_.load()) return;
    if (rc == 0) continue;
    if (rc < 0 || (pfd.revents & (POLLERR | POLLHUP | POLLNVAL))) {
      if (!(pfd.revents & POLLIN)) return;
    }
    uint8_t type;
    std::vector<uint8_t> body;
    try {
      std::tie(type, body) = recv_frame(fd, Clock::now() + Millis(10000));
    } catch (...) {
      return;  // peer closed / garbage
    }
    auto deadline = Clock::now() + Millis(30000);
    try {
      Reader r(body.data(), body.size());
      switch (type) {
        case kLighthouseHeartbeatReq: {
          std::string rid = r.str();
          {
            std::lock_guard<std::mutex> lk(mu_);
            state_.heartbeats[rid] = Clock::now();
          }
          Writer w;
          send_frame(fd, kLighthouseHeartbeatResp, w.buf, deadline);
          break;
        }
        case kLighthouseQuorumReq: {
          QuorumMember requester = QuorumMember::decode(r);
          int64_t timeout_ms = r.i64();
          TimePoint rpc_deadline = Clock::now() + Millis(timeout_ms);
          Quorum result;
          bool ok = false;
          {
            std::unique_lock<std::mutex> lk(mu_);
            // implicit heartbeat + (re-)registration
            state_.heartbeats[requester.replica_id] = Clock::now();
            state_.participants[requester.replica_id] = {Clock::now(), requester};
            uint64_t seen = quorum_seq_;
            tick_locked(lk);  // proactive tick
            while (!stop_.load()) {
              if (quorum_seq_ != seen && latest_.has_value()) {
                seen = quorum_seq_;
                bool in_quorum = false;
                for (auto& p : latest_->participants)
                  if (p.replica_id == requester.replica_id) in_quorum = true;
                if (in_quorum) {
                  result = *latest_;
                  ok = true;
                  break;
                }
                // Not in this quorum (participants were cleared on broadcast):
                // re-register and keep waiting, mirroring the reference's
                // retry loop (lighthouse.rs:484-551).
                state_.heartbeats[requester.replica_id] = Clock::now();
                state_.participants[requester.replica_id] = {Clock::now(), requester};
              }
              if (Clock::now() >= rpc_deadline) break;
              // Wait in slices shorter than the heartbeat timeout and
              // refresh our own heartbeat each wake: a pending quorum
              // request IS a liveness signal. Without this, a requester
              // blocked here longer than heartbeat_timeout_ms (e.g. while
              // straggler heartbeats block the split-brain guard) goes
              // stale itself and can never be part of the quorum it is
              // waiting for. (The participant entry is NOT refreshed —
              // its `joined` time drives the join timeout.)
              TimePoint wake = Clock::now() +
                  Millis(std::max<int64_t>(opt_.heartbeat_timeout_ms / 4, 50));
              if (wake > rpc_deadline) wake = rpc_deadline;
              cv_.wait_until(lk, wake);
              state_.heartbeats[requester.replica_id] = Clock::now();
            }
          }
          if (ok) {
            Writer w;
            result.encode(w);
            send_frame(fd, kLighthouseQuorumResp, w.buf, deadline);
          } else {
            send_error(fd, kErrTimeout, "lighthouse quorum timed out", deadline);
          }
          break;
        }
        default:
          send_error(fd, kErrInvalid, "unknown message type", deadline);
      }
    } catch (const std::exception& e) {
      try {
        send_error(fd, kErrGeneric, e.what(), Clock::now() + Millis(5000));
      } catch (...) {
        return;
      }
    }
  }
}

std::string Lighthouse::render_status() {
  std::unique_lock<std::mutex> lk(mu_);
  auto [_, reason] = quorum_compute(Clock::now(), state_, opt_);
  std::ostringstream os;
  os << "<html><head><title>torchft_amd lighthouse</title></head><body>";
  os << "<h1>torchft_amd Lighthouse</h1>";
  os << "<p>quorum_id: " << state_.quorum_id << "</p>";
  os << "<p>status: " << reason << "</p>";
  int64_t max_step = -1;
  if (state_.prev_quorum.has_value()) {
    os << "<h2>Previous quorum (" << state_.prev_quorum->participants.size()
       << " participants)</h2><table border=1><tr><th>replica</th><th>address</th><th>step</th>"
          "<th>world_size</th><th>kill</th></tr>";
    for (auto& p : state_.prev_quorum->participants) {
      max_step = std::max(max_step, p.step);
      os << "<tr><td>" << p.replica_id << "</td><td>" << p.address << "</td><td>" << p.step
         << "</td><td>" << p.world_size << "</td><td><form method=post action=\"/replica/"
         << p.replica_id << "/kill\"><button>kill</button></form></td></tr>";
    }
    os << "</table>";
  }
  os << "<p>max_step: " << max_step << "</p><h2>Heartbeats</h2><table border=1>"
     << "<tr><th>replica</th><th>age_ms</th></tr>";
  auto now = Clock::now();
  for (auto& [rid, t] : state_.heartbeats) {
    auto age = std::chrono::duration_cast<Millis>(now - t).count();
    os << "<tr><td>" << rid << "</td><td>" << age << "</td></tr>";
  }
  os << "</table></body></html>";
  return os.str();
}

void Lighthouse::handle_http(int fd, const std::string& request) {
  auto line_end = request.find("\r\n");
  std::string line = request.substr(0, line_end);
  auto deadline = Clock::now() + Millis(5000);
  auto respond = [&](int code, const std::string& status, const std::string& body) {
    std::ostringstream os;
    os << "HTTP/1.1 " << code << " " << status << "\r\nContent-Type: text/html\r\nContent-Length: "
       << body.size() << "\r\nConnection: close\r\n\r\n" << body;
    auto s = os.str();
    try {
      write_all(fd, (const uint8_t*)s.data(), s.size(), deadline);
    } catch (...) {
    }
  };

  if (line.rfind("GET / ", 0) == 0 || line.rfind("GET /status", 0) == 0) {
    respond(200, "OK", render_status());
    return;
  }
  // POST /replica/{id}/kill — forward a kill RPC to that replica's manager.
  if (line.rfind("POST /replica/", 0) == 0) {
    auto path = line.substr(5);  // strip "POST "
    auto sp = path.find(' ');
    if (sp != std::string::npos) path = path.substr(0, sp);
    // path = /replica/{id}/kill
    std::string prefix = "/replica/";
    std::string suffix = "/kill";
    if (path.size() > prefix.size() + suffix.size() &&
        path.compare(path.size() - suffix.size(), suffix.size(), suffix) == 0) {
      std::string rid = path.substr(prefix.size(), path.size() - prefix.size() - suffix.size());
      std::string addr;
      {
        std::unique_lock<std::mutex> lk(mu_);
        if (state_.prev_quorum.has_value()) {
          for (auto& p : state_.prev_quorum->participants)
            if (p.replica_id == rid) addr = p.address;
        }
      }
      if (addr.empty()) {
        respond(500, "Internal Server Error", "failed to find replica");
        return;
      }
      try {
        Client c(addr, Millis(10000));
        Writer w;
        w.str("killed from dashboard");
        c.call(kKillReq, w.buf, Millis(10000));
      } catch (const std::exception& e) {
        // Managers exit(1) on kill without replying; treat conn reset as OK.
      }
      respond(200, "OK", "ok");
      return;
    }
  }
  respond(404, "Not Found", "not found");
}

// ------------------------------------------------------------- client

std::pair<uint8_t, std::vector<uint8_t>> Client::call(uint8_t type,
                                                      const std::vector<uint8_t>& body,
                                                      Millis timeout) {
  std::lock_guard<std::mutex> g(mu_);
  TimePoint deadline = Clock::now() + timeout;
  for (int attempt = 0; attempt < 2; attempt++) {
    if (fd_ < 0) fd_ = tcp_connect(addr_, connect_timeout_);
    try {
      send_frame(fd_, type, body, deadline);
      auto resp = recv_frame(fd_, deadline);
      if (resp.first == kError) {
        Reader r(resp.second.data(), resp.second.size());
        throw_wire_error(r);
      }
      return resp;
    } catch (const ConnError&) {
      // stale connection (peer restarted) — retry once with a fresh one
      ::close(fd_);
      fd_ = -1;
      if (attempt == 1) throw;
    } catch (const TimeoutError&) {
      // leave the connection in an indeterminate state; drop it
      ::close(fd_);
      fd_ = -1;
      throw;
    }
  }
  throw ConnError("unreachable");
}

void Client::close() {
  std::lock_guard<std::mutex> g(mu_);
  if (fd_ >= 0) {
    ::close(fd_);
    fd_ = -1;
  }
}

}  // namespace ftcoord
