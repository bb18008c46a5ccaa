// ManagerSrv: per-replica-group aggregator.
// Reference semantics: /root/reference/src/manager.rs (Rust/tokio/tonic);
// re-implemented as a blocking-thread C++ TCP server (see coord.h).
#include "coord.h"

#include <algorithm>
#include <cstdio>

namespace ftcoord {

ManagerQuorumResult compute_quorum_results(const std::string& replica_id, int64_t group_rank,
                                           const Quorum& quorum, bool init_sync) {
  std::vector<QuorumMember> participants = quorum.participants;
  std::sort(participants.begin(), participants.end(),
            [](const QuorumMember& a, const QuorumMember& b) { return a.replica_id < b.replica_id; });

  // Rank of this replica within the quorum.
  int64_t replica_rank = -1;
  for (size_t i = 0; i < participants.size(); i++) {
    if (participants[i].replica_id == replica_id) {
      replica_rank = (int64_t)i;
      break;
    }
  }
  if (replica_rank < 0)
    throw std::runtime_error("replica " + replica_id + " not participating in returned quorum");

  // Participants at max step; the primary TCPStore rotates by group_rank.
  int64_t max_step = participants[0].step;
  for (auto& p : participants) max_step = std::max(max_step, p.step);
  std::vector<const QuorumMember*> max_participants;
  for (auto& p : participants)
    if (p.step == max_step) max_participants.push_back(&p);

  std::optional<int64_t> max_replica_rank;
  for (size_t i = 0; i < max_participants.size(); i++)
    if (max_participants[i]->replica_id == replica_id) max_replica_rank = (int64_t)i;

  size_t primary_replica_rank = (size_t)(group_rank % (int64_t)max_participants.size());
  const QuorumMember* primary = max_participants[primary_replica_rank];

  // Recovery assignment: replicas behind max_step (or, at step 0 with
  // init_sync, everyone but the primary) are recover-dsts; sources are
  // assigned round-robin over up-to-date replicas offset by group_rank so
  // each rank inside a group pulls from a different source.
  bool force_recover = init_sync && max_step == 0;
  std::vector<size_t> dst_ranks;
  for (size_t i = 0; i < participants.size(); i++) {
    const auto& p = participants[i];
    if (p.step != max_step || (force_recover && primary->replica_id != p.replica_id))
      dst_ranks.push_back(i);
  }
  std::set<size_t> dst_set(dst_ranks.begin(), dst_ranks.end());
  std::vector<size_t> up_to_date;
  for (size_t i = 0; i < participants.size(); i++)
    if (!dst_set.count(i)) up_to_date.push_back(i);

  std::map<size_t, std::vector<int64_t>> assignments;
  std::optional<int64_t> recover_src_replica_rank;
  if (!up_to_date.empty()) {
    for (size_t i = 0; i < dst_ranks.size(); i++) {
      size_t idx = (i + (size_t)group_rank) % up_to_date.size();
      size_t src = up_to_date[idx];
      assignments[src].push_back((int64_t)dst_ranks[i]);
      if ((int64_t)dst_ranks[i] == replica_rank) recover_src_replica_rank = (int64_t)src;
    }
  }

  ManagerQuorumResult out;
  out.quorum_id = quorum.quorum_id;
  out.recover_src_replica_rank = recover_src_replica_rank;
  out.recover_src_manager_address =
      recover_src_replica_rank ? participants[(size_t)*recover_src_replica_rank].address : "";
  auto it = assignments.find((size_t)replica_rank);
  if (it != assignments.end()) out.recover_dst_replica_ranks = it->second;
  out.store_address = primary->store_address;
  out.max_step = max_step;
  out.max_replica_rank = max_replica_rank;
  out.max_world_size = (int64_t)max_participants.size();
  out.replica_rank = replica_rank;
  out.replica_world_size = (int64_t)participants.size();
  out.heal = recover_src_replica_rank.has_value();
  int64_t cf = 0;
  for (auto& p : participants) cf = std::max(cf, p.commit_failures);
  out.commit_failures = cf;
  for (auto& p : participants) out.replica_ids.push_back(p.replica_id);
  return out;
}

// ------------------------------------------------------------- server

ManagerSrv::ManagerSrv(std::string replica_id, std::string lighthouse_addr, std::string hostname,
                       const std::string& bind, std::string store_addr, int64_t world_size,
                       Millis heartbeat_interval, Millis connect_timeout, int64_t quorum_retries)
    : replica_id_(std::move(replica_id)),
      lighthouse_addr_(std::move(lighthouse_addr)),
      hostname_(std::move(hostname)),
      store_address_(std::move(store_addr)),
      world_size_(world_size),
      heartbeat_interval_(heartbeat_interval),
      connect_timeout_(connect_timeout),
      quorum_retries_(quorum_retries) {
  listen_fd_ = tcp_listen(bind, port_);
  if (hostname_.empty()) {
    const char* env = ::getenv("TORCHFT_AMD_HOSTNAME");
    hostname_ = env ? env : my_hostname_or_loopback();
  }
  accept_thread_ = std::thread([this] { accept_loop(); });
  heartbeat_thread_ = std::thread([this] { heartbeat_loop(); });
  quorum_worker_ = std::thread([this] { quorum_worker_loop(); });
}

ManagerSrv::~ManagerSrv() { shutdown(); }

std::string ManagerSrv::address() const {
  return "http://" + hostname_ + ":" + std::to_string(port_);
}

void ManagerSrv::shutdown() {
  bool expected = false;
  if (!stop_.compare_exchange_strong(expected, true)) return;
  if (listen_fd_ >= 0) {
    ::shutdown(listen_fd_, SHUT_RDWR);
    ::close(listen_fd_);
  }
  cv_.notify_all();
  qcv_.notify_all();
  {
    std::lock_guard<std::mutex> g(conns_mu_);
    for (int fd : conns_) ::shutdown(fd, SHUT_RDWR);
  }
  if (accept_thread_.joinable()) accept_thread_.join();
  if (heartbeat_thread_.joinable()) heartbeat_thread_.join();
  if (quorum_worker_.joinable()) quorum_worker_.join();
  for (int i = 0; i < 600 && active_conns_.load() > 0; i++) {
    std::this_thread::sleep_for(Millis(10));
  }
}

void ManagerSrv::heartbeat_loop() {
  // Periodic heartbeat to the lighthouse (reference: manager.rs:194-216).
  Client client(lighthouse_addr_, connect_timeout_);
  while (!stop_.load()) {
    try {
      Writer w;
      w.str(replica_id_);
      client.call(kLighthouseHeartbeatReq, w.buf, connect_timeout_);
    } catch (...) {
      // lighthouse may be down/restarting; Client reconnects lazily
    }
    auto deadline = Clock::now() + heartbeat_interval_;
    while (!stop_.load() && Clock::now() < deadline) std::this_thread::sleep_for(Millis(10));
  }
}

void ManagerSrv::run_quorum(QuorumMember member, Millis timeout) {
  // One lighthouse quorum request on behalf of the whole group, with
  // retries and client re-creation (reference: manager.rs:218-327).
  Writer w;
  member.encode(w);
  w.i64((int64_t)timeout.count());

  std::string last_err;
  for (int64_t attempt = 0; attempt <= std::max<int64_t>(quorum_retries_, 0); attempt++) {
    if (stop_.load()) return;
    try {
      Client client(lighthouse_addr_, connect_timeout_);
      auto [type, body] = client.call(kLighthouseQuorumReq, w.buf, timeout);
      if (type != kLighthouseQuorumResp) throw std::runtime_error("unexpected response type");
      Reader r(body.data(), body.size());
      Quorum q = Quorum::decode(r);
      std::lock_guard<std::mutex> lk(mu_);
      latest_ = std::move(q);
      quorum_error_.clear();
      quorum_seq_ += 1;
      cv_.notify_all();
      return;
    } catch (const std::exception& e) {
      last_err = e.what();
      int64_t sleep_ms =
          std::max<int64_t>(100, (int64_t)timeout.count() / std::max<int64_t>(quorum_retries_ + 1, 1));
      for (int64_t t = 0; t < sleep_ms && !stop_.load(); t += 50)
        std::this_thread::sleep_for(Millis(50));
    }
  }
  // All retries exhausted: fail the waiting ranks instead of hanging them
  // (the reference has a TODO about this hang; we broadcast the error).
  std::lock_guard<std::mutex> lk(mu_);
  quorum_error_ = "lighthouse quorum failed after retries: " + last_err;
  quorum_seq_ += 1;
  cv_.notify_all();
}

void ManagerSrv::accept_loop() {
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

void ManagerSrv::quorum_worker_loop() {
  // one long-lived runner; handle_conn drops (member, timeout) into the
  // latest-wins slot when the last rank of a round joins
  while (!stop_.load()) {
    std::pair<QuorumMember, Millis> job;
    {
      std::unique_lock<std::mutex> lk(mu_);
      qcv_.wait_for(lk, Millis(250),
                    [&] { return stop_.load() || pending_quorum_.has_value(); });
      if (stop_.load()) return;
      if (!pending_quorum_.has_value()) continue;
      job = std::move(*pending_quorum_);
      pending_quorum_.reset();
    }
    run_quorum(job.first, job.second);
  }
}

void ManagerSrv::handle_conn(int fd) {
  while (!stop_.load()) {
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
      return;
    }
    auto deadline = Clock::now() + Millis(30000);
    try {
      Reader r(body.data(), body.size());
      switch (type) {
        case kManagerQuorumReq: {
          int64_t group_rank = r.i64();
          int64_t step = r.i64();
          std::string ckpt_meta = r.str();
          bool shrink_only = r.u8() != 0;
          bool init_sync = r.u8() != 0;
          int64_t commit_failures = r.i64();
          int64_t timeout_ms = r.i64();
          TimePoint rpc_deadline = Clock::now() + Millis(timeout_ms);

          std::optional<Quorum> quorum;
          std::string err;
          {
            std::unique_lock<std::mutex> lk(mu_);
            checkpoint_metadata_[group_rank] = ckpt_meta;
            QuorumMember member;
            member.replica_id = replica_id_;
            member.address = address();
            member.store_address = store_address_;
            member.step = step;
            member.world_size = world_size_;
            member.shrink_only = shrink_only;
            member.commit_failures = commit_failures;
            participants_[group_rank] = member;
            uint64_t seen = quorum_seq_;

            if ((int64_t)participants_.size() == world_size_) {
              participants_.clear();
              pending_quorum_ = {member, Millis(timeout_ms)};
              qcv_.notify_one();
            }

            while (!stop_.load() && quorum_seq_ == seen) {
              if (cv_.wait_until(lk, rpc_deadline) == std::cv_status::timeout) break;
            }
            if (quorum_seq_ != seen) {
              if (quorum_error_.empty() && latest_.has_value())
                quorum = latest_;
              else
                err = quorum_error_.empty() ? "no quorum" : quorum_error_;
            } else {
              err = "timed out waiting for quorum";
            }
          }
          if (quorum.has_value()) {
            try {
              auto result = compute_quorum_results(replica_id_, group_rank, *quorum, init_sync);
              Writer w;
              result.encode(w);
              send_frame(fd, kManagerQuorumResp, w.buf, deadline);
            } catch (const std::exception& e) {
              send_error(fd, kErrNotFound, e.what(), deadline);
            }
          } else {
            send_error(fd, err.find("timed out") != std::string::npos ? kErrTimeout : kErrGeneric,
                       err, deadline);
          }
          break;
        }
        case kCheckpointMetadataReq: {
          int64_t rank = r.i64();
          std::string meta;
          bool found = false;
          {
            std::lock_guard<std::mutex> lk(mu_);
            auto it = checkpoint_metadata_.find(rank);
            if (it != checkpoint_metadata_.end()) {
              meta = it->second;
              found = true;
            }
          }
          if (found) {
            Writer w;
            w.str(meta);
            send_frame(fd, kCheckpointMetadataResp, w.buf, deadline);
          } else {
            send_error(fd, kErrInvalid, "rank not found", deadline);
          }
          break;
        }
        case kShouldCommitReq: {
          int64_t group_rank = r.i64();
          (void)r.i64();  // step — reserved, unchecked (matches reference TODO)
          bool ok = r.u8() != 0;
          int64_t timeout_ms = r.i64();
          TimePoint rpc_deadline = Clock::now() + Millis(timeout_ms);

          bool decision = false;
          bool timed_out = false;
          {
            std::unique_lock<std::mutex> lk(mu_);
            if (!ok) commit_failures_.insert(group_rank);
            commit_count_.insert(group_rank);
            uint64_t my_round = commit_round_;
            if ((int64_t)commit_count_.size() == world_size_) {
              // Whole group reported: decision = no failures. Barrier over
              // ranks, false if ANY rank reported failure (manager.rs:423-479).
              decision = commit_failures_.empty();
              last_decision_ = decision;
              commit_count_.clear();
              commit_failures_.clear();
              commit_round_ += 1;
              cv_.notify_all();
            } else {
              while (!stop_.load() && commit_round_ == my_round) {
                if (cv_.wait_until(lk, rpc_deadline) == std::cv_status::timeout) break;
              }
              if (commit_round_ != my_round)
                decision = last_decision_;
              else
                timed_out = true;
            }
          }
          if (timed_out) {
            send_error(fd, kErrTimeout, "should_commit timed out", deadline);
          } else {
            Writer w;
            w.u8(decision ? 1 : 0);
            send_frame(fd, kShouldCommitResp, w.buf, deadline);
          }
          break;
        }
        case kKillReq: {
          std::string msg = r.str();
          fprintf(stderr, "[torchft_amd manager %s] got kill request: %s\n", replica_id_.c_str(),
                  msg.c_str());
          fflush(stderr);
          ::_exit(1);
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

}  // namespace ftcoord
