// torchft_amd coordination services: Lighthouse (job-wide quorum) and
// Manager (per-replica-group aggregator).
//
// Native C++ re-implementation of the reference's Rust control plane
// (reference: /root/reference/src/lighthouse.rs, /root/reference/src/manager.rs).
// Same protocol semantics — heartbeat-filtered membership, fast quorum,
// min_replicas, majority-of-alive split-brain guard, join timeout,
// shrink_only, quorum_id bump rules, round-robin recovery assignment,
// should_commit all-ranks barrier — implemented as blocking-thread servers
// over the FTP/1 framing in wire.h instead of tokio/tonic/gRPC.
#pragma once

#include <atomic>
#include <condition_variable>
#include <map>
#include <memory>
#include <mutex>
#include <set>
#include <thread>
#include <unordered_map>
#include <unordered_set>

#include "wire.h"

namespace ftcoord {

// ------------------------------------------------------------- pure logic

struct LighthouseOptions {
  uint64_t min_replicas = 1;
  uint64_t join_timeout_ms = 60000;
  uint64_t quorum_tick_ms = 100;
  uint64_t heartbeat_timeout_ms = 5000;
};

struct MemberDetails {
  TimePoint joined;
  QuorumMember member;
};

struct LighthouseState {
  std::map<std::string, MemberDetails> participants;
  std::map<std::string, TimePoint> heartbeats;
  std::optional<Quorum> prev_quorum;
  int64_t quorum_id = 0;
};

// Pure quorum-formation check (reference semantics: src/lighthouse.rs:141-269).
// Returns (participants or nullopt, human-readable reason).
std::pair<std::optional<std::vector<QuorumMember>>, std::string> quorum_compute(
    TimePoint now, const LighthouseState& state, const LighthouseOptions& opt);

// Pure per-rank quorum result computation (reference: src/manager.rs:489-625).
// Throws std::runtime_error if replica_id is not in the quorum.
ManagerQuorumResult compute_quorum_results(const std::string& replica_id,
                                           int64_t group_rank, const Quorum& quorum,
                                           bool init_sync);

// ------------------------------------------------------------- client

// Thread-safe blocking RPC client with lazy (re)connect.
class Client {
 public:
  Client(std::string addr, Millis connect_timeout)
      : addr_(std::move(addr)), connect_timeout_(connect_timeout) {}
  ~Client() { close(); }

  // Send one frame, wait for the reply frame. Reconnects when the cached
  // connection is dead. Throws TimeoutError / ConnError / runtime_error.
  std::pair<uint8_t, std::vector<uint8_t>> call(uint8_t type, const std::vector<uint8_t>& body,
                                                Millis timeout);
  void close();
  const std::string& addr() const { return addr_; }

 private:
  std::string addr_;
  Millis connect_timeout_;
  std::mutex mu_;
  int fd_ = -1;
};

// ------------------------------------------------------------- lighthouse

class Lighthouse {
 public:
  Lighthouse(const std::string& bind, LighthouseOptions opt);
  ~Lighthouse();

  std::string address() const;
  void shutdown();

  // exposed for tests / dashboard
  std::string status_reason();
  int64_t quorum_id();

 private:
  void accept_loop();
  void tick_loop();
  void handle_conn(int fd);
  void handle_http(int fd, const std::string& request);
  void tick_locked(std::unique_lock<std::mutex>& lk);
  std::string render_status();

  LighthouseOptions opt_;
  int listen_fd_ = -1;
  int port_ = 0;
  std::string hostname_;

  std::mutex mu_;
  std::condition_variable cv_;
  LighthouseState state_;
  uint64_t quorum_seq_ = 0;   // bumped on every broadcast
  std::optional<Quorum> latest_;
  std::string last_reason_;

  std::atomic<bool> stop_{false};
  std::thread accept_thread_;
  std::thread tick_thread_;
  std::mutex conns_mu_;
  std::set<int> conns_;
  // connection threads are detached; this counts the live ones so
  // shutdown() can wait them out (a per-thread vector would grow
  // unreaped over a long-running job)
  std::atomic<int> active_conns_{0};
};

// ------------------------------------------------------------- manager

class ManagerSrv {
 public:
  ManagerSrv(std::string replica_id, std::string lighthouse_addr, std::string hostname,
             const std::string& bind, std::string store_addr, int64_t world_size,
             Millis heartbeat_interval, Millis connect_timeout, int64_t quorum_retries);
  ~ManagerSrv();

  std::string address() const;
  void shutdown();

 private:
  void accept_loop();
  void heartbeat_loop();
  void handle_conn(int fd);
  void quorum_worker_loop();
  void run_quorum(QuorumMember member, Millis timeout);

  std::string replica_id_;
  std::string lighthouse_addr_;
  std::string hostname_;
  std::string store_address_;
  int64_t world_size_;
  Millis heartbeat_interval_;
  Millis connect_timeout_;
  int64_t quorum_retries_;

  int listen_fd_ = -1;
  int port_ = 0;

  std::mutex mu_;
  std::condition_variable cv_;
  std::unordered_map<int64_t, std::string> checkpoint_metadata_;
  std::unordered_map<int64_t, QuorumMember> participants_;
  uint64_t quorum_seq_ = 0;
  std::optional<Quorum> latest_;
  std::string quorum_error_;  // nonempty => last quorum round failed

  uint64_t commit_round_ = 0;
  bool last_decision_ = false;
  std::unordered_set<int64_t> commit_count_;
  std::unordered_set<int64_t> commit_failures_;

  std::atomic<bool> stop_{false};
  std::thread accept_thread_;
  std::thread heartbeat_thread_;
  std::mutex conns_mu_;
  std::set<int> conns_;
  std::atomic<int> active_conns_{0};

  // single long-lived quorum runner (latest-wins trigger slot): one
  // std::thread per quorum round would accumulate unreaped over the job
  std::thread quorum_worker_;
  std::condition_variable qcv_;
  std::optional<std::pair<QuorumMember, Millis>> pending_quorum_;
};

}  // namespace ftcoord
