// pybind11 bindings for the torchft_amd coordination core.
// API mirrors the reference's pyo3 surface (/root/reference/src/lib.rs,
// /root/reference/torchft/_torchft.pyi): ManagerServer / ManagerClient /
// LighthouseServer / LighthouseClient / QuorumResult, plus the pure
// functions quorum_compute / compute_quorum_results exposed for unit tests.
#include <pybind11/chrono.h>
#include <pybind11/pybind11.h>
#include <pybind11/stl.h>

#include "coord.h"

namespace py = pybind11;
using namespace ftcoord;

static Millis to_ms(std::chrono::duration<double> d) {
  double ms = d.count() * 1000.0;
  if (ms < 0) ms = 0;
  if (ms > 86400000.0) ms = 86400000.0;  // clamp to 24h to avoid overflow
  return Millis((int64_t)ms);
}

// Python-facing client wrappers -------------------------------------------

struct PyLighthouseClient {
  Client client;
  PyLighthouseClient(const std::string& addr, std::chrono::duration<double> connect_timeout)
      : client(addr, to_ms(connect_timeout)) {}

  Quorum quorum(const std::string& replica_id, std::chrono::duration<double> timeout,
                const std::string& address, const std::string& store_address, int64_t step,
                int64_t world_size, bool shrink_only, const std::string& data,
                int64_t commit_failures) {
    QuorumMember m;
    m.replica_id = replica_id;
    m.address = address;
    m.store_address = store_address;
    m.step = step;
    m.world_size = world_size;
    m.shrink_only = shrink_only;
    m.commit_failures = commit_failures;
    m.data = data;
    Writer w;
    m.encode(w);
    Millis ms = to_ms(timeout);
    w.i64((int64_t)ms.count());
    auto [type, body] = client.call(kLighthouseQuorumReq, w.buf, ms + Millis(2000));
    if (type != kLighthouseQuorumResp) throw std::runtime_error("unexpected response");
    Reader r(body.data(), body.size());
    return Quorum::decode(r);
  }

  void heartbeat(const std::string& replica_id, std::chrono::duration<double> timeout) {
    Writer w;
    w.str(replica_id);
    client.call(kLighthouseHeartbeatReq, w.buf, to_ms(timeout));
  }
};

struct PyManagerClient {
  Client client;
  PyManagerClient(const std::string& addr, std::chrono::duration<double> connect_timeout)
      : client(addr, to_ms(connect_timeout)) {}

  ManagerQuorumResult quorum(int64_t group_rank, int64_t step,
                             const std::string& checkpoint_metadata, bool shrink_only,
                             std::chrono::duration<double> timeout, bool init_sync,
                             int64_t commit_failures) {
    Writer w;
    w.i64(group_rank);
    w.i64(step);
    w.str(checkpoint_metadata);
    w.u8(shrink_only ? 1 : 0);
    w.u8(init_sync ? 1 : 0);
    w.i64(commit_failures);
    Millis ms = to_ms(timeout);
    w.i64((int64_t)ms.count());
    auto [type, body] = client.call(kManagerQuorumReq, w.buf, ms + Millis(2000));
    if (type != kManagerQuorumResp) throw std::runtime_error("unexpected response");
    Reader r(body.data(), body.size());
    return ManagerQuorumResult::decode(r);
  }

  std::string checkpoint_metadata(int64_t rank, std::chrono::duration<double> timeout) {
    Writer w;
    w.i64(rank);
    auto [type, body] = client.call(kCheckpointMetadataReq, w.buf, to_ms(timeout));
    if (type != kCheckpointMetadataResp) throw std::runtime_error("unexpected response");
    Reader r(body.data(), body.size());
    return r.str();
  }

  bool should_commit(int64_t group_rank, int64_t step, bool ok,
                     std::chrono::duration<double> timeout) {
    Writer w;
    w.i64(group_rank);
    w.i64(step);
    w.u8(ok ? 1 : 0);
    Millis ms = to_ms(timeout);
    w.i64((int64_t)ms.count());
    auto [type, body] = client.call(kShouldCommitReq, w.buf, ms + Millis(2000));
    if (type != kShouldCommitResp) throw std::runtime_error("unexpected response");
    Reader r(body.data(), body.size());
    return r.u8() != 0;
  }

  void kill(const std::string& msg) {
    Writer w;
    w.str(msg);
    try {
      client.call(kKillReq, w.buf, Millis(10000));
    } catch (const ConnError&) {
      // target exits without replying; connection reset is expected
    }
  }
};

// Test helper: run quorum_compute against a synthetic state ----------------

static py::tuple py_quorum_compute(
    const std::vector<std::pair<QuorumMember, int64_t>>& participants_with_age,
    const std::map<std::string, int64_t>& heartbeat_ages,
    const std::optional<std::vector<QuorumMember>>& prev_participants, int64_t prev_quorum_id,
    uint64_t min_replicas, uint64_t join_timeout_ms, uint64_t heartbeat_timeout_ms) {
  TimePoint now = Clock::now();
  LighthouseState state;
  for (auto& [m, age_ms] : participants_with_age)
    state.participants[m.replica_id] = {now - Millis(age_ms), m};
  for (auto& [rid, age_ms] : heartbeat_ages) state.heartbeats[rid] = now - Millis(age_ms);
  if (prev_participants.has_value()) {
    Quorum q;
    q.quorum_id = prev_quorum_id;
    q.participants = *prev_participants;
    state.prev_quorum = q;
  }
  LighthouseOptions opt;
  opt.min_replicas = min_replicas;
  opt.join_timeout_ms = join_timeout_ms;
  opt.heartbeat_timeout_ms = heartbeat_timeout_ms;
  auto [result, reason] = quorum_compute(now, state, opt);
  if (result.has_value()) return py::make_tuple(py::cast(*result), reason);
  return py::make_tuple(py::none(), reason);
}

PYBIND11_MODULE(_ftcore, m) {
  m.doc() = "torchft_amd coordination core (C++): lighthouse + manager services";

  py::register_exception<TimeoutError>(m, "CoordTimeoutError", PyExc_TimeoutError);
  py::register_exception<ConnError>(m, "CoordConnectionError", PyExc_ConnectionError);

  py::class_<QuorumMember>(m, "QuorumMember")
      .def(py::init([](const std::string& replica_id, const std::string& address,
                       const std::string& store_address, int64_t step, int64_t world_size,
                       bool shrink_only, int64_t commit_failures, const std::string& data) {
             QuorumMember mm;
             mm.replica_id = replica_id;
             mm.address = address;
             mm.store_address = store_address;
             mm.step = step;
             mm.world_size = world_size;
             mm.shrink_only = shrink_only;
             mm.commit_failures = commit_failures;
             mm.data = data;
             return mm;
           }),
           py::arg("replica_id"), py::arg("address") = "", py::arg("store_address") = "",
           py::arg("step") = 0, py::arg("world_size") = 1, py::arg("shrink_only") = false,
           py::arg("commit_failures") = 0, py::arg("data") = "")
      .def_readwrite("replica_id", &QuorumMember::replica_id)
      .def_readwrite("address", &QuorumMember::address)
      .def_readwrite("store_address", &QuorumMember::store_address)
      .def_readwrite("step", &QuorumMember::step)
      .def_readwrite("world_size", &QuorumMember::world_size)
      .def_readwrite("shrink_only", &QuorumMember::shrink_only)
      .def_readwrite("commit_failures", &QuorumMember::commit_failures)
      .def_readwrite("data", &QuorumMember::data)
      .def("__repr__", [](const QuorumMember& mm) {
        return "QuorumMember(replica_id='" + mm.replica_id + "', step=" + std::to_string(mm.step) +
               ")";
      });

  py::class_<Quorum>(m, "Quorum")
      .def(py::init<>())
      .def_readwrite("quorum_id", &Quorum::quorum_id)
      .def_readwrite("participants", &Quorum::participants)
      .def_readonly("created_sec", &Quorum::created_sec)
      .def_readonly("created_nanos", &Quorum::created_nanos);

  py::class_<ManagerQuorumResult>(m, "QuorumResult")
      .def(py::init<>())
      .def_readwrite("quorum_id", &ManagerQuorumResult::quorum_id)
      .def_readwrite("replica_rank", &ManagerQuorumResult::replica_rank)
      .def_readwrite("replica_world_size", &ManagerQuorumResult::replica_world_size)
      .def_readwrite("recover_src_manager_address",
                     &ManagerQuorumResult::recover_src_manager_address)
      .def_readwrite("recover_src_replica_rank", &ManagerQuorumResult::recover_src_replica_rank)
      .def_readwrite("recover_dst_replica_ranks", &ManagerQuorumResult::recover_dst_replica_ranks)
      .def_readwrite("store_address", &ManagerQuorumResult::store_address)
      .def_readwrite("max_step", &ManagerQuorumResult::max_step)
      .def_readwrite("max_replica_rank", &ManagerQuorumResult::max_replica_rank)
      .def_readwrite("max_world_size", &ManagerQuorumResult::max_world_size)
      .def_readwrite("heal", &ManagerQuorumResult::heal)
      .def_readwrite("commit_failures", &ManagerQuorumResult::commit_failures)
      .def_readwrite("replica_ids", &ManagerQuorumResult::replica_ids);

  py::class_<Lighthouse>(m, "LighthouseServer")
      .def(py::init([](const std::string& bind, uint64_t min_replicas,
                       std::optional<uint64_t> join_timeout_ms,
                       std::optional<uint64_t> quorum_tick_ms,
                       std::optional<uint64_t> heartbeat_timeout_ms) {
             LighthouseOptions opt;
             opt.min_replicas = min_replicas;
             opt.join_timeout_ms = join_timeout_ms.value_or(100);
             opt.quorum_tick_ms = quorum_tick_ms.value_or(100);
             opt.heartbeat_timeout_ms = heartbeat_timeout_ms.value_or(5000);
             return std::make_unique<Lighthouse>(bind, opt);
           }),
           py::arg("bind"), py::arg("min_replicas"), py::arg("join_timeout_ms") = py::none(),
           py::arg("quorum_tick_ms") = py::none(), py::arg("heartbeat_timeout_ms") = py::none(),
           py::call_guard<py::gil_scoped_release>())
      .def("address", &Lighthouse::address)
      .def("shutdown", &Lighthouse::shutdown, py::call_guard<py::gil_scoped_release>())
      .def("quorum_id", &Lighthouse::quorum_id, py::call_guard<py::gil_scoped_release>())
      .def("status_reason", &Lighthouse::status_reason,
           py::call_guard<py::gil_scoped_release>());

  py::class_<ManagerSrv>(m, "ManagerServer")
      .def(py::init([](const std::string& replica_id, const std::string& lighthouse_addr,
                       const std::string& hostname, const std::string& bind,
                       const std::string& store_addr, int64_t world_size,
                       std::chrono::duration<double> heartbeat_interval,
                       std::chrono::duration<double> connect_timeout, int64_t quorum_retries) {
             return std::make_unique<ManagerSrv>(replica_id, lighthouse_addr, hostname, bind,
                                                 store_addr, world_size, to_ms(heartbeat_interval),
                                                 to_ms(connect_timeout), quorum_retries);
           }),
           py::arg("replica_id"), py::arg("lighthouse_addr"), py::arg("hostname"), py::arg("bind"),
           py::arg("store_addr"), py::arg("world_size"), py::arg("heartbeat_interval"),
           py::arg("connect_timeout"), py::arg("quorum_retries") = 0,
           py::call_guard<py::gil_scoped_release>())
      .def("address", &ManagerSrv::address)
      .def("shutdown", &ManagerSrv::shutdown, py::call_guard<py::gil_scoped_release>());

  py::class_<PyLighthouseClient>(m, "LighthouseClient")
      .def(py::init<const std::string&, std::chrono::duration<double>>(), py::arg("addr"),
           py::arg("connect_timeout"), py::call_guard<py::gil_scoped_release>())
      .def("quorum", &PyLighthouseClient::quorum, py::arg("replica_id"), py::arg("timeout"),
           py::arg("address") = "", py::arg("store_address") = "", py::arg("step") = 0,
           py::arg("world_size") = 1, py::arg("shrink_only") = false, py::arg("data") = "",
           py::arg("commit_failures") = 0, py::call_guard<py::gil_scoped_release>())
      .def("heartbeat", &PyLighthouseClient::heartbeat, py::arg("replica_id"),
           py::arg("timeout") = std::chrono::duration<double>(5.0),
           py::call_guard<py::gil_scoped_release>());

  py::class_<PyManagerClient>(m, "ManagerClient")
      .def(py::init<const std::string&, std::chrono::duration<double>>(), py::arg("addr"),
           py::arg("connect_timeout"), py::call_guard<py::gil_scoped_release>())
      .def("_quorum", &PyManagerClient::quorum, py::arg("group_rank"), py::arg("step"),
           py::arg("checkpoint_metadata"), py::arg("shrink_only"), py::arg("timeout"),
           py::arg("init_sync") = true, py::arg("commit_failures") = 0,
           py::call_guard<py::gil_scoped_release>())
      .def("_checkpoint_metadata", &PyManagerClient::checkpoint_metadata, py::arg("rank"),
           py::arg("timeout"), py::call_guard<py::gil_scoped_release>())
      .def("should_commit", &PyManagerClient::should_commit, py::arg("group_rank"),
           py::arg("step"), py::arg("should_commit"), py::arg("timeout"),
           py::call_guard<py::gil_scoped_release>())
      .def("kill", &PyManagerClient::kill, py::arg("msg") = "",
           py::call_guard<py::gil_scoped_release>());

  m.def("quorum_compute", &py_quorum_compute, py::arg("participants_with_age"),
        py::arg("heartbeat_ages"), py::arg("prev_participants") = py::none(),
        py::arg("prev_quorum_id") = 0, py::arg("min_replicas") = 1,
        py::arg("join_timeout_ms") = 60000, py::arg("heartbeat_timeout_ms") = 5000,
        "Pure quorum-formation check against a synthetic lighthouse state (for tests)");

  m.def("compute_quorum_results", &compute_quorum_results, py::arg("replica_id"),
        py::arg("group_rank"), py::arg("quorum"), py::arg("init_sync") = true,
        "Pure per-rank quorum result computation (for tests)");
}
