// torchft_amd coordination wire protocol ("FTP/1" — fault-tolerance protocol).
//
// MI355X-native replacement for the reference's gRPC/tonic stack
// (reference: /root/reference/proto/torchft.proto, src/net.rs, src/timeout.rs).
// Instead of HTTP/2 + protobuf we use a length-prefixed binary framing over
// plain TCP: each frame is
//     u32 little-endian payload length | u8 message type | body
// Bodies are encoded with the Writer/Reader below (i64 LE, u8, length-prefixed
// strings, vectors). The message *fields* mirror torchft.proto so the
// capability surface is identical; the encoding is our own.
//
// The same port also answers plain HTTP GET/POST (dashboard /status and
// /replica/:id/kill) — the server peeks the first bytes of a connection and
// dispatches, mirroring the reference's accept_http1 dual-protocol listener.
#pragma once

#include <arpa/inet.h>
#include <netdb.h>
#include <netinet/in.h>
#include <netinet/tcp.h>
#include <poll.h>
#include <string.h>
#include <sys/socket.h>
#include <unistd.h>

#include <chrono>
#include <cstdint>
#include <optional>
#include <stdexcept>
#include <string>
#include <vector>

namespace ftcoord {

using Clock = std::chrono::steady_clock;
using TimePoint = Clock::time_point;
using Millis = std::chrono::milliseconds;

class TimeoutError : public std::runtime_error {
 public:
  explicit TimeoutError(const std::string& what) : std::runtime_error(what) {}
};

class ConnError : public std::runtime_error {
 public:
  explicit ConnError(const std::string& what) : std::runtime_error(what) {}
};

// ---------------------------------------------------------------- messages

enum MsgType : uint8_t {
  kLighthouseQuorumReq = 1,
  kLighthouseQuorumResp = 2,
  kLighthouseHeartbeatReq = 3,
  kLighthouseHeartbeatResp = 4,
  kManagerQuorumReq = 5,
  kManagerQuorumResp = 6,
  kCheckpointMetadataReq = 7,
  kCheckpointMetadataResp = 8,
  kShouldCommitReq = 9,
  kShouldCommitResp = 10,
  kKillReq = 11,
  kKillResp = 12,
  kLighthouseSubscribeReq = 13,   // subscribe to quorum broadcasts (no join)
  kError = 255,
};

enum ErrCode : uint8_t {
  kErrGeneric = 1,
  kErrTimeout = 2,
  kErrNotFound = 3,
  kErrInvalid = 4,
};

struct Writer {
  std::vector<uint8_t> buf;
  void u8(uint8_t v) { buf.push_back(v); }
  void i64(int64_t v) {
    for (int i = 0; i < 8; i++) buf.push_back((uint8_t)(((uint64_t)v >> (8 * i)) & 0xff));
  }
  void u32(uint32_t v) {
    for (int i = 0; i < 4; i++) buf.push_back((uint8_t)((v >> (8 * i)) & 0xff));
  }
  void str(const std::string& s) {
    u32((uint32_t)s.size());
    buf.insert(buf.end(), s.begin(), s.end());
  }
};

struct Reader {
  const uint8_t* p;
  size_t n;
  size_t off = 0;
  Reader(const uint8_t* data, size_t len) : p(data), n(len) {}
  void need(size_t k) const {
    if (off + k > n) throw ConnError("wire: truncated message");
  }
  uint8_t u8() {
    need(1);
    return p[off++];
  }
  uint32_t u32() {
    need(4);
    uint32_t v = 0;
    for (int i = 0; i < 4; i++) v |= (uint32_t)p[off + i] << (8 * i);
    off += 4;
    return v;
  }
  int64_t i64() {
    need(8);
    uint64_t v = 0;
    for (int i = 0; i < 8; i++) v |= (uint64_t)p[off + i] << (8 * i);
    off += 8;
    return (int64_t)v;
  }
  std::string str() {
    uint32_t len = u32();
    need(len);
    std::string s((const char*)p + off, len);
    off += len;
    return s;
  }
};

// Mirrors QuorumMember in the reference proto (torchft.proto:37-47).
struct QuorumMember {
  std::string replica_id;
  std::string address;
  std::string store_address;
  int64_t step = 0;
  int64_t world_size = 0;
  bool shrink_only = false;
  int64_t commit_failures = 0;
  std::string data;  // JSON-encoded user dict

  void encode(Writer& w) const {
    w.str(replica_id);
    w.str(address);
    w.str(store_address);
    w.i64(step);
    w.i64(world_size);
    w.u8(shrink_only ? 1 : 0);
    w.i64(commit_failures);
    w.str(data);
  }
  static QuorumMember decode(Reader& r) {
    QuorumMember m;
    m.replica_id = r.str();
    m.address = r.str();
    m.store_address = r.str();
    m.step = r.i64();
    m.world_size = r.i64();
    m.shrink_only = r.u8() != 0;
    m.commit_failures = r.i64();
    m.data = r.str();
    return m;
  }
};

struct Quorum {
  int64_t quorum_id = 0;
  std::vector<QuorumMember> participants;
  int64_t created_sec = 0;
  int64_t created_nanos = 0;

  void encode(Writer& w) const {
    w.i64(quorum_id);
    w.u32((uint32_t)participants.size());
    for (auto& m : participants) m.encode(w);
    w.i64(created_sec);
    w.i64(created_nanos);
  }
  static Quorum decode(Reader& r) {
    Quorum q;
    q.quorum_id = r.i64();
    uint32_t n = r.u32();
    q.participants.reserve(n);
    for (uint32_t i = 0; i < n; i++) q.participants.push_back(QuorumMember::decode(r));
    q.created_sec = r.i64();
    q.created_nanos = r.i64();
    return q;
  }
};

// Mirrors ManagerQuorumResponse (torchft.proto:84-100).
struct ManagerQuorumResult {
  int64_t quorum_id = 0;
  std::string recover_src_manager_address;
  std::optional<int64_t> recover_src_replica_rank;
  std::vector<int64_t> recover_dst_replica_ranks;
  std::string store_address;
  int64_t max_step = 0;
  std::optional<int64_t> max_replica_rank;
  int64_t max_world_size = 0;
  int64_t replica_rank = 0;
  int64_t replica_world_size = 0;
  bool heal = false;
  int64_t commit_failures = 0;
  std::vector<std::string> replica_ids;

  void encode(Writer& w) const {
    w.i64(quorum_id);
    w.str(recover_src_manager_address);
    w.u8(recover_src_replica_rank.has_value() ? 1 : 0);
    w.i64(recover_src_replica_rank.value_or(0));
    w.u32((uint32_t)recover_dst_replica_ranks.size());
    for (auto v : recover_dst_replica_ranks) w.i64(v);
    w.str(store_address);
    w.i64(max_step);
    w.u8(max_replica_rank.has_value() ? 1 : 0);
    w.i64(max_replica_rank.value_or(0));
    w.i64(max_world_size);
    w.i64(replica_rank);
    w.i64(replica_world_size);
    w.u8(heal ? 1 : 0);
    w.i64(commit_failures);
    w.u32((uint32_t)replica_ids.size());
    for (auto& s : replica_ids) w.str(s);
  }
  static ManagerQuorumResult decode(Reader& r) {
    ManagerQuorumResult q;
    q.quorum_id = r.i64();
    q.recover_src_manager_address = r.str();
    bool has_src = r.u8() != 0;
    int64_t src = r.i64();
    if (has_src) q.recover_src_replica_rank = src;
    uint32_t nd = r.u32();
    for (uint32_t i = 0; i < nd; i++) q.recover_dst_replica_ranks.push_back(r.i64());
    q.store_address = r.str();
    q.max_step = r.i64();
    bool has_max = r.u8() != 0;
    int64_t maxr = r.i64();
    if (has_max) q.max_replica_rank = maxr;
    q.max_world_size = r.i64();
    q.replica_rank = r.i64();
    q.replica_world_size = r.i64();
    q.heal = r.u8() != 0;
    q.commit_failures = r.i64();
    uint32_t ni = r.u32();
    for (uint32_t i = 0; i < ni; i++) q.replica_ids.push_back(r.str());
    return q;
  }
};

// ---------------------------------------------------------------- sockets

inline void set_nodelay(int fd) {
  int one = 1;
  ::setsockopt(fd, IPPROTO_TCP, TCP_NODELAY, &one, sizeof(one));
}

inline void set_keepalive(int fd) {
  // Reference keeps HTTP2 keep-alives at 60s interval / 20s timeout
  // (src/net.rs:16-28); we use TCP keepalive with similar cadence.
  int one = 1;
  ::setsockopt(fd, SOL_SOCKET, SO_KEEPALIVE, &one, sizeof(one));
  int idle = 60, intvl = 20, cnt = 3;
  ::setsockopt(fd, IPPROTO_TCP, TCP_KEEPIDLE, &idle, sizeof(idle));
  ::setsockopt(fd, IPPROTO_TCP, TCP_KEEPINTVL, &intvl, sizeof(intvl));
  ::setsockopt(fd, IPPROTO_TCP, TCP_KEEPCNT, &cnt, sizeof(cnt));
}

// Parse "host:port", "[::]:port", "http://host:port".
inline void parse_hostport(const std::string& addr_in, std::string& host, std::string& port) {
  std::string addr = addr_in;
  auto scheme = addr.find("://");
  if (scheme != std::string::npos) addr = addr.substr(scheme + 3);
  auto slash = addr.find('/');
  if (slash != std::string::npos) addr = addr.substr(0, slash);
  if (!addr.empty() && addr[0] == '[') {
    auto close = addr.find(']');
    if (close == std::string::npos) throw ConnError("bad address: " + addr_in);
    host = addr.substr(1, close - 1);
    if (close + 1 < addr.size() && addr[close + 1] == ':')
      port = addr.substr(close + 2);
    else
      throw ConnError("bad address (no port): " + addr_in);
  } else {
    auto colon = addr.rfind(':');
    if (colon == std::string::npos) throw ConnError("bad address (no port): " + addr_in);
    host = addr.substr(0, colon);
    port = addr.substr(colon + 1);
  }
}

// Bind + listen; returns fd, fills bound port.
inline int tcp_listen(const std::string& bind_addr, int& out_port) {
  std::string host, port;
  parse_hostport(bind_addr, host, port);
  struct addrinfo hints = {};
  hints.ai_family = AF_UNSPEC;
  hints.ai_socktype = SOCK_STREAM;
  hints.ai_flags = AI_PASSIVE;
  struct addrinfo* res = nullptr;
  const char* node = host.empty() ? nullptr : host.c_str();
  int rc = ::getaddrinfo(node, port.c_str(), &hints, &res);
  if (rc != 0) throw ConnError("getaddrinfo(" + bind_addr + "): " + gai_strerror(rc));
  int fd = -1;
  for (auto* ai = res; ai; ai = ai->ai_next) {
    fd = ::socket(ai->ai_family, ai->ai_socktype, ai->ai_protocol);
    if (fd < 0) continue;
    int one = 1;
    ::setsockopt(fd, SOL_SOCKET, SO_REUSEADDR, &one, sizeof(one));
    if (::bind(fd, ai->ai_addr, ai->ai_addrlen) == 0 && ::listen(fd, 256) == 0) break;
    ::close(fd);
    fd = -1;
  }
  ::freeaddrinfo(res);
  if (fd < 0) throw ConnError("failed to bind " + bind_addr);
  struct sockaddr_storage ss;
  socklen_t slen = sizeof(ss);
  ::getsockname(fd, (struct sockaddr*)&ss, &slen);
  if (ss.ss_family == AF_INET)
    out_port = ntohs(((struct sockaddr_in*)&ss)->sin_port);
  else
    out_port = ntohs(((struct sockaddr_in6*)&ss)->sin6_port);
  return fd;
}

// Connect with deadline (nonblocking connect + poll), mirroring the
// reference's backoff-wrapped connect (src/retry.rs:14-49).
inline int tcp_connect(const std::string& addr, Millis timeout) {
  std::string host, port;
  parse_hostport(addr, host, port);
  struct addrinfo hints = {};
  hints.ai_family = AF_UNSPEC;
  hints.ai_socktype = SOCK_STREAM;
  struct addrinfo* res = nullptr;
  int rc = ::getaddrinfo(host.c_str(), port.c_str(), &hints, &res);
  if (rc != 0) throw ConnError("getaddrinfo(" + addr + "): " + gai_strerror(rc));

  TimePoint deadline = Clock::now() + timeout;
  int fd = -1;
  std::string last_err = "no addresses";
  for (auto* ai = res; ai; ai = ai->ai_next) {
    fd = ::socket(ai->ai_family, ai->ai_socktype | SOCK_NONBLOCK, ai->ai_protocol);
    if (fd < 0) continue;
    rc = ::connect(fd, ai->ai_addr, ai->ai_addrlen);
    if (rc == 0) break;
    if (errno == EINPROGRESS) {
      auto left = std::chrono::duration_cast<Millis>(deadline - Clock::now()).count();
      if (left < 0) left = 0;
      struct pollfd pfd = {fd, POLLOUT, 0};
      rc = ::poll(&pfd, 1, (int)left);
      if (rc > 0) {
        int err = 0;
        socklen_t elen = sizeof(err);
        ::getsockopt(fd, SOL_SOCKET, SO_ERROR, &err, &elen);
        if (err == 0) break;
        last_err = ::strerror(err);
      } else if (rc == 0) {
        ::close(fd);
        ::freeaddrinfo(res);
        throw TimeoutError("connect to " + addr + " timed out");
      }
    } else {
      last_err = ::strerror(errno);
    }
    ::close(fd);
    fd = -1;
  }
  ::freeaddrinfo(res);
  if (fd < 0) throw ConnError("connect to " + addr + " failed: " + last_err);
  // back to blocking; rely on poll for timeouts
  int flags = 0;
  ::setsockopt(fd, SOL_SOCKET, SO_RCVTIMEO, &flags, 0);  // no-op, keep nonblocking off via fcntl below
  set_nodelay(fd);
  set_keepalive(fd);
  return fd;
}

inline void poll_wait(int fd, short events, TimePoint deadline, const char* what) {
  auto left = std::chrono::duration_cast<Millis>(deadline - Clock::now()).count();
  if (left < 0) left = 0;
  struct pollfd pfd = {fd, events, 0};
  int rc = ::poll(&pfd, 1, (int)left);
  if (rc == 0) throw TimeoutError(std::string(what) + " timed out");
  if (rc < 0) throw ConnError(std::string(what) + " poll failed: " + strerror(errno));
  if (pfd.revents & (POLLERR | POLLNVAL)) throw ConnError(std::string(what) + ": socket error");
}

inline void write_all(int fd, const uint8_t* data, size_t len, TimePoint deadline) {
  size_t off = 0;
  while (off < len) {
    ssize_t k = ::send(fd, data + off, len - off, MSG_NOSIGNAL | MSG_DONTWAIT);
    if (k > 0) {
      off += (size_t)k;
      continue;
    }
    if (k < 0 && (errno == EAGAIN || errno == EWOULDBLOCK)) {
      poll_wait(fd, POLLOUT, deadline, "send");
      continue;
    }
    if (k < 0 && errno == EINTR) continue;
    throw ConnError(std::string("send failed: ") + strerror(errno));
  }
}

inline void read_all(int fd, uint8_t* data, size_t len, TimePoint deadline) {
  size_t off = 0;
  while (off < len) {
    ssize_t k = ::recv(fd, data + off, len - off, MSG_DONTWAIT);
    if (k > 0) {
      off += (size_t)k;
      continue;
    }
    if (k == 0) throw ConnError("connection closed by peer");
    if (errno == EAGAIN || errno == EWOULDBLOCK) {
      poll_wait(fd, POLLIN, deadline, "recv");
      continue;
    }
    if (errno == EINTR) continue;
    throw ConnError(std::string("recv failed: ") + strerror(errno));
  }
}

constexpr uint32_t kMaxFrame = 64u << 20;  // 64 MiB

inline void send_frame(int fd, uint8_t type, const std::vector<uint8_t>& body, TimePoint deadline) {
  std::vector<uint8_t> hdr(5);
  uint32_t len = (uint32_t)body.size() + 1;
  for (int i = 0; i < 4; i++) hdr[i] = (uint8_t)((len >> (8 * i)) & 0xff);
  hdr[4] = type;
  write_all(fd, hdr.data(), 5, deadline);
  if (!body.empty()) write_all(fd, body.data(), body.size(), deadline);
}

// Returns (type, body). first_byte: optional already-peeked first length byte.
inline std::pair<uint8_t, std::vector<uint8_t>> recv_frame(int fd, TimePoint deadline) {
  uint8_t hdr[5];
  read_all(fd, hdr, 5, deadline);
  uint32_t len = 0;
  for (int i = 0; i < 4; i++) len |= (uint32_t)hdr[i] << (8 * i);
  if (len == 0 || len > kMaxFrame) throw ConnError("wire: bad frame length");
  uint8_t type = hdr[4];
  std::vector<uint8_t> body(len - 1);
  if (len > 1) read_all(fd, body.data(), len - 1, deadline);
  return {type, std::move(body)};
}

inline void send_error(int fd, uint8_t code, const std::string& msg, TimePoint deadline) {
  Writer w;
  w.u8(code);
  w.str(msg);
  send_frame(fd, kError, w.buf, deadline);
}

// Raise the error contained in an kError frame.
[[noreturn]] inline void throw_wire_error(Reader& r) {
  uint8_t code = r.u8();
  std::string msg = r.str();
  if (code == kErrTimeout) throw TimeoutError(msg);
  throw std::runtime_error(msg);
}

inline std::string my_hostname_or_loopback() {
  char buf[256] = {0};
  if (::gethostname(buf, sizeof(buf) - 1) != 0) return "127.0.0.1";
  // If the hostname does not resolve (common in containers), fall back to
  // loopback so single-node jobs work out of the box.
  struct addrinfo hints = {};
  hints.ai_family = AF_UNSPEC;
  hints.ai_socktype = SOCK_STREAM;
  struct addrinfo* res = nullptr;
  if (::getaddrinfo(buf, nullptr, &hints, &res) != 0) return "127.0.0.1";
  ::freeaddrinfo(res);
  return std::string(buf);
}

}  // namespace ftcoord
