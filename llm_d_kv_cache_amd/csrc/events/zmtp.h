// From-scratch ZMTP 3.0 PUB/SUB transport over TCP.
//
// The KVEvents plane speaks ZeroMQ wire protocol (engines publish with
// pyzmq/libzmq). This image ships no libzmq, so — like the reference's
// pure-Go zmq4 dependency (docs/architecture.md:328) — the protocol is
// implemented directly: 64-byte greeting, NULL-mechanism READY handshake,
// short/long frames, 3.0-style subscription messages (0x01/0x00 prefix)
// plus 3.1 SUBSCRIBE/CANCEL commands for newer peers.
//
// Capability parity with the reference zmqSubscriber/SubscriberManager
// (pkg/kvevents/zmq_subscriber.go, subscriber_manager.go): SUB supports
// both bind (centralized fan-in) and dial (pod-discovery fan-out) with a
// 5s reconnect loop; PUB supports bind and dial and honors per-peer
// subscription filters.
#pragma once

#include <arpa/inet.h>
#include <netdb.h>
#include <netinet/in.h>
#include <netinet/tcp.h>
#include <poll.h>
#include <sys/socket.h>
#include <unistd.h>

#include <atomic>
#include <chrono>
#include <algorithm>
#include <cstring>
#include <functional>
#include <memory>
#include <mutex>
#include <set>
#include <stdexcept>
#include <string>
#include <thread>
#include <unordered_map>
#include <vector>

#include "events.h"  // RawMessage lives in pool.h; forward-declare instead

namespace kvc {

struct ZmtpError : std::runtime_error {
  using std::runtime_error::runtime_error;
};

namespace zmtp {

// Self-reaping thread set for per-connection reader threads: a finished
// thread marks itself done and the next spawn() (or join_all()) joins and
// erases it, so a flapping peer reconnecting every few seconds does not
// grow an unjoined-handle list for the process lifetime.
class ThreadSet {
 public:
  template <class F>
  void spawn(F&& fn) {
    std::lock_guard<std::mutex> g(mu_);
    reap_finished_locked();
    const uint64_t id = next_id_++;
    threads_.emplace(id,
                     std::thread([this, id, f = std::forward<F>(fn)]() mutable {
                       f();
                       std::lock_guard<std::mutex> g2(mu_);
                       finished_.push_back(id);
                     }));
  }

  void join_all() {
    std::unordered_map<uint64_t, std::thread> taken;
    {
      std::lock_guard<std::mutex> g(mu_);
      taken.swap(threads_);
      finished_.clear();
    }
    for (auto& [id, t] : taken)
      if (t.joinable()) t.join();
  }

 private:
  void reap_finished_locked() {
    for (uint64_t id : finished_) {
      auto it = threads_.find(id);
      if (it != threads_.end()) {
        it->second.join();  // thread is past its body; join returns at once
        threads_.erase(it);
      }
    }
    finished_.clear();
  }

  std::mutex mu_;
  uint64_t next_id_ = 0;
  std::unordered_map<uint64_t, std::thread> threads_;
  std::vector<uint64_t> finished_;
};

// ---- endpoint parsing: "tcp://host:port" -----------------------------------
struct Endpoint {
  std::string host;
  uint16_t port = 0;
};

inline Endpoint parse_endpoint(const std::string& ep) {
  const std::string prefix = "tcp://";
  if (ep.rfind(prefix, 0) != 0)
    throw ZmtpError("unsupported endpoint (tcp:// only): " + ep);
  std::string rest = ep.substr(prefix.size());
  size_t colon = rest.rfind(':');
  if (colon == std::string::npos) throw ZmtpError("endpoint missing port: " + ep);
  Endpoint out;
  out.host = rest.substr(0, colon);
  if (out.host == "*") out.host = "0.0.0.0";
  out.port = static_cast<uint16_t>(std::stoi(rest.substr(colon + 1)));
  return out;
}

// ---- blocking socket helpers ------------------------------------------------
inline bool write_all(int fd, const void* buf, size_t n) {
  const char* p = static_cast<const char*>(buf);
  while (n > 0) {
    ssize_t w = ::send(fd, p, n, MSG_NOSIGNAL);
    if (w <= 0) {
      if (w < 0 && (errno == EINTR)) continue;
      return false;
    }
    p += w;
    n -= static_cast<size_t>(w);
  }
  return true;
}

inline bool read_all(int fd, void* buf, size_t n) {
  char* p = static_cast<char*>(buf);
  while (n > 0) {
    ssize_t r = ::recv(fd, p, n, 0);
    if (r <= 0) {
      if (r < 0 && errno == EINTR) continue;
      return false;
    }
    p += r;
    n -= static_cast<size_t>(r);
  }
  return true;
}

// ---- greeting + handshake ---------------------------------------------------
// Mechanisms: NULL (default) and PLAIN (RFC 24 username/password — cleartext,
// for in-cluster deployments that want more than nothing; not CURVE).
inline bool send_greeting(int fd, const char* mech = "NULL",
                          bool as_server = false) {
  uint8_t g[64] = {0};
  g[0] = 0xff;
  g[9] = 0x7f;
  g[10] = 3;  // version major
  g[11] = 0;  // version minor: 3.0 (peers downgrade; subscriptions as messages)
  std::memcpy(g + 12, mech, std::min<size_t>(20, std::strlen(mech)));
  g[32] = as_server ? 1 : 0;
  return write_all(fd, g, sizeof(g));
}

inline bool recv_greeting(int fd, std::string* mech_out = nullptr) {
  uint8_t g[64];
  if (!read_all(fd, g, 64)) return false;
  if (g[0] != 0xff || g[9] != 0x7f) return false;
  if (g[10] < 3) return false;  // ZMTP < 3 unsupported
  std::string mech(reinterpret_cast<char*>(g + 12), 20);
  mech.erase(mech.find_last_not_of('\0') + 1);
  if (mech.empty()) mech = "NULL";
  if (mech_out != nullptr) *mech_out = mech;
  return true;
}

struct Frame {
  bool more = false;
  bool command = false;
  std::string body;
};

inline bool send_frame(int fd, const void* body, size_t n, bool more, bool command = false) {
  uint8_t hdr[9];
  size_t hlen;
  uint8_t flags = (more ? 1 : 0) | (command ? 4 : 0);
  if (n <= 255) {
    hdr[0] = flags;
    hdr[1] = static_cast<uint8_t>(n);
    hlen = 2;
  } else {
    hdr[0] = flags | 2;
    uint64_t len = n;
    for (int i = 0; i < 8; ++i) hdr[1 + i] = static_cast<uint8_t>((len >> (8 * (7 - i))) & 0xff);
    hlen = 9;
  }
  return write_all(fd, hdr, hlen) && write_all(fd, body, n);
}

inline bool recv_frame(int fd, Frame& f, size_t max_size = 256ull << 20) {
  uint8_t flags;
  if (!read_all(fd, &flags, 1)) return false;
  f.more = flags & 1;
  f.command = flags & 4;
  uint64_t n = 0;
  if (flags & 2) {
    uint8_t len[8];
    if (!read_all(fd, len, 8)) return false;
    for (int i = 0; i < 8; ++i) n = (n << 8) | len[i];
  } else {
    uint8_t len;
    if (!read_all(fd, &len, 1)) return false;
    n = len;
  }
  if (n > max_size) return false;
  f.body.resize(n);
  return n == 0 || read_all(fd, f.body.data(), n);
}

inline bool send_command(int fd, const std::string& name,
                         const std::string& payload = "") {
  std::string body;
  body.push_back(static_cast<char>(name.size()));
  body += name;
  body += payload;
  return send_frame(fd, body.data(), body.size(), /*more=*/false,
                    /*command=*/true);
}

inline bool recv_command(int fd, std::string& name, std::string& payload) {
  Frame f;
  if (!recv_frame(fd, f)) return false;
  if (!f.command || f.body.empty()) return false;
  uint8_t nlen = static_cast<uint8_t>(f.body[0]);
  if (f.body.size() < 1u + nlen) return false;
  name = f.body.substr(1, nlen);
  payload = f.body.substr(1 + nlen);
  return true;
}

// ZMTP 3.1 heartbeat (RFC 37): a peer configured with heartbeats sends
// PING [2-byte TTL][context] and CLOSES the connection unless a PONG
// echoing the context comes back — reply or the whole event stream dies
// silently. Returns true when the frame was a PING it answered.
inline bool maybe_pong(int fd, const Frame& f) {
  if (!f.command || f.body.empty()) return false;
  const uint8_t nlen = static_cast<uint8_t>(f.body[0]);
  if (f.body.size() < 1u + nlen || f.body.compare(1, nlen, "PING") != 0)
    return false;
  std::string context;
  if (f.body.size() > 1u + nlen + 2) context = f.body.substr(1 + nlen + 2);
  std::string body;
  body.push_back(4);
  body += "PONG";
  body += context;
  send_frame(fd, body.data(), body.size(), /*more=*/false, /*command=*/true);
  return true;
}

// Metadata body: a Socket-Type property (READY / INITIATE payloads).
inline std::string metadata_body(const std::string& socket_type) {
  std::string body;
  body.push_back(11);
  body += "Socket-Type";
  uint32_t vlen = static_cast<uint32_t>(socket_type.size());
  for (int i = 3; i >= 0; --i)
    body.push_back(static_cast<char>((vlen >> (8 * i)) & 0xff));
  body += socket_type;
  return body;
}

// READY command with a Socket-Type property.
inline bool send_ready(int fd, const std::string& socket_type) {
  return send_command(fd, "READY", metadata_body(socket_type));
}

inline bool recv_ready(int fd) {
  std::string name, payload;
  return recv_command(fd, name, payload) && name == "READY";
}

// PLAIN handshake (RFC 24): client HELLO(user, pass) -> server WELCOME ->
// client INITIATE(metadata) -> server READY(metadata). The binding side
// is the PLAIN server.
inline bool plain_client(int fd, const std::string& user,
                         const std::string& pass,
                         const std::string& socket_type) {
  std::string hello;
  hello.push_back(static_cast<char>(user.size()));
  hello += user;
  hello.push_back(static_cast<char>(pass.size()));
  hello += pass;
  if (!send_command(fd, "HELLO", hello)) return false;
  std::string name, payload;
  if (!recv_command(fd, name, payload) || name != "WELCOME") return false;
  if (!send_command(fd, "INITIATE", metadata_body(socket_type))) return false;
  return recv_command(fd, name, payload) && name == "READY";
}

inline bool plain_server(int fd, const std::string& user,
                         const std::string& pass,
                         const std::string& socket_type) {
  std::string name, payload;
  if (!recv_command(fd, name, payload) || name != "HELLO") return false;
  if (payload.empty()) return false;
  size_t ul = static_cast<uint8_t>(payload[0]);
  if (payload.size() < 2 + ul) return false;
  size_t pl = static_cast<uint8_t>(payload[1 + ul]);
  if (payload.size() < 2 + ul + pl) return false;
  if (payload.substr(1, ul) != user || payload.substr(2 + ul, pl) != pass) {
    std::string reason = "Invalid username or password";
    send_command(fd, "ERROR",
                 std::string(1, static_cast<char>(reason.size())) + reason);
    return false;
  }
  if (!send_command(fd, "WELCOME")) return false;
  if (!recv_command(fd, name, payload) || name != "INITIATE") return false;
  return send_command(fd, "READY", metadata_body(socket_type));
}

// Full greeting + security handshake for either mechanism. `is_server`
// is the TCP role (accepted vs dialed); empty username selects NULL.
// The whole exchange runs under a receive timeout so a peer that
// connects and stalls cannot pin a handshake thread forever; the
// timeout is cleared before application traffic (long-idle SUBs are
// normal).
inline bool handshake_auth(int fd, bool is_server, const std::string& user,
                           const std::string& pass,
                           const std::string& socket_type,
                           int timeout_ms = 10000) {
  struct timeval tv = {timeout_ms / 1000, (timeout_ms % 1000) * 1000};
  ::setsockopt(fd, SOL_SOCKET, SO_RCVTIMEO, &tv, sizeof(tv));
  struct Clear {
    int fd;
    ~Clear() {
      struct timeval off = {0, 0};
      ::setsockopt(fd, SOL_SOCKET, SO_RCVTIMEO, &off, sizeof(off));
    }
  } clear{fd};
  const char* mech = user.empty() ? "NULL" : "PLAIN";
  if (!send_greeting(fd, mech, is_server && !user.empty())) return false;
  std::string peer_mech;
  if (!recv_greeting(fd, &peer_mech)) return false;
  if (peer_mech != mech) return false;  // mechanism mismatch: refuse
  if (user.empty())
    return send_ready(fd, socket_type) && recv_ready(fd);
  return is_server ? plain_server(fd, user, pass, socket_type)
                   : plain_client(fd, user, pass, socket_type);
}

inline int connect_to(const Endpoint& ep, int timeout_ms = 5000) {
  struct addrinfo hints = {};
  hints.ai_family = AF_UNSPEC;
  hints.ai_socktype = SOCK_STREAM;
  struct addrinfo* res = nullptr;
  std::string port = std::to_string(ep.port);
  if (::getaddrinfo(ep.host.c_str(), port.c_str(), &hints, &res) != 0 || !res)
    return -1;
  int fd = ::socket(res->ai_family, SOCK_STREAM, 0);
  if (fd >= 0) {
    struct timeval tv = {timeout_ms / 1000, (timeout_ms % 1000) * 1000};
    ::setsockopt(fd, SOL_SOCKET, SO_SNDTIMEO, &tv, sizeof(tv));
    int one = 1;
    ::setsockopt(fd, IPPROTO_TCP, TCP_NODELAY, &one, sizeof(one));
    if (::connect(fd, res->ai_addr, res->ai_addrlen) != 0) {
      ::close(fd);
      fd = -1;
    }
  }
  ::freeaddrinfo(res);
  return fd;
}

inline int listen_on(const Endpoint& ep) {
  int fd = ::socket(AF_INET, SOCK_STREAM, 0);
  if (fd < 0) throw ZmtpError("socket() failed");
  int one = 1;
  ::setsockopt(fd, SOL_SOCKET, SO_REUSEADDR, &one, sizeof(one));
  struct sockaddr_in addr = {};
  addr.sin_family = AF_INET;
  addr.sin_port = htons(ep.port);
  if (::inet_pton(AF_INET, ep.host.c_str(), &addr.sin_addr) != 1)
    addr.sin_addr.s_addr = INADDR_ANY;
  if (::bind(fd, reinterpret_cast<sockaddr*>(&addr), sizeof(addr)) != 0) {
    ::close(fd);
    throw ZmtpError("bind failed on " + ep.host + ":" + std::to_string(ep.port));
  }
  if (::listen(fd, 64) != 0) {
    ::close(fd);
    throw ZmtpError("listen failed");
  }
  return fd;
}

inline uint16_t bound_port(int fd) {
  struct sockaddr_in addr = {};
  socklen_t len = sizeof(addr);
  ::getsockname(fd, reinterpret_cast<sockaddr*>(&addr), &len);
  return ntohs(addr.sin_port);
}

}  // namespace zmtp

// ---- PUB socket -------------------------------------------------------------

class ZmtpPublisher {
 public:
  // bind=true: listen for SUB peers (centralized fan-in topology).
  // bind=false: dial a bound SUB (engine-pod publishing to a central host).
  // username non-empty selects the PLAIN mechanism (password checked on
  // the binding side, presented by the dialing side).
  explicit ZmtpPublisher(const std::string& endpoint, bool bind = true,
                         std::string username = "", std::string password = "")
      : ep_(zmtp::parse_endpoint(endpoint)), bind_(bind),
        user_(std::move(username)), pass_(std::move(password)) {
    if (bind_) {
      listen_fd_ = zmtp::listen_on(ep_);
      ep_.port = zmtp::bound_port(listen_fd_);  // resolve port 0
      accept_thread_ = std::thread([this] { accept_loop(); });
    } else {
      dial_thread_ = std::thread([this] { dial_loop(); });
    }
  }

  ~ZmtpPublisher() { close(); }

  uint16_t port() const { return ep_.port; }

  // 3-frame KVEvents message: [topic, 8-byte BE seq, payload].
  void publish(const std::string& topic, uint64_t seq, const std::string& payload) {
    uint8_t seqb[8];
    for (int i = 0; i < 8; ++i) seqb[i] = static_cast<uint8_t>((seq >> (8 * (7 - i))) & 0xff);
    std::lock_guard<std::mutex> g(peers_mu_);
    for (auto it = peers_.begin(); it != peers_.end();) {
      auto& peer = **it;
      bool ok = true;
      {
        std::lock_guard<std::mutex> pg(peer.mu);
        if (peer.ready && peer.subscribed(topic)) {
          ok = zmtp::send_frame(peer.fd, topic.data(), topic.size(), true) &&
               zmtp::send_frame(peer.fd, seqb, 8, true) &&
               zmtp::send_frame(peer.fd, payload.data(), payload.size(), false);
        }
      }
      if (!ok) {
        ::shutdown(peer.fd, SHUT_RDWR);
        ++it;  // reader thread reaps the connection
      } else {
        ++it;
      }
    }
  }

  size_t peer_count() {
    std::lock_guard<std::mutex> g(peers_mu_);
    return peers_.size();
  }

  void close() {
    bool expected = false;
    if (!closing_.compare_exchange_strong(expected, true)) return;
    if (listen_fd_ >= 0) ::shutdown(listen_fd_, SHUT_RDWR), ::close(listen_fd_);
    {
      std::lock_guard<std::mutex> g(peers_mu_);
      for (auto& p : peers_) ::shutdown(p->fd, SHUT_RDWR);
    }
    if (accept_thread_.joinable()) accept_thread_.join();
    if (dial_thread_.joinable()) dial_thread_.join();
    reader_threads_.join_all();
    std::lock_guard<std::mutex> g(peers_mu_);
    for (auto& p : peers_) ::close(p->fd);
    peers_.clear();
  }

 private:
  struct Peer {
    int fd;
    std::mutex mu;
    bool ready = false;
    std::set<std::string> subs;  // subscribed prefixes ("" = all)

    bool subscribed(const std::string& topic) const {
      for (const auto& s : subs)
        if (topic.rfind(s, 0) == 0) return true;
      return false;
    }
  };

  void accept_loop() {
    while (!closing_) {
      int fd = ::accept(listen_fd_, nullptr, nullptr);
      if (fd < 0) {
        if (closing_) return;
        continue;
      }
      add_peer(fd);
    }
  }

  void dial_loop() {
    while (!closing_) {
      int fd = zmtp::connect_to(ep_);
      if (fd >= 0) {
        auto* peer = add_peer_sync(fd);
        // block in the reader until the connection dies, then reconnect
        if (peer) reader(peer);
      }
      for (int i = 0; i < 50 && !closing_; ++i)
        std::this_thread::sleep_for(std::chrono::milliseconds(100));
    }
  }

  Peer* add_peer_sync(int fd) {
    auto peer = std::make_unique<Peer>();
    peer->fd = fd;
    Peer* raw = peer.get();
    {
      std::lock_guard<std::mutex> g(peers_mu_);
      peers_.push_back(std::move(peer));
    }
    if (!handshake(raw)) {
      reap(raw);
      return nullptr;
    }
    return raw;
  }

  void add_peer(int fd) {
    int one = 1;
    ::setsockopt(fd, IPPROTO_TCP, TCP_NODELAY, &one, sizeof(one));
    auto peer = std::make_unique<Peer>();
    peer->fd = fd;
    Peer* raw = peer.get();
    {
      std::lock_guard<std::mutex> g(peers_mu_);
      peers_.push_back(std::move(peer));
    }
    reader_threads_.spawn([this, raw] {
      if (handshake(raw)) reader(raw);
      reap(raw);
    });
  }

  bool handshake(Peer* peer) {
    if (!zmtp::handshake_auth(peer->fd, bind_, user_, pass_, "PUB"))
      return false;
    std::lock_guard<std::mutex> pg(peer->mu);
    peer->ready = true;
    return true;
  }

  // Reads subscription updates: 3.0-style messages (0x01 sub / 0x00 unsub)
  // and 3.1 SUBSCRIBE/CANCEL commands.
  void reader(Peer* peer) {
    zmtp::Frame f;
    while (!closing_ && zmtp::recv_frame(peer->fd, f)) {
      std::lock_guard<std::mutex> pg(peer->mu);
      if (f.command) {
        if (zmtp::maybe_pong(peer->fd, f)) continue;
        if (f.body.size() >= 1) {
          uint8_t nlen = static_cast<uint8_t>(f.body[0]);
          if (f.body.size() >= 1u + nlen) {
            std::string cmd = f.body.substr(1, nlen);
            std::string arg = f.body.substr(1 + nlen);
            if (cmd == "SUBSCRIBE") peer->subs.insert(arg);
            if (cmd == "CANCEL") peer->subs.erase(arg);
          }
        }
      } else if (!f.body.empty() && f.body[0] == 0x01) {
        peer->subs.insert(f.body.substr(1));
      } else if (!f.body.empty() && f.body[0] == 0x00) {
        peer->subs.erase(f.body.substr(1));
      } else if (f.body.empty()) {
        peer->subs.insert("");
      }
    }
  }

  void reap(Peer* peer) {
    std::lock_guard<std::mutex> g(peers_mu_);
    for (auto it = peers_.begin(); it != peers_.end(); ++it) {
      if (it->get() == peer) {
        ::close(peer->fd);
        peers_.erase(it);
        return;
      }
    }
  }

  zmtp::Endpoint ep_;
  bool bind_;
  std::string user_, pass_;
  int listen_fd_ = -1;
  std::atomic<bool> closing_{false};
  std::thread accept_thread_;
  std::thread dial_thread_;
  zmtp::ThreadSet reader_threads_;
  std::mutex peers_mu_;
  std::vector<std::unique_ptr<Peer>> peers_;
};

// ---- SUB socket -------------------------------------------------------------

class ZmtpSubscriber {
 public:
  using Handler = std::function<void(std::string topic, uint64_t seq, std::string payload)>;

  // bind=false (default): dial the publisher and reconnect every
  // reconnect_ms on failure (pod-discovery topology). bind=true: accept
  // publisher connections (centralized topology).
  ZmtpSubscriber(const std::string& endpoint, std::string topic_filter,
                 Handler handler, bool bind = false, int reconnect_ms = 5000,
                 std::string username = "", std::string password = "")
      : ep_(zmtp::parse_endpoint(endpoint)), topic_(std::move(topic_filter)),
        handler_(std::move(handler)), bind_(bind), reconnect_ms_(reconnect_ms),
        user_(std::move(username)), pass_(std::move(password)) {
    if (bind_) {
      listen_fd_ = zmtp::listen_on(ep_);
      ep_.port = zmtp::bound_port(listen_fd_);
      main_thread_ = std::thread([this] { accept_loop(); });
    } else {
      main_thread_ = std::thread([this] { dial_loop(); });
    }
  }

  ~ZmtpSubscriber() { close(); }

  uint16_t port() const { return ep_.port; }

  void close() {
    bool expected = false;
    if (!closing_.compare_exchange_strong(expected, true)) return;
    if (listen_fd_ >= 0) ::shutdown(listen_fd_, SHUT_RDWR), ::close(listen_fd_);
    {
      std::lock_guard<std::mutex> g(fds_mu_);
      for (int fd : live_fds_) ::shutdown(fd, SHUT_RDWR);
    }
    if (main_thread_.joinable()) main_thread_.join();
    reader_threads_.join_all();
  }

 private:
  void dial_loop() {
    while (!closing_) {
      int fd = zmtp::connect_to(ep_);
      if (fd >= 0) {
        track(fd, true);
        if (handshake_sub(fd)) read_messages(fd);
        track(fd, false);
        ::close(fd);
      }
      int waited = 0;
      while (waited < reconnect_ms_ && !closing_) {
        std::this_thread::sleep_for(std::chrono::milliseconds(50));
        waited += 50;
      }
    }
  }

  void accept_loop() {
    while (!closing_) {
      int fd = ::accept(listen_fd_, nullptr, nullptr);
      if (fd < 0) {
        if (closing_) return;
        continue;
      }
      track(fd, true);
      reader_threads_.spawn([this, fd] {
        if (handshake_sub(fd)) read_messages(fd);
        track(fd, false);
        ::close(fd);
      });
    }
  }

  bool handshake_sub(int fd) {
    int one = 1;
    ::setsockopt(fd, IPPROTO_TCP, TCP_NODELAY, &one, sizeof(one));
    if (!zmtp::handshake_auth(fd, bind_, user_, pass_, "SUB")) return false;
    // 3.0-style subscription message: 0x01 + prefix ("" = everything).
    std::string sub;
    sub.push_back(0x01);
    sub += topic_;
    return zmtp::send_frame(fd, sub.data(), sub.size(), false);
  }

  void read_messages(int fd) {
    std::vector<zmtp::Frame> parts;
    zmtp::Frame f;
    while (!closing_ && zmtp::recv_frame(fd, f)) {
      if (f.command) {
        zmtp::maybe_pong(fd, f);  // heartbeat peers close without a PONG
        continue;
      }
      parts.push_back(f);
      if (f.more) continue;
      deliver(parts);
      parts.clear();
    }
  }

  void deliver(const std::vector<zmtp::Frame>& parts) {
    // KVEvents wire: [topic, 8-byte BE seq, payload]. Tolerate 2-frame
    // [topic, payload] by treating seq as 0.
    if (parts.empty()) return;
    std::string topic = parts[0].body;
    uint64_t seq = 0;
    std::string payload;
    if (parts.size() >= 3 && parts[1].body.size() == 8) {
      for (int i = 0; i < 8; ++i)
        seq = (seq << 8) | static_cast<uint8_t>(parts[1].body[i]);
      payload = parts[2].body;
    } else if (parts.size() == 2) {
      payload = parts[1].body;
    } else {
      return;
    }
    if (handler_) handler_(std::move(topic), seq, std::move(payload));
  }

  void track(int fd, bool add) {
    std::lock_guard<std::mutex> g(fds_mu_);
    if (add)
      live_fds_.insert(fd);
    else
      live_fds_.erase(fd);
  }

  zmtp::Endpoint ep_;
  std::string topic_;
  Handler handler_;
  bool bind_;
  int reconnect_ms_;
  std::string user_, pass_;
  int listen_fd_ = -1;
  std::atomic<bool> closing_{false};
  std::thread main_thread_;
  zmtp::ThreadSet reader_threads_;
  std::mutex fds_mu_;
  std::set<int> live_fds_;
};

}  // namespace kvc
