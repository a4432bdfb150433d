// Minimal RESP2 (Redis/Valkey wire protocol) client: blocking TCP with
// command pipelining. No external dependency — the framework talks to
// Valkey/Redis with its own codec, the same way the events plane carries
// its own ZMTP implementation.
#pragma once

#include <arpa/inet.h>
#include <netdb.h>
#include <netinet/tcp.h>
#include <sys/socket.h>
#include <unistd.h>

#include <cstring>
#include <memory>
#include <mutex>
#include <optional>
#include <stdexcept>
#include <string>
#include <variant>
#include <vector>

namespace kvc {

struct RespError : std::runtime_error {
  using std::runtime_error::runtime_error;
};

// A RESP reply: monostate = nil, string covers simple+bulk, int64, error
// text, or nested array.
struct RespReply;
using RespValue = std::variant<std::monostate, std::string, int64_t,
                               std::vector<RespReply>>;
struct RespReply {
  RespValue value;
  bool is_error = false;

  bool is_nil() const { return std::holds_alternative<std::monostate>(value); }
  const std::string& str() const { return std::get<std::string>(value); }
  int64_t integer() const { return std::get<int64_t>(value); }
  const std::vector<RespReply>& array() const {
    return std::get<std::vector<RespReply>>(value);
  }
};

class RespConnection {
 public:
  RespConnection(const std::string& host, int port, int timeout_ms = 5000) {
    struct addrinfo hints = {};
    hints.ai_family = AF_UNSPEC;
    hints.ai_socktype = SOCK_STREAM;
    struct addrinfo* res = nullptr;
    if (::getaddrinfo(host.c_str(), std::to_string(port).c_str(), &hints, &res) != 0 ||
        !res)
      throw RespError("resolve failed: " + host);
    fd_ = ::socket(res->ai_family, SOCK_STREAM, 0);
    if (fd_ < 0) {
      ::freeaddrinfo(res);
      throw RespError("socket() failed");
    }
    struct timeval tv = {timeout_ms / 1000, (timeout_ms % 1000) * 1000};
    ::setsockopt(fd_, SOL_SOCKET, SO_RCVTIMEO, &tv, sizeof(tv));
    ::setsockopt(fd_, SOL_SOCKET, SO_SNDTIMEO, &tv, sizeof(tv));
    int one = 1;
    ::setsockopt(fd_, IPPROTO_TCP, TCP_NODELAY, &one, sizeof(one));
    int rc = ::connect(fd_, res->ai_addr, res->ai_addrlen);
    ::freeaddrinfo(res);
    if (rc != 0) {
      ::close(fd_);
      fd_ = -1;
      throw RespError("connect failed: " + host + ":" + std::to_string(port));
    }
  }

  ~RespConnection() {
    if (fd_ >= 0) ::close(fd_);
  }
  RespConnection(const RespConnection&) = delete;
  RespConnection& operator=(const RespConnection&) = delete;

  // Pipeline: send all commands, then read all replies.
  std::vector<RespReply> pipeline(
      const std::vector<std::vector<std::string>>& commands) {
    std::string out;
    for (const auto& cmd : commands) encode(out, cmd);
    send_all(out);
    std::vector<RespReply> replies;
    replies.reserve(commands.size());
    for (size_t i = 0; i < commands.size(); ++i) replies.push_back(read_reply());
    return replies;
  }

  RespReply command(const std::vector<std::string>& cmd) {
    return pipeline({cmd})[0];
  }

 private:
  static void encode(std::string& out, const std::vector<std::string>& cmd) {
    out += "*" + std::to_string(cmd.size()) + "\r\n";
    for (const auto& a : cmd) {
      out += "$" + std::to_string(a.size()) + "\r\n";
      out += a;
      out += "\r\n";
    }
  }

  void send_all(const std::string& data) {
    size_t off = 0;
    while (off < data.size()) {
      ssize_t w = ::send(fd_, data.data() + off, data.size() - off, MSG_NOSIGNAL);
      if (w <= 0) throw RespError("send failed");
      off += static_cast<size_t>(w);
    }
  }

  char take() {
    if (rpos_ >= rbuf_.size()) {
      char buf[4096];
      ssize_t r = ::recv(fd_, buf, sizeof(buf), 0);
      if (r <= 0) throw RespError("recv failed / connection closed");
      rbuf_.assign(buf, buf + r);
      rpos_ = 0;
    }
    return rbuf_[rpos_++];
  }

  std::string read_line() {
    std::string line;
    for (;;) {
      char c = take();
      if (c == '\r') {
        take();  // \n
        return line;
      }
      line.push_back(c);
    }
  }

  std::string read_exact(size_t n) {
    std::string out;
    out.reserve(n);
    while (out.size() < n) out.push_back(take());
    take();
    take();  // trailing \r\n
    return out;
  }

  RespReply read_reply() {
    char t = take();
    std::string line = read_line();
    RespReply r;
    switch (t) {
      case '+':
        r.value = line;
        return r;
      case '-':
        r.value = line;
        r.is_error = true;
        return r;
      case ':':
        r.value = static_cast<int64_t>(std::stoll(line));
        return r;
      case '$': {
        long n = std::stol(line);
        if (n < 0) return r;  // nil
        r.value = read_exact(static_cast<size_t>(n));
        return r;
      }
      case '*': {
        long n = std::stol(line);
        if (n < 0) return r;  // nil array
        std::vector<RespReply> items;
        items.reserve(static_cast<size_t>(n));
        for (long i = 0; i < n; ++i) items.push_back(read_reply());
        r.value = std::move(items);
        return r;
      }
      default:
        throw RespError(std::string("unexpected RESP type byte: ") + t);
    }
  }

  int fd_ = -1;
  std::string rbuf_;
  size_t rpos_ = 0;
};

// Thread-safe pool of RESP connections.
class RespPool {
 public:
  RespPool(std::string host, int port, size_t size = 4)
      : host_(std::move(host)), port_(port), size_(size) {}

  template <typename F>
  auto with(F f) {
    std::unique_ptr<RespConnection> conn;
    {
      std::lock_guard<std::mutex> g(mu_);
      if (!idle_.empty()) {
        conn = std::move(idle_.back());
        idle_.pop_back();
      }
    }
    if (!conn) conn = std::make_unique<RespConnection>(host_, port_);
    try {
      auto result = f(*conn);
      std::lock_guard<std::mutex> g(mu_);
      if (idle_.size() < size_) idle_.push_back(std::move(conn));
      return result;
    } catch (...) {
      // broken connection: drop it, next call reconnects
      throw;
    }
  }

 private:
  std::string host_;
  int port_;
  size_t size_;
  std::mutex mu_;
  std::vector<std::unique_ptr<RespConnection>> idle_;
};

}  // namespace kvc
