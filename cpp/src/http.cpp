#include "cpilot/http.hpp"

#include <sys/epoll.h>

#include <arpa/inet.h>
#include <fcntl.h>
#include <netdb.h>
#include <netinet/in.h>
#include <netinet/tcp.h>
#include <poll.h>
#include <openssl/err.h>
#include <openssl/ssl.h>
#include <sys/socket.h>
#include <sys/un.h>
#include <unistd.h>

#include <cctype>
#include <cstring>

#include "cpilot/log.hpp"

namespace cpilot {
namespace http {

const char* statusText(int code) {
  switch (code) {
    case 200: return "OK";
    case 400: return "Bad Request";
    case 404: return "Not Found";
    case 405: return "Method Not Allowed";
    case 411: return "Length Required";
    case 413: return "Payload Too Large";
    case 422: return "Unprocessable Entity";
    case 431: return "Request Header Fields Too Large";
    case 500: return "Internal Server Error";
    default: return "";
  }
}

// hardening limits: the reference gets equivalent protection from
// net/http's defaults (1 MiB header cap, read deadlines)
static constexpr size_t kMaxHeaderBytes = 64 * 1024;
static constexpr size_t kMaxBodyBytes = 4 * 1024 * 1024;
static constexpr int kConnDeadlineSecs = 10;

struct Server::Conn {
  int fd = -1;
  std::string inbuf;
  bool headersDone = false;
  size_t contentLength = 0;
  size_t headerEnd = 0;
  Request req;
  // async response write state (flushed via EPOLLOUT, never by
  // blocking the reactor)
  std::string outbuf;
  size_t outoff = 0;
  bool writing = false;
  uint64_t deadlineTimer = 0;
};

Server::Server(Loop& loop, Handler handler)
    : loop_(loop), handler_(std::move(handler)) {}

Server::~Server() { stop(); }

static void setNonblock(int fd) {
  fcntl(fd, F_SETFL, fcntl(fd, F_GETFL, 0) | O_NONBLOCK);
  fcntl(fd, F_SETFD, FD_CLOEXEC);
}

bool Server::listenUnix(const std::string& path, std::string* err) {
  int fd = socket(AF_UNIX, SOCK_STREAM | SOCK_CLOEXEC, 0);
  if (fd < 0) {
    *err = strerror(errno);
    return false;
  }
  struct sockaddr_un addr;
  memset(&addr, 0, sizeof(addr));
  addr.sun_family = AF_UNIX;
  strncpy(addr.sun_path, path.c_str(), sizeof(addr.sun_path) - 1);
  if (bind(fd, (struct sockaddr*)&addr, sizeof(addr)) != 0 ||
      listen(fd, 128) != 0) {
    *err = strerror(errno);
    close(fd);
    return false;
  }
  setNonblock(fd);
  listenFd_ = fd;
  loop_.watchFd(fd, EPOLLIN, [this](uint32_t) { acceptReady(); });
  return true;
}

bool Server::listenTcp(const std::string& ip, int port, std::string* err) {
  int fd = socket(AF_INET, SOCK_STREAM | SOCK_CLOEXEC, 0);
  if (fd < 0) {
    *err = strerror(errno);
    return false;
  }
  int one = 1;
  setsockopt(fd, SOL_SOCKET, SO_REUSEADDR, &one, sizeof(one));
  struct sockaddr_in addr;
  memset(&addr, 0, sizeof(addr));
  addr.sin_family = AF_INET;
  addr.sin_port = htons(port);
  if (ip.empty() || ip == "0.0.0.0") {
    addr.sin_addr.s_addr = INADDR_ANY;
  } else if (inet_pton(AF_INET, ip.c_str(), &addr.sin_addr) != 1) {
    addr.sin_addr.s_addr = INADDR_ANY;
  }
  if (bind(fd, (struct sockaddr*)&addr, sizeof(addr)) != 0 ||
      listen(fd, 128) != 0) {
    *err = strerror(errno);
    close(fd);
    return false;
  }
  setNonblock(fd);
  listenFd_ = fd;
  loop_.watchFd(fd, EPOLLIN, [this](uint32_t) { acceptReady(); });
  return true;
}

void Server::stop() {
  if (listenFd_ >= 0) {
    loop_.unwatchFd(listenFd_);
    close(listenFd_);
    listenFd_ = -1;
  }
  for (auto& kv : conns_) {
    // cancel deadline timers: their callbacks capture this server
    if (kv.second->deadlineTimer) {
      loop_.cancelTimer(kv.second->deadlineTimer);
      kv.second->deadlineTimer = 0;
    }
    loop_.unwatchFd(kv.first);
    close(kv.first);
    kv.second->fd = -1;
  }
  conns_.clear();
}

void Server::acceptReady() {
  while (true) {
    int fd = accept(listenFd_, nullptr, nullptr);
    if (fd < 0) return;
    setNonblock(fd);
    auto c = std::make_shared<Conn>();
    c->fd = fd;
    conns_[fd] = c;
    // hard per-connection deadline: a client that dribbles its request
    // (slow-loris) or refuses to read the response is dropped, freeing
    // the fd and buffer; reset once when the response starts
    c->deadlineTimer = loop_.addTimeout(
        std::chrono::seconds(kConnDeadlineSecs),
        [this, c] {
          c->deadlineTimer = 0;
          closeConn(c);
        });
    loop_.watchFd(fd, EPOLLIN | EPOLLHUP, [this, c](uint32_t events) {
      if (c->writing)
        connWritable(c);
      else
        connReadable(c);
      (void)events;
    });
  }
}

void Server::closeConn(const std::shared_ptr<Conn>& c) {
  if (c->fd < 0) return;
  if (c->deadlineTimer) {
    loop_.cancelTimer(c->deadlineTimer);
    c->deadlineTimer = 0;
  }
  loop_.unwatchFd(c->fd);
  close(c->fd);
  conns_.erase(c->fd);
  c->fd = -1;
}

// Serialize the response and flush as much as the socket accepts now;
// the remainder drains via EPOLLOUT so a slow reader never blocks the
// reactor (it is dropped at the connection deadline instead).
void Server::beginWrite(const std::shared_ptr<Conn>& c,
                        const Response& resp) {
  std::string out = "HTTP/1.1 " + std::to_string(resp.status) + " " +
                    statusText(resp.status) + "\r\n";
  out += "Content-Type: " + resp.contentType + "\r\n";
  out += "Content-Length: " + std::to_string(resp.body.size()) + "\r\n";
  out += "Connection: close\r\n\r\n";
  out += resp.body;
  c->outbuf = std::move(out);
  c->outoff = 0;
  c->writing = true;
  // fresh deadline for the write phase
  if (c->deadlineTimer) loop_.cancelTimer(c->deadlineTimer);
  c->deadlineTimer = loop_.addTimeout(
      std::chrono::seconds(kConnDeadlineSecs),
      [this, c] {
        c->deadlineTimer = 0;
        closeConn(c);
      });
  connWritable(c);
}

void Server::connWritable(std::shared_ptr<Conn> c) {
  if (c->fd < 0) return;
  while (c->outoff < c->outbuf.size()) {
    ssize_t n =
        write(c->fd, c->outbuf.data() + c->outoff, c->outbuf.size() - c->outoff);
    if (n > 0) {
      c->outoff += n;
    } else if (n < 0 && errno == EINTR) {
      continue;
    } else if (n < 0 && errno == EAGAIN) {
      loop_.modifyFd(c->fd, EPOLLOUT | EPOLLHUP);
      return;  // resume when the socket drains
    } else {
      break;  // peer gone
    }
  }
  closeConn(c);
}

static bool parseHeaders(Server::Conn* c);

void Server::connReadable(std::shared_ptr<Conn> c) {
  char buf[8192];
  while (true) {
    ssize_t n = read(c->fd, buf, sizeof(buf));
    if (n > 0) {
      c->inbuf.append(buf, n);
      if (c->inbuf.size() > kMaxHeaderBytes + kMaxBodyBytes) {
        closeConn(c);
        return;
      }
      continue;
    }
    if (n < 0 && errno == EAGAIN) break;
    if (n < 0 && errno == EINTR) continue;
    // EOF or error before a complete request
    if (!c->headersDone || c->inbuf.size() < c->headerEnd + c->contentLength) {
      closeConn(c);
      return;
    }
    break;
  }

  if (!c->headersDone) {
    size_t end = c->inbuf.find("\r\n\r\n");
    if (end == std::string::npos) {
      if (c->inbuf.size() > kMaxHeaderBytes) {
        beginWrite(c, errorResponse(431));
      }
      return;  // wait for more (bounded by the connection deadline)
    }
    c->headerEnd = end + 4;
    if (c->headerEnd > kMaxHeaderBytes) {
      beginWrite(c, errorResponse(431));
      return;
    }
    if (!parseHeaders(c.get())) {
      closeConn(c);
      return;
    }
    c->headersDone = true;
    // no chunked request support: require a length (net/http would
    // dechunk; our endpoints only ever see small bodies)
    if (c->req.headers.count("transfer-encoding")) {
      beginWrite(c, errorResponse(411));
      return;
    }
    if (c->contentLength > kMaxBodyBytes) {
      beginWrite(c, errorResponse(413));
      return;
    }
  }
  if (c->inbuf.size() < c->headerEnd + c->contentLength) return;
  c->req.body = c->inbuf.substr(c->headerEnd, c->contentLength);

  beginWrite(c, handler_(c->req));
}

Response Server::errorResponse(int status) {
  Response resp;
  resp.status = status;
  resp.body = std::string(statusText(status)) + "\n";
  return resp;
}

static bool parseHeaders(Server::Conn* c) {
  const std::string& b = c->inbuf;
  size_t lineEnd = b.find("\r\n");
  if (lineEnd == std::string::npos) return false;
  std::string reqLine = b.substr(0, lineEnd);
  size_t sp1 = reqLine.find(' ');
  size_t sp2 = reqLine.rfind(' ');
  if (sp1 == std::string::npos || sp2 == sp1) return false;
  c->req.method = reqLine.substr(0, sp1);
  std::string target = reqLine.substr(sp1 + 1, sp2 - sp1 - 1);
  size_t q = target.find('?');
  if (q != std::string::npos) {
    c->req.path = target.substr(0, q);
    c->req.query = target.substr(q + 1);
  } else {
    c->req.path = target;
  }
  size_t pos = lineEnd + 2;
  while (pos < c->headerEnd - 2) {
    size_t eol = b.find("\r\n", pos);
    if (eol == std::string::npos || eol >= c->headerEnd - 2) break;
    std::string line = b.substr(pos, eol - pos);
    pos = eol + 2;
    size_t colon = line.find(':');
    if (colon == std::string::npos) continue;
    std::string key = line.substr(0, colon);
    for (auto& ch : key) ch = tolower((unsigned char)ch);
    size_t vstart = colon + 1;
    while (vstart < line.size() && line[vstart] == ' ') vstart++;
    c->req.headers[key] = line.substr(vstart);
  }
  auto it = c->req.headers.find("content-length");
  c->contentLength = (it != c->req.headers.end())
                         ? (size_t)atoll(it->second.c_str())
                         : 0;
  return true;
}

// ---------------- blocking client ----------------

void CancelToken::arm(int fd) {
  std::lock_guard<std::mutex> l(mu_);
  fd_ = fd;
  if (cancelled_) ::shutdown(fd, SHUT_RDWR);
}

void CancelToken::disarm() {
  std::lock_guard<std::mutex> l(mu_);
  fd_ = -1;
}

void CancelToken::cancel() {
  std::lock_guard<std::mutex> l(mu_);
  cancelled_ = true;
  if (fd_ >= 0) ::shutdown(fd_, SHUT_RDWR);
}

bool CancelToken::cancelled() {
  std::lock_guard<std::mutex> l(mu_);
  return cancelled_;
}

namespace {

int connectTarget(const std::string& target, int timeoutMs, std::string* err) {
  int fd = -1;
  if (target.rfind("unix:", 0) == 0) {
    fd = socket(AF_UNIX, SOCK_STREAM | SOCK_CLOEXEC, 0);
    if (fd < 0) {
      *err = strerror(errno);
      return -1;
    }
    struct sockaddr_un addr;
    memset(&addr, 0, sizeof(addr));
    addr.sun_family = AF_UNIX;
    strncpy(addr.sun_path, target.c_str() + 5, sizeof(addr.sun_path) - 1);
    if (connect(fd, (struct sockaddr*)&addr, sizeof(addr)) != 0) {
      *err = strerror(errno);
      close(fd);
      return -1;
    }
  } else {
    std::string host = target;
    std::string port = "80";
    size_t colon = target.rfind(':');
    if (colon != std::string::npos) {
      host = target.substr(0, colon);
      port = target.substr(colon + 1);
    }
    struct addrinfo hints;
    memset(&hints, 0, sizeof(hints));
    hints.ai_family = AF_UNSPEC;
    hints.ai_socktype = SOCK_STREAM;
    struct addrinfo* res = nullptr;
    int rc = getaddrinfo(host.c_str(), port.c_str(), &hints, &res);
    if (rc != 0) {
      *err = gai_strerror(rc);
      return -1;
    }
    for (struct addrinfo* ai = res; ai; ai = ai->ai_next) {
      fd = socket(ai->ai_family, ai->ai_socktype | SOCK_CLOEXEC,
                  ai->ai_protocol);
      if (fd < 0) continue;
      if (connect(fd, ai->ai_addr, ai->ai_addrlen) == 0) break;
      close(fd);
      fd = -1;
    }
    freeaddrinfo(res);
    if (fd < 0) {
      *err = "connection refused";
      return -1;
    }
  }
  struct timeval tv;
  tv.tv_sec = timeoutMs / 1000;
  tv.tv_usec = (timeoutMs % 1000) * 1000;
  setsockopt(fd, SOL_SOCKET, SO_RCVTIMEO, &tv, sizeof(tv));
  setsockopt(fd, SOL_SOCKET, SO_SNDTIMEO, &tv, sizeof(tv));
  return fd;
}

}  // namespace

namespace {

// thin transport abstraction so the request logic is shared between
// plain sockets and TLS sessions
struct Transport {
  int fd = -1;
  SSL_CTX* ctx = nullptr;
  SSL* ssl = nullptr;

  ~Transport() {
    if (ssl) {
      SSL_shutdown(ssl);
      SSL_free(ssl);
    }
    if (ctx) SSL_CTX_free(ctx);
    if (fd >= 0) close(fd);
  }

  bool startTls(const std::string& host, const TlsOptions& tls,
                std::string* err) {
    ctx = SSL_CTX_new(TLS_client_method());
    if (!ctx) {
      *err = "SSL_CTX_new failed";
      return false;
    }
    if (!tls.caFile.empty() || !tls.caPath.empty()) {
      if (SSL_CTX_load_verify_locations(
              ctx, tls.caFile.empty() ? nullptr : tls.caFile.c_str(),
              tls.caPath.empty() ? nullptr : tls.caPath.c_str()) != 1) {
        *err = "failed to load CA certificates";
        return false;
      }
    } else {
      SSL_CTX_set_default_verify_paths(ctx);
    }
    if (!tls.certFile.empty() &&
        (SSL_CTX_use_certificate_chain_file(ctx, tls.certFile.c_str()) != 1 ||
         SSL_CTX_use_PrivateKey_file(ctx, tls.keyFile.c_str(),
                                     SSL_FILETYPE_PEM) != 1)) {
      *err = "failed to load client certificate/key";
      return false;
    }
    SSL_CTX_set_verify(
        ctx, tls.insecureSkipVerify ? SSL_VERIFY_NONE : SSL_VERIFY_PEER,
        nullptr);
    ssl = SSL_new(ctx);
    SSL_set_fd(ssl, fd);
    std::string sniHost = tls.serverName.empty() ? host : tls.serverName;
    // strip :port for SNI/verification
    size_t colon = sniHost.rfind(':');
    if (colon != std::string::npos) sniHost = sniHost.substr(0, colon);
    SSL_set_tlsext_host_name(ssl, sniHost.c_str());
    if (!tls.insecureSkipVerify) SSL_set1_host(ssl, sniHost.c_str());
    if (SSL_connect(ssl) != 1) {
      unsigned long e = ERR_get_error();
      char ebuf[256];
      ERR_error_string_n(e, ebuf, sizeof(ebuf));
      *err = std::string("TLS handshake failed: ") + ebuf;
      return false;
    }
    return true;
  }

  ssize_t send(const char* data, size_t len) {
    return ssl ? SSL_write(ssl, data, (int)len) : write(fd, data, len);
  }
  ssize_t recv(char* data, size_t len) {
    return ssl ? SSL_read(ssl, data, (int)len) : read(fd, data, len);
  }
};

}  // namespace

namespace {

// Thread-local keep-alive pool: one cached plain-TCP/unix connection
// per target, giving the same connection reuse the reference's consul
// client inherits from Go's http.Transport. Plain + uncancellable
// requests only (TLS sessions and cancellable long-polls stay
// per-request). At thousands of TTL updates/sec, per-request connects
// burned measurable CPU on both the daemon and the agent.
struct ConnPool {
  std::map<std::string, int> conns;
  // close pooled fds at thread exit: consul workers are joined on
  // every generation teardown, so a leak here compounds per reload
  ~ConnPool() {
    for (auto& kv : conns) close(kv.second);
  }
};
thread_local ConnPool connPool;

int poolTake(const std::string& target) {
  auto it = connPool.conns.find(target);
  if (it == connPool.conns.end()) return -1;
  int fd = it->second;
  connPool.conns.erase(it);
  return fd;
}

void poolStore(const std::string& target, int fd) {
  auto it = connPool.conns.find(target);
  if (it != connPool.conns.end()) {
    close(it->second);
    it->second = fd;
  } else {
    connPool.conns[target] = fd;
  }
}

// incremental chunked-body decoder: returns true once the terminal
// chunk is complete, filling *out with the decoded bytes
bool tryDechunk(const std::string& data, std::string* out) {
  out->clear();
  size_t pos = 0;
  while (true) {
    size_t eol = data.find("\r\n", pos);
    if (eol == std::string::npos) return false;
    long len = strtol(data.substr(pos, eol - pos).c_str(), nullptr, 16);
    if (len < 0) return false;
    if (len == 0) return true;  // terminal chunk (ignore trailers)
    if (data.size() < eol + 2 + (size_t)len + 2) return false;
    out->append(data, eol + 2, (size_t)len);
    pos = eol + 2 + (size_t)len + 2;
  }
}

}  // namespace

ClientResult request(const std::string& target, const std::string& method,
                     const std::string& path, const std::string& body,
                     const std::string& contentType,
                     const std::map<std::string, std::string>& headers,
                     int timeoutMs, const TlsOptions* tls,
                     CancelToken* cancel) {
  ClientResult result;
  if (cancel && cancel->cancelled()) {
    result.error = "cancelled";
    return result;
  }
  const bool tlsOn = tls && tls->enabled;
  const bool reusable = !tlsOn && cancel == nullptr;

  for (int attempt = 0; attempt < 2; attempt++) {
    result = ClientResult{};
    Transport t;
    bool fromPool = false;
    if (reusable && attempt == 0) {
      t.fd = poolTake(target);
      fromPool = t.fd >= 0;
    }
    if (t.fd < 0) {
      t.fd = connectTarget(target, timeoutMs, &result.error);
      if (t.fd < 0) return result;
    } else {
      // refresh the timeouts: the pooled fd carries the previous
      // request's deadline settings
      struct timeval tv;
      tv.tv_sec = timeoutMs / 1000;
      tv.tv_usec = (timeoutMs % 1000) * 1000;
      setsockopt(t.fd, SOL_SOCKET, SO_RCVTIMEO, &tv, sizeof(tv));
      setsockopt(t.fd, SOL_SOCKET, SO_SNDTIMEO, &tv, sizeof(tv));
    }
    struct Disarm {
      CancelToken* ct;
      ~Disarm() {
        if (ct) ct->disarm();
      }
    } disarm{cancel};
    if (cancel) cancel->arm(t.fd);

    std::string host = target.rfind("unix:", 0) == 0 ? "localhost" : target;
    if (tlsOn) {
      if (!t.startTls(host, *tls, &result.error)) return result;
    }

    std::string req = method + " " + path + " HTTP/1.1\r\n";
    req += "Host: " + host + "\r\n";
    req += reusable ? "Connection: keep-alive\r\n" : "Connection: close\r\n";
    for (auto& kv : headers) req += kv.first + ": " + kv.second + "\r\n";
    if (!body.empty() || method == "POST" || method == "PUT") {
      req += "Content-Type: " + contentType + "\r\n";
      req += "Content-Length: " + std::to_string(body.size()) + "\r\n";
    }
    req += "\r\n";
    req += body;

    bool ioFailed = false;
    size_t off = 0;
    while (off < req.size()) {
      ssize_t n = t.send(req.data() + off, req.size() - off);
      if (n <= 0) {
        if (n < 0 && errno == EINTR && !t.ssl) continue;
        ioFailed = true;
        break;
      }
      off += n;
    }
    if (ioFailed) {
      if (fromPool) continue;  // stale keep-alive: retry on a fresh conn
      result.error = "write failed";
      return result;
    }

    // read incrementally: headers first, then exactly the framed body
    std::string resp;
    char buf[8192];
    size_t headerEnd = std::string::npos;
    auto readMore = [&]() -> bool {
      while (true) {
        ssize_t n = t.recv(buf, sizeof(buf));
        if (n > 0) {
          resp.append(buf, n);
          return true;
        }
        if (n < 0 && errno == EINTR && !t.ssl) continue;
        return false;  // EOF or error/timeout
      }
    };
    bool eof = false;
    while ((headerEnd = resp.find("\r\n\r\n")) == std::string::npos) {
      if (!readMore()) {
        eof = true;
        break;
      }
    }
    if (headerEnd == std::string::npos || resp.compare(0, 5, "HTTP/") != 0) {
      if (fromPool && resp.empty()) continue;  // server closed the idle conn
      result.error = (cancel && cancel->cancelled()) ? "cancelled"
                                                     : "malformed response";
      return result;
    }

    size_t sp = resp.find(' ');
    result.status = atoi(resp.c_str() + sp + 1);
    // response headers (lower-cased keys)
    size_t pos = resp.find("\r\n") + 2;
    while (pos < headerEnd) {
      size_t eol = resp.find("\r\n", pos);
      if (eol == std::string::npos || eol > headerEnd) break;
      std::string line = resp.substr(pos, eol - pos);
      pos = eol + 2;
      size_t colon = line.find(':');
      if (colon == std::string::npos) continue;
      std::string key = line.substr(0, colon);
      for (auto& ch : key) ch = (char)tolower((unsigned char)ch);
      size_t vstart = colon + 1;
      while (vstart < line.size() && line[vstart] == ' ') vstart++;
      result.headers[key] = line.substr(vstart);
    }

    auto lower = [](std::string v) {
      for (auto& ch : v) ch = (char)tolower((unsigned char)ch);
      return v;
    };
    bool chunked = false;
    auto te = result.headers.find("transfer-encoding");
    if (te != result.headers.end() &&
        lower(te->second).find("chunked") != std::string::npos)
      chunked = true;
    bool bodyFramed = true;
    if (chunked) {
      std::string decoded;
      while (!tryDechunk(resp.substr(headerEnd + 4), &decoded)) {
        if (eof || !readMore()) {
          // timeout/EOF mid-body: a truncated framed body is an error,
          // not a short success (Go's client errors the same way)
          result.error = "truncated response body";
          return result;
        }
      }
      result.body = decoded;
    } else if (result.headers.count("content-length")) {
      size_t want =
          (size_t)atoll(result.headers["content-length"].c_str());
      while (resp.size() < headerEnd + 4 + want) {
        if (eof || !readMore()) {
          result.error = "truncated response body";
          return result;
        }
      }
      result.body = resp.substr(headerEnd + 4, want);
    } else {
      // no framing: read to EOF, connection not reusable
      while (readMore()) {
      }
      result.body = resp.substr(headerEnd + 4);
      bodyFramed = false;
    }

    result.ok = true;
    // reuse only when both sides agreed and the body was fully framed
    bool serverKeeps = true;
    auto ch = result.headers.find("connection");
    if (ch != result.headers.end() &&
        lower(ch->second).find("close") != std::string::npos)
      serverKeeps = false;
    if (reusable && bodyFramed && serverKeeps && result.status > 0) {
      int fd = t.fd;
      t.fd = -1;  // keep it out of Transport's destructor
      poolStore(target, fd);
    }
    return result;
  }
  result.error = "write failed";
  return result;
}

}  // namespace http
}  // namespace cpilot
