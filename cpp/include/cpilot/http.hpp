// Minimal HTTP/1.1: a nonblocking server that runs on the reactor
// (serves the unix-socket control plane and the TCP telemetry endpoint)
// and a small blocking client (Consul API calls from worker threads, and
// the control-socket client used by subcommands).
// Connections are close-after-response; the reference also disables
// keep-alives (control/control.go:110).
#pragma once

#include <functional>
#include <map>
#include <mutex>
#include <memory>
#include <string>

#include "cpilot/loop.hpp"

namespace cpilot {
namespace http {

struct Request {
  std::string method;
  std::string path;
  std::string query;
  std::map<std::string, std::string> headers;  // lower-cased keys
  std::string body;
};

struct Response {
  int status = 200;
  std::string contentType = "text/plain; charset=utf-8";
  std::string body;
};

const char* statusText(int code);

using Handler = std::function<Response(const Request&)>;

class Server {
 public:
  Server(Loop& loop, Handler handler);
  ~Server();

  // Bind + listen on a unix socket path (unlinks nothing; caller manages
  // stale sockets) or a TCP port. Return false on bind/listen failure.
  bool listenUnix(const std::string& path, std::string* err);
  bool listenTcp(const std::string& ip, int port, std::string* err);

  void stop();  // close listener and all connections

 public:
  struct Conn;

 private:
  void acceptReady();
  void connReadable(std::shared_ptr<Conn> c);
  void connWritable(std::shared_ptr<Conn> c);
  void beginWrite(const std::shared_ptr<Conn>& c, const Response& resp);
  void closeConn(const std::shared_ptr<Conn>& c);
  static Response errorResponse(int status);

  Loop& loop_;
  Handler handler_;
  int listenFd_ = -1;
  std::map<int, std::shared_ptr<Conn>> conns_;
};

// ---- blocking client ----

struct ClientResult {
  bool ok = false;
  int status = 0;
  std::string body;
  std::string error;
  std::map<std::string, std::string> headers;  // lower-cased keys
};

// Cancellation for long-polling requests (Consul blocking queries):
// cancel() shuts the socket down from another thread so the blocked
// read returns immediately.
class CancelToken {
 public:
  void arm(int fd);     // called by request() once connected
  void disarm();        // called by request() before closing
  void cancel();        // any thread
  bool cancelled();

 private:
  std::mutex mu_;
  int fd_ = -1;
  bool cancelled_ = false;
};

// TLS options for https targets (OpenSSL; reference parity with the
// consul api.TLSConfig fields, discovery/config.go:29-61)
struct TlsOptions {
  bool enabled = false;
  std::string caFile;
  std::string caPath;
  std::string certFile;
  std::string keyFile;
  std::string serverName;          // SNI + hostname verification override
  bool insecureSkipVerify = false;
};

// target: "unix:<path>" or "host:port"
ClientResult request(const std::string& target, const std::string& method,
                     const std::string& path, const std::string& body,
                     const std::string& contentType = "application/json",
                     const std::map<std::string, std::string>& headers = {},
                     int timeoutMs = 10000,
                     const TlsOptions* tls = nullptr,
                     CancelToken* cancel = nullptr);

}  // namespace http
}  // namespace cpilot
