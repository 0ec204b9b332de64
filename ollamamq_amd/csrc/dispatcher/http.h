// Minimal dependency-free HTTP/1.1 server + client for the dispatcher.
// Native equivalent of the reference's axum server + reqwest client
// (reference src/main.rs:310-339 serve, src/dispatcher.rs:742-778 proxy).
// Thread-per-connection with keep-alive; chunked streaming responses for
// token streams (Ollama JSON-lines and OpenAI SSE); 1 GB body cap
// (reference main.rs:302).
#pragma once

#include <atomic>
#include <functional>
#include <map>
#include <memory>
#include <string>
#include <thread>
#include <vector>

namespace omq {

struct HttpRequest {
    std::string method;
    std::string path;        // path only (no query)
    std::string query;       // raw query string ("" if none)
    std::vector<std::pair<std::string, std::string>> headers;
    std::string body;
    std::string client_ip;

    std::string header(const std::string& name) const;  // case-insensitive
};

// Response writer bound to one connection.  Either send() once, or
// begin_stream() + write_chunk()* + end_stream().
class HttpConn {
public:
    explicit HttpConn(int fd) : fd_(fd) {}
    bool send(int status,
              const std::vector<std::pair<std::string, std::string>>& headers,
              const std::string& body);
    bool begin_stream(
        int status,
        const std::vector<std::pair<std::string, std::string>>& headers);
    bool write_chunk(const std::string& data);  // false => client gone
    bool end_stream();
    bool alive() const { return alive_; }
    bool streaming() const { return streaming_; }
    bool responded() const { return responded_; }

private:
    bool write_all(const char* p, size_t n);
    int fd_;
    bool streaming_ = false;
    bool responded_ = false;
    bool alive_ = true;
};

using HttpHandler = std::function<void(const HttpRequest&, HttpConn&)>;

class HttpServer {
public:
    HttpServer(std::string host, int port, HttpHandler handler);
    ~HttpServer();
    bool start(std::string* err);   // binds + spawns accept loop
    void stop();
    int port() const { return bound_port_; }  // actual port (0 => ephemeral)

private:
    void accept_loop();
    void handle_conn(int fd, std::string peer_ip);
    std::string host_;
    int port_;
    int bound_port_ = 0;
    HttpHandler handler_;
    int listen_fd_ = -1;
    std::thread accept_thread_;
    std::atomic<bool> stopping_{false};
    // liveness guard: thread-per-connection needs a hard cap — a storm
    // of half-open connections must shed load instead of starving accept
    std::atomic<int> live_conns_{0};
};

// ------------------------------------------------------------- client side
struct HttpResponse {
    int status = -1;          // <0: transport error
    std::vector<std::pair<std::string, std::string>> headers;
    std::string body;
    std::string error;
    std::string header(const std::string& name) const;
};

// Blocking request to http://host:port/path.  timeout_ms covers connect +
// total transfer.  If on_chunk is set, body bytes stream into it as they
// arrive (body stays empty) — used by the proxy executor; returning false
// from on_chunk aborts the transfer (client disconnected).
HttpResponse http_request(
    const std::string& method, const std::string& url,
    const std::vector<std::pair<std::string, std::string>>& headers,
    const std::string& body, int timeout_ms,
    const std::function<bool(const char*, size_t)>& on_chunk = nullptr,
    const std::function<void(int, const std::vector<std::pair<std::string,
                             std::string>>&)>& on_status = nullptr);

}  // namespace omq
