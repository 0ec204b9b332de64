// C1/C3/C4/C5/C8/C9 wiring: the dispatcher server.
// Router surface (reference src/main.rs:264-308): 20 proxied routes +
// 3 admin + /health (+ optional fallback proxy), auth middleware
// (opt-in API key), ingress -> per-user queues -> scheduler -> executor.
#pragma once

#include <memory>
#include <string>
#include <thread>
#include <vector>

#include "core.h"
#include "http.h"

namespace omq {

class Server {
public:
    explicit Server(std::string config_path = "appconf.yaml");
    ~Server();

    AppState& state() { return st_; }
    void add_http_backend(const std::string& url);
    void add_backend(std::shared_ptr<Backend> impl, const std::string& name);

    bool start(std::string* err);       // binds HTTP, spawns worker+health
    void stop();
    int port() const;

    // exposed for tests
    void handle(const HttpRequest& req, HttpConn& conn);
    std::string config_path;

private:
    void proxy_handler(const HttpRequest& req, HttpConn& conn);
    void run_worker();
    void health_loop();
    bool auth_ok(const HttpRequest& req) const;

    AppState st_;
    std::unique_ptr<HttpServer> http_;
    std::thread worker_, health_;
    std::atomic<bool> started_{false};
};

bool is_proxied_route(const std::string& path);

}  // namespace omq
