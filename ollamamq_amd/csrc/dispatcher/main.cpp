// C1: CLI entry point for the ollamamq-server dispatcher binary.
// Flag surface and precedence (CLI > appconf.yaml > defaults) mirror the
// reference (reference src/main.rs:26-71 Args; :160-191 merge; :220-257
// wiring; defaults: port 11435, host 127.0.0.1, timeout 300 s,
// load_keep_alive 86400 s, stuck_timeout 60 s, default backend
// http://localhost:11434).  Auth key via env OLLAMA_MQ_API_KEY.
#include <csignal>
#include <unistd.h>

#include <cstdlib>
#include <cstring>
#include <iostream>
#include <sstream>
#include <thread>

#include "config.h"
#include "control.h"
#include "core.h"
#include "server.h"
#include "tui.h"

using namespace omq;

static void usage() {
    std::cout <<
        "ollamamq-server — MI355X-native multi-user LLM dispatcher\n"
        "\n"
        "  -p, --port <P>            listen port (default 11435)\n"
        "  -H, --host <H>            bind host (default 127.0.0.1)\n"
        "  -o, --backend-urls <U,..> backend URLs (alias --ollama-urls;\n"
        "                            default http://localhost:11434)\n"
        "  -w, --workers <SPEC,..>   in-process GPU worker backends\n"
        "                            (unix socket paths of running workers)\n"
        "  -t, --timeout <S>         request timeout seconds (default 300)\n"
        "      --load-keep-alive <S> control-load keep_alive (default 86400)\n"
        "      --stuck-timeout <S>   queue stuck timeout -> 503 (default 60)\n"
        "      --probe-interval-ms <M> health probe cadence (default 10000)\n"
        "      --allow-all-routes    proxy unknown routes too\n"
        "      --no-tui              headless (logs to stderr)\n"
        "  -c, --model-config <F>    config file (default appconf.yaml)\n"
        "  -V, --version             print version\n";
}

int main(int argc, char** argv) {
    std::string config_path = "appconf.yaml";
    std::string cli_host, cli_backends, cli_workers;
    int cli_port = -1;
    int64_t cli_timeout = -1, cli_keep = -1, cli_stuck = -1;
    int cli_probe = -1;
    bool cli_allow_all = false, no_tui = false;

    for (int i = 1; i < argc; i++) {
        const std::string a = argv[i];
        auto next = [&]() -> const char* {
            return i + 1 < argc ? argv[++i] : "";
        };
        if (a == "-p" || a == "--port") cli_port = atoi(next());
        else if (a == "-H" || a == "--host") cli_host = next();
        else if (a == "-o" || a == "--backend-urls" || a == "--ollama-urls")
            cli_backends = next();
        else if (a == "-w" || a == "--workers") cli_workers = next();
        else if (a == "-t" || a == "--timeout") cli_timeout = atoll(next());
        else if (a == "--load-keep-alive") cli_keep = atoll(next());
        else if (a == "--stuck-timeout") cli_stuck = atoll(next());
        else if (a == "--probe-interval-ms") cli_probe = atoi(next());
        else if (a == "--allow-all-routes") cli_allow_all = true;
        else if (a == "--no-tui") no_tui = true;
        else if (a == "-c" || a == "--model-config") config_path = next();
        else if (a == "-h" || a == "--help") { usage(); return 0; }
        else if (a == "-V" || a == "--version") {
            std::cout << "ollamamq-amd 0.2.0\n";
            return 0;
        }
        else {
            std::cerr << "unknown flag " << a << "\n";
            usage();
            return 2;
        }
    }

    AppConfig cfg;
    std::string err;
    if (!load_config(config_path, &cfg, &err)) {
        std::cerr << "config error in " << config_path << ": " << err << "\n";
        return 2;
    }

    Server server(config_path);
    auto& st = server.state();
    // precedence: CLI > file > defaults (reference main.rs:160-191)
    st.settings = cfg.settings;
    if (cli_port >= 0) st.settings.port = cli_port;
    if (!cli_host.empty()) st.settings.host = cli_host;
    if (cli_timeout >= 0) st.settings.timeout_s = cli_timeout;
    if (cli_keep >= 0) st.settings.load_keep_alive_s = cli_keep;
    if (cli_stuck >= 0) st.settings.stuck_timeout_s = cli_stuck;
    if (cli_probe > 0) st.settings.probe_interval_ms = cli_probe;
    if (cli_allow_all) st.settings.allow_all_routes = true;
    if (const char* k = getenv("OLLAMA_MQ_API_KEY")) st.settings.api_key = k;
    {
        std::lock_guard<std::mutex> g(st.models_mu);
        st.model_config = cfg.models;
    }

    std::vector<std::string> urls;
    if (!cli_backends.empty()) {
        std::stringstream ss(cli_backends);
        std::string u;
        while (std::getline(ss, u, ',')) urls.push_back(u);
    } else {
        urls = cfg.backends;
    }
    if (urls.empty() && cli_workers.empty())
        urls.push_back("http://localhost:11434");
    for (const auto& u : urls) server.add_http_backend(u);
    if (!cli_workers.empty()) {
        std::stringstream ss(cli_workers);
        std::string w;
        while (std::getline(ss, w, ','))
            add_worker_backend(server, w);
    }

    if (no_tui)
        st.log.set_stderr_sink(true);
    else
        st.log.set_file_sink("ollamamq.log");
    // log-level env filter (reference RUST_LOG, src/main.rs:208,215);
    // OLLAMAMQ_LOG is ours, RUST_LOG honored for migration friendliness
    for (const char* var : {"OLLAMAMQ_LOG", "RUST_LOG"})
        if (const char* lv = getenv(var))
            if (std::string(lv).find("debug") != std::string::npos)
                st.log.set_debug(true);

    if (!server.start(&err)) {
        std::cerr << "failed to start: " << err << "\n";
        return 1;
    }
    std::cerr << "ollamamq-server listening on " << st.settings.host << ":"
              << server.port() << " with "
              << st.backends.size() << " backend(s)\n";

    // startup config-apply: wait (≤30 s) until every backend has an api
    // type, then apply the models section (reference main.rs:234-257)
    std::thread([&st] {
        for (int i = 0; i < 60; i++) {
            std::this_thread::sleep_for(std::chrono::milliseconds(500));
            std::lock_guard<std::mutex> g(st.backends_mu);
            bool all = !st.backends.empty();
            for (const auto& b : st.backends)
                if (b.api_type == ApiType::Unknown && b.is_online) all = false;
            if (all) break;
        }
        apply_model_config(st);
    }).detach();

    if (no_tui) {
        // headless: run until SIGINT/SIGTERM, then stop cleanly
        static std::atomic<bool> quit{false};
        std::signal(SIGINT, [](int) { quit = true; });
        std::signal(SIGTERM, [](int) { quit = true; });
        while (!quit)
            std::this_thread::sleep_for(std::chrono::milliseconds(200));
        server.stop();
    } else {
        run_tui(server);  // returns on 'q'
        server.stop();
    }
    return 0;
}
