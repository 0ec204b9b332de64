#include "core.h"

#include <fstream>

#include "json.h"

namespace omq {

// ------------------------------------------------------- ResponseChannel
bool ResponseChannel::send_status(
    int st, std::vector<std::pair<std::string, std::string>> hdrs) {
    std::lock_guard<std::mutex> g(mu);
    if (client_gone) return false;
    status = st;
    headers = std::move(hdrs);
    started = true;
    cv.notify_all();
    return true;
}

bool ResponseChannel::send_chunk(std::string data) {
    std::unique_lock<std::mutex> g(mu);
    // bounded channel (reference mpsc cap 32, src/dispatcher.rs:872):
    // block the producer until the consumer drains or disconnects
    cv.wait(g, [&] { return chunks.size() < cap || client_gone; });
    if (client_gone) return false;
    chunks.push_back(std::move(data));
    cv.notify_all();
    return true;
}

void ResponseChannel::finish() {
    std::lock_guard<std::mutex> g(mu);
    done = true;
    started = true;  // even if no status was sent (error path)
    cv.notify_all();
}

bool ResponseChannel::wait_started(int timeout_ms) {
    std::unique_lock<std::mutex> g(mu);
    return cv.wait_for(g, std::chrono::milliseconds(timeout_ms),
                       [&] { return started || done; });
}

bool ResponseChannel::next_chunk(std::string* out, int timeout_ms) {
    std::unique_lock<std::mutex> g(mu);
    if (!cv.wait_for(g, std::chrono::milliseconds(timeout_ms),
                     [&] { return !chunks.empty() || done; }))
        return false;
    if (chunks.empty()) return false;  // done
    *out = std::move(chunks.front());
    chunks.pop_front();
    cv.notify_all();
    return true;
}

void ResponseChannel::mark_client_gone() {
    std::lock_guard<std::mutex> g(mu);
    client_gone = true;
    cv.notify_all();
}

// ------------------------------------------------------------- blocklist
// C17 (reference src/dispatcher.rs:19-25,279-334): JSON file in CWD,
// loaded at startup, rewritten on every change.
void AppState::load_blocked() {
    std::ifstream f(blocked_path);
    if (!f) return;
    std::string text((std::istreambuf_iterator<char>(f)),
                     std::istreambuf_iterator<char>());
    auto j = Json::parse(text);
    if (!j) return;
    std::lock_guard<std::mutex> g(blocked_mu);
    // the reference's serde field names are "users"/"ips"
    // (src/dispatcher.rs:22-25) — a migrated blocked_items.json uses
    // those; also accept our earlier "blocked_*" spelling
    for (const char* key : {"users", "blocked_users"})
        if (const Json* users = j->find(key))
            for (const auto& u : users->arr)
                if (u.is_str()) blocked_users.insert(u.str);
    for (const char* key : {"ips", "blocked_ips"})
        if (const Json* ips = j->find(key))
            for (const auto& ip : ips->arr)
                if (ip.is_str()) blocked_ips.insert(ip.str);
}

void AppState::save_blocked() const {
    Json j = Json::object();
    {
        std::lock_guard<std::mutex> g(blocked_mu);
        Json users = Json::array();
        for (const auto& u : blocked_users)
            users.arr.push_back(Json::string(u));
        Json ips = Json::array();
        for (const auto& ip : blocked_ips)
            ips.arr.push_back(Json::string(ip));
        // write the reference's field names (src/dispatcher.rs:22-25)
        j.set("users", std::move(users));
        j.set("ips", std::move(ips));
    }
    std::ofstream f(blocked_path, std::ios::trunc);
    f << j.dump();
}

}  // namespace omq
