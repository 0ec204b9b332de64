// C19: terminal dashboard — native ANSI-escape renderer mirroring the
// reference's panels and keymap (reference src/tui.rs: Backends / Users /
// Blocked / Logs panels, 100 ms frame cadence, snapshot-per-frame so no
// state lock is held while rendering; keys: j/k nav, Tab panel cycle,
// L/U load/unload (typed model name), r config reload, p VIP, b Boost,
// x/X block user/IP, u unblock, ? help, q/Esc quit).
#include "tui.h"

#include <poll.h>
#include <termios.h>
#include <unistd.h>

#include <algorithm>
#include <cstdio>
#include <set>
#include <sstream>
#include <vector>

#include "control.h"
#include "core.h"
#include "scheduler.h"
#include "server.h"

namespace omq {

namespace {

struct Snapshot {
    int64_t processed_total = 0, dropped_total = 0, queued_total = 0;
    double p50_wait_ms = 0;
    struct B {
        std::string url;
        bool online;
        std::string api;
        int active;
        int64_t processed;
        std::string loaded;
        std::string avail;
        std::vector<std::string> models;   // sorted available (cursor)
        std::string op;
    };
    struct U {
        std::string name;
        size_t queued;
        int64_t processing, processed, dropped;
        bool vip, boost;
    };
    std::vector<B> backends;
    std::vector<U> users;
    std::vector<std::string> blocked;
    std::vector<LogEvent> logs;
    uint64_t counter;
};

Snapshot capture(AppState& st) {
    Snapshot s;
    {
        std::scoped_lock lk(st.control_mu, st.backends_mu);
        for (size_t i = 0; i < st.backends.size(); i++) {
            const auto& b = st.backends[i];
            Snapshot::B e;
            e.url = b.url;
            e.online = b.is_online;
            e.api = api_type_name(b.api_type);
            e.active = b.active_requests;
            e.processed = b.processed_count;
            for (const auto& m : b.loaded_models) {
                if (!e.loaded.empty()) e.loaded += ",";
                e.loaded += m;
            }
            for (const auto& m : b.available_models) {
                if (!e.avail.empty()) e.avail += ",";
                e.avail += m;
                e.models.push_back(m);
            }
            std::sort(e.models.begin(), e.models.end());
            auto op = st.control_ops.find(i);
            if (op != st.control_ops.end()) {
                e.op = (op->second.action == ControlAction::Load ? "load "
                                                                 : "unload ") +
                       op->second.model;
            } else {
                // flash the latest completed result for 10 s (reference
                // control.rs:33-37,91-101 history visibility)
                for (auto it = st.control_history.rbegin();
                     it != st.control_history.rend(); ++it) {
                    if (it->backend != i) continue;
                    if (now_ms() - it->finished_ms < 10000)
                        e.op = std::string(it->ok ? "done: " : "FAILED: ") +
                               it->model;
                    break;
                }
            }
            s.backends.push_back(std::move(e));
        }
    }
    {
        std::scoped_lock lk(st.queues_mu, st.prio_mu);
        for (const auto& [name, us] : st.users) {
            s.processed_total += us.processed;
            s.dropped_total += us.dropped;
            s.queued_total += (int64_t)us.queue.size();
            Snapshot::U u;
            u.name = name;
            u.queued = us.queue.size();
            u.processing = us.processing;
            u.processed = us.processed;
            u.dropped = us.dropped;
            u.vip = name == st.vip_user;
            u.boost = name == st.boost_user;
            s.users.push_back(std::move(u));
        }
    }
    {
        std::lock_guard<std::mutex> g(st.blocked_mu);
        for (const auto& u : st.blocked_users) s.blocked.push_back("user " + u);
        for (const auto& ip : st.blocked_ips) s.blocked.push_back("ip " + ip);
    }
    s.logs = st.log.snapshot();
    s.counter = st.sched_counter.load();
    {
        std::lock_guard<std::mutex> g(st.waits_mu);
        std::vector<int64_t> w(st.wait_samples_ms.begin(),
                               st.wait_samples_ms.end());
        if (!w.empty()) {
            std::sort(w.begin(), w.end());
            s.p50_wait_ms = (double)w[w.size() / 2];
        }
    }
    return s;
}

struct RawTerm {
    termios orig{};
    bool ok = false;
    RawTerm() {
        if (tcgetattr(STDIN_FILENO, &orig) == 0) {
            termios raw = orig;
            raw.c_lflag &= ~(ECHO | ICANON);
            raw.c_cc[VMIN] = 0;
            raw.c_cc[VTIME] = 0;
            tcsetattr(STDIN_FILENO, TCSANOW, &raw);
            ok = true;
        }
        printf("\x1b[?1049h\x1b[?25l");  // alt screen, hide cursor
        fflush(stdout);
    }
    ~RawTerm() {
        printf("\x1b[?1049l\x1b[?25h");
        fflush(stdout);
        if (ok) tcsetattr(STDIN_FILENO, TCSANOW, &orig);
    }
};

std::string pad(std::string s, size_t w) {
    if (s.size() > w) return s.substr(0, w - 1) + "…";
    s.resize(w, ' ');
    return s;
}

}  // namespace

void run_tui(Server& server) {
    AppState& st = server.state();
    RawTerm term;
    int sel_user = 0;
    int sel_backend = 0;
    int focus = 1;            // 0 = Backends panel, 1 = Users panel (Tab)
    bool show_all = false;    // 'a': show available models per backend
    std::set<int> expanded;   // Space/Enter: expand one backend's models
    std::map<int, int> mcur;  // per-backend model cursor (reference
                              // tui.rs:607-636): Tab walks the expanded
                              // backend's sorted models; L/U act on the
                              // highlighted model directly
    std::set<int> unfolded;   // expanded lists fold to 5 until the
                              // cursor walks past (tui.rs:631-634)
    std::string input;        // typed model name for L/U
    char input_mode = 0;      // 'L' or 'U' when typing
    bool help = false;
    std::string status_msg;

    // One key from stdin.  CSI sequences (arrow keys, Shift-Tab — reference
    // tui.rs supports ↑/↓ nav and S-Tab panel cycle via crossterm) are
    // translated to 256+vi-equivalent so bare Esc (27) still means quit and
    // input mode can ignore navigation keys; unknown sequences return 0.
    auto read_key = [&]() -> int {
        char c;
        if (read(STDIN_FILENO, &c, 1) != 1) return 0;
        if (c != 27) return (unsigned char)c;
        pollfd p2{STDIN_FILENO, POLLIN, 0};
        if (poll(&p2, 1, 10) <= 0) return 27;  // bare Esc
        char b = 0;
        if (read(STDIN_FILENO, &b, 1) != 1 || b != '[') return 27;
        char f = 0;
        if (poll(&p2, 1, 10) > 0 && read(STDIN_FILENO, &f, 1) != 1) f = 0;
        switch (f) {
            case 'A': return 256 + 'k';  // up
            case 'B': return 256 + 'j';  // down
            case 'C': return 256 + 'l';  // right
            case 'D': return 256 + 'h';  // left
            case 'Z': return 256 + 'Z';  // Shift-Tab
        }
        return 0;
    };

    while (true) {
        Snapshot s = capture(st);
        std::ostringstream out;
        out << "\x1b[H\x1b[2J";
        out << "\x1b[1m ollamamq-amd dispatcher — MI355X \x1b[0m"
            << "  backends:" << s.backends.size()
            << "  users:" << s.users.size()
            << "  queued:" << s.queued_total
            << "  done:" << s.processed_total
            << "  dropped:" << s.dropped_total
            << "  p50-wait:" << (int)s.p50_wait_ms << "ms"
            << "  sched:" << s.counter << "\r\n";
        if (help) {
            out << "\r\n \x1b[1mKeys\x1b[0m\r\n"
                   "  Tab     model cursor / panel focus (S-Tab, h/l: panel)\r\n"
                   "  j/k/arrows  move selection in the focused panel\r\n"
                   "  a       toggle available-models listing\r\n"
                   "  p / b   toggle VIP / Boost for selected user\r\n"
                   "  x / X   block selected user / their IP\r\n"
                   "  u       unblock all\r\n"
                   "  L / U   load / unload model (type name, Enter)\r\n"
                   "  r       reload models from appconf.yaml\r\n"
                   "  ?       close help\r\n"
                   "  q/Esc   quit\r\n";
            fputs(out.str().c_str(), stdout);
            fflush(stdout);
            pollfd pfd{STDIN_FILENO, POLLIN, 0};
            if (poll(&pfd, 1, 100) > 0) {
                const int c = read_key();
                if (c == 'q' || c == 27) return;
                if (c) help = false;
            }
            continue;
        }
        out << "\x1b[7m" << pad(focus == 0 ? " Backends [focused]"
                                          : " Backends", 90)
            << "\x1b[0m\r\n";
        for (int i = 0; i < (int)s.backends.size(); i++) {
            const auto& b = s.backends[i];
            out << (focus == 0 && i == sel_backend ? ">" : " ")
                << (b.online ? "\x1b[32m●\x1b[0m " : "\x1b[31m○\x1b[0m ")
                << pad(b.url, 32) << pad(b.api, 8)
                << "act:" << b.active << " done:" << b.processed << " "
                << pad(b.loaded, 24)
                << (b.op.empty() ? "" : " [" + b.op + "]") << "\r\n";
            if (expanded.count(i) && !b.models.empty()) {
                // per-model rows with the cursor marker; folded to 5
                // unless 'a' or the cursor forced the full list
                const bool full = show_all || unfolded.count(i);
                const size_t lim = full ? b.models.size()
                                        : std::min<size_t>(5,
                                                           b.models.size());
                auto cit = mcur.find(i);
                for (size_t mi = 0; mi < lim; mi++) {
                    const bool cur = cit != mcur.end() &&
                                     (int)mi == cit->second;
                    out << (cur ? "    \x1b[7m> " : "      ")
                        << pad(b.models[mi], 40)
                        << (cur ? "\x1b[0m" : "") << "\r\n";
                }
                if (lim < b.models.size())
                    out << "      … +"
                        << (b.models.size() - lim) << " more (Tab)\r\n";
            } else if (show_all && !b.avail.empty()) {
                out << "    available: " << pad(b.avail, 80) << "\r\n";
            }
        }
        out << "\x1b[7m"
            << pad(focus == 1 ? " Users [focused] (j/k, p VIP, b Boost, "
                                "x block)"
                              : " Users", 90)
            << "\x1b[0m\r\n";
        for (int i = 0; i < (int)s.users.size(); i++) {
            const auto& u = s.users[i];
            out << (focus == 1 && i == sel_user ? " >" : "  ")
                << pad(u.name + (u.vip ? " ★" : "") + (u.boost ? " ⚡" : ""),
                       24)
                << " q:" << u.queued << " run:" << u.processing
                << " done:" << u.processed << " drop:" << u.dropped
                << "\r\n";
        }
        {   // Queues panel: per-user load bars (queued + in flight as a
            // share of the total — reference tui.rs:1124-1163; VIP '*',
            // boost '+', processing shown cyan)
            int64_t total_q = 0;
            for (const auto& u : s.users)
                total_q += (int64_t)u.queued + u.processing;
            out << "\x1b[7m" << pad(" Queues (" +
                                    std::to_string(total_q) +
                                    " waiting/in-flight)", 90)
                << "\x1b[0m\r\n";
            for (const auto& u : s.users) {
                const int64_t ql = (int64_t)u.queued + u.processing;
                const int barw = (int)(std::min<double>(ql / 20.0, 1.0)
                                       * 36.0);
                const double pct = total_q > 0
                                       ? 100.0 * (double)ql / total_q
                                       : 0.0;
                std::string bar(barw, '#');
                const char* col = u.vip ? "\x1b[35m"
                                  : u.boost ? "\x1b[33m"
                                  : u.processing > 0 ? "\x1b[36m"
                                                     : "\x1b[32m";
                char pbuf[32];
                snprintf(pbuf, sizeof pbuf, "%lld (%.0f%%)",
                         (long long)ql, pct);
                out << "  " << pad(u.name, 18) << col << pad(bar, 38)
                    << "\x1b[0m " << pbuf << "\r\n";
            }
        }
        if (!s.blocked.empty()) {
            out << "\x1b[7m" << pad(" Blocked (u unblock)", 90)
                << "\x1b[0m\r\n";
            for (const auto& b : s.blocked) out << "  " << b << "\r\n";
        }
        out << "\x1b[7m" << pad(" Logs", 90) << "\x1b[0m\r\n";
        const size_t n0 = s.logs.size() > 9 ? s.logs.size() - 9 : 0;
        for (size_t i = n0; i < s.logs.size(); i++)
            out << "  " << pad(s.logs[i].kind, 4) << s.logs[i].text
                << "\r\n";
        if (input_mode)
            out << "\r\n " << (input_mode == 'L' ? "load" : "unload")
                << " model: " << input << "_\r\n";
        else
            out << "\r\n " << status_msg
                << "  [Tab]panel [L]oad [U]nload [r]eload [p]VIP [b]Boost "
                   "[x]block [u]unblock [a]ll [?]help [q]uit\r\n";
        fputs(out.str().c_str(), stdout);
        fflush(stdout);

        pollfd pfd{STDIN_FILENO, POLLIN, 0};
        if (poll(&pfd, 1, 100) > 0) {
            int c = read_key();
            if (c) {
                if (c >= 256 && input_mode) continue;  // nav keys ignored
                if (c >= 256) c -= 256;                // arrows → j/k/h/l
                if (input_mode) {
                    if (c == 27) {  // Esc
                        input_mode = 0;
                        input.clear();
                    } else if (c == '\n' || c == '\r') {
                        ControlRequest req;
                        req.action = input_mode == 'L'
                                         ? ControlAction::Load
                                         : ControlAction::Unload;
                        req.model = input;
                        req.backend_idx = (size_t)sel_backend;
                        auto r = start_model_control(st, req);
                        status_msg = r.http_status == 202
                                         ? "accepted"
                                         : r.body.get_str("error");
                        input_mode = 0;
                        input.clear();
                    } else if (c == 127 || c == 8) {
                        if (!input.empty()) input.pop_back();
                    } else if (c >= 32) {
                        input += c;
                    }
                    continue;
                }
                auto sel_name = [&]() -> std::string {
                    if (sel_user < (int)s.users.size())
                        return s.users[sel_user].name;
                    return "";
                };
                switch (c) {
                    case 'q':
                    case 27:
                        return;
                    case '\t':
                        // Tab advances the model cursor of the selected
                        // expanded backend; falls through to panel cycle
                        // when there is none (reference tui.rs:607-636)
                        if (focus == 0 && expanded.count(sel_backend) &&
                            sel_backend < (int)s.backends.size() &&
                            !s.backends[sel_backend].models.empty()) {
                            const int len =
                                (int)s.backends[sel_backend].models.size();
                            auto it = mcur.find(sel_backend);
                            const int next =
                                it == mcur.end() ? 0
                                                 : (it->second + 1) % len;
                            mcur[sel_backend] = next;
                            if (next >= 5) unfolded.insert(sel_backend);
                            break;
                        }
                        focus ^= 1;
                        break;
                    case 'Z':
                        // Shift-Tab: model cursor BACKWARD when the
                        // selected backend is expanded (reference
                        // tui.rs:288-295 forward=false), else panels
                        if (focus == 0 && expanded.count(sel_backend) &&
                            sel_backend < (int)s.backends.size() &&
                            !s.backends[sel_backend].models.empty()) {
                            const int len =
                                (int)s.backends[sel_backend].models.size();
                            auto it = mcur.find(sel_backend);
                            const int prev =
                                it == mcur.end()
                                    ? len - 1
                                    : (it->second + len - 1) % len;
                            mcur[sel_backend] = prev;
                            if (prev >= 5) unfolded.insert(sel_backend);
                            break;
                        }
                        focus ^= 1;
                        break;
                    case 'h':
                    case 'l':
                        focus ^= 1;
                        break;
                    case ' ':
                    case '\r':
                    case '\n':
                        if (focus == 0) {
                            if (expanded.count(sel_backend))
                                expanded.erase(sel_backend);
                            else
                                expanded.insert(sel_backend);
                        }
                        break;
                    case 'a':
                        show_all = !show_all;
                        break;
                    case 'j':
                        if (focus == 1)
                            sel_user = std::min<int>(
                                sel_user + 1, (int)s.users.size() - 1);
                        else
                            sel_backend = std::min<int>(
                                sel_backend + 1,
                                (int)s.backends.size() - 1);
                        break;
                    case 'k':
                        if (focus == 1)
                            sel_user = std::max(sel_user - 1, 0);
                        else
                            sel_backend = std::max(sel_backend - 1, 0);
                        break;
                    case 'L':
                    case 'U': {
                        // cursor-model direct control when the selected
                        // backend is expanded with a highlighted model
                        // (reference tui.rs:345-387); typed-name input
                        // mode otherwise
                        auto it = mcur.find(sel_backend);
                        if (focus == 0 && expanded.count(sel_backend) &&
                            it != mcur.end() &&
                            sel_backend < (int)s.backends.size() &&
                            it->second <
                                (int)s.backends[sel_backend].models
                                    .size()) {
                            ControlRequest req;
                            req.action = c == 'L' ? ControlAction::Load
                                                  : ControlAction::Unload;
                            req.model =
                                s.backends[sel_backend].models[it->second];
                            req.backend_idx = (size_t)sel_backend;
                            auto r = start_model_control(st, req);
                            status_msg =
                                r.http_status == 202
                                    ? (std::string(c == 'L' ? "load "
                                                            : "unload ") +
                                       req.model + " accepted")
                                    : r.body.get_str("error");
                            break;
                        }
                        input_mode = c;
                        input.clear();
                        break;
                    }
                    case 'r':
                        reload_model_config(st, server.config_path);
                        status_msg = "config reloaded";
                        break;
                    case 'p': {
                        std::lock_guard<std::mutex> g(st.prio_mu);
                        const std::string u = sel_name();
                        st.vip_user = st.vip_user == u ? "" : u;
                        if (st.vip_user == st.boost_user)
                            st.boost_user.clear();
                        break;
                    }
                    case 'b': {
                        std::lock_guard<std::mutex> g(st.prio_mu);
                        const std::string u = sel_name();
                        st.boost_user = st.boost_user == u ? "" : u;
                        if (st.boost_user == st.vip_user)
                            st.vip_user.clear();
                        break;
                    }
                    case 'x': {
                        const std::string u = sel_name();
                        if (!u.empty()) {
                            {
                                std::lock_guard<std::mutex> g(st.blocked_mu);
                                st.blocked_users.insert(u);
                            }
                            st.save_blocked();
                        }
                        break;
                    }
                    case 'X': {
                        const std::string u = sel_name();
                        std::string ip;
                        {
                            std::lock_guard<std::mutex> g(st.queues_mu);
                            auto it = st.user_ips.find(u);
                            if (it != st.user_ips.end()) ip = it->second;
                        }
                        if (!ip.empty()) {
                            {
                                std::lock_guard<std::mutex> g(st.blocked_mu);
                                st.blocked_ips.insert(ip);
                            }
                            st.save_blocked();
                        }
                        break;
                    }
                    case '?':
                        help = true;
                        break;
                    case 'u': {
                        {
                            std::lock_guard<std::mutex> g(st.blocked_mu);
                            st.blocked_users.clear();
                            st.blocked_ips.clear();
                        }
                        st.save_blocked();
                        status_msg = "unblocked all";
                        break;
                    }
                    default:
                        break;
                }
            }
        }
    }
}

}  // namespace omq
