#include "http.h"

#include <arpa/inet.h>
#include <netdb.h>
#include <netinet/in.h>
#include <netinet/tcp.h>
#include <poll.h>
#include <signal.h>
#include <sys/socket.h>
#include <unistd.h>

#include <cstring>
#include <sstream>

namespace omq {

static const size_t BODY_CAP = 1ull << 30;  // 1 GB (reference main.rs:302)

static bool ieq(const std::string& a, const std::string& b) {
    if (a.size() != b.size()) return false;
    for (size_t i = 0; i < a.size(); i++)
        if (std::tolower((unsigned char)a[i]) !=
            std::tolower((unsigned char)b[i]))
            return false;
    return true;
}

std::string HttpRequest::header(const std::string& name) const {
    for (const auto& [k, v] : headers)
        if (ieq(k, name)) return v;
    return "";
}
std::string HttpResponse::header(const std::string& name) const {
    for (const auto& [k, v] : headers)
        if (ieq(k, name)) return v;
    return "";
}

// --------------------------------------------------------------- HttpConn
bool HttpConn::write_all(const char* p, size_t n) {
    while (n > 0) {
        ssize_t w = ::send(fd_, p, n, MSG_NOSIGNAL);
        if (w <= 0) {
            alive_ = false;
            return false;
        }
        p += w;
        n -= (size_t)w;
    }
    return true;
}

static const char* status_text(int s) {
    switch (s) {
        case 200: return "OK";
        case 202: return "Accepted";
        case 400: return "Bad Request";
        case 401: return "Unauthorized";
        case 403: return "Forbidden";
        case 404: return "Not Found";
        case 409: return "Conflict";
        case 429: return "Too Many Requests";
        case 500: return "Internal Server Error";
        case 502: return "Bad Gateway";
        case 503: return "Service Unavailable";
        default: return "OK";
    }
}

bool HttpConn::send(
    int status,
    const std::vector<std::pair<std::string, std::string>>& headers,
    const std::string& body) {
    std::ostringstream h;
    h << "HTTP/1.1 " << status << " " << status_text(status) << "\r\n";
    bool has_ct = false;
    for (const auto& [k, v] : headers) {
        if (ieq(k, "Content-Length") || ieq(k, "Transfer-Encoding")) continue;
        if (ieq(k, "Content-Type")) has_ct = true;
        h << k << ": " << v << "\r\n";
    }
    if (!has_ct) h << "Content-Type: application/json\r\n";
    h << "Content-Length: " << body.size() << "\r\n\r\n";
    responded_ = true;
    const std::string head = h.str();
    return write_all(head.data(), head.size()) &&
           write_all(body.data(), body.size());
}

bool HttpConn::begin_stream(
    int status,
    const std::vector<std::pair<std::string, std::string>>& headers) {
    std::ostringstream h;
    h << "HTTP/1.1 " << status << " " << status_text(status) << "\r\n";
    for (const auto& [k, v] : headers) {
        if (ieq(k, "Content-Length") || ieq(k, "Transfer-Encoding")) continue;
        h << k << ": " << v << "\r\n";
    }
    h << "Transfer-Encoding: chunked\r\n\r\n";
    responded_ = true;
    streaming_ = true;
    const std::string head = h.str();
    return write_all(head.data(), head.size());
}

bool HttpConn::write_chunk(const std::string& data) {
    if (data.empty()) return alive_;
    char sz[20];
    int n = snprintf(sz, sizeof sz, "%zx\r\n", data.size());
    return write_all(sz, n) && write_all(data.data(), data.size()) &&
           write_all("\r\n", 2);
}

bool HttpConn::end_stream() {
    streaming_ = false;
    return write_all("0\r\n\r\n", 5);
}

// ------------------------------------------------------------- HttpServer
HttpServer::HttpServer(std::string host, int port, HttpHandler handler)
    : host_(std::move(host)), port_(port), handler_(std::move(handler)) {}

HttpServer::~HttpServer() { stop(); }

bool HttpServer::start(std::string* err) {
    signal(SIGPIPE, SIG_IGN);
    listen_fd_ = ::socket(AF_INET, SOCK_STREAM, 0);
    if (listen_fd_ < 0) {
        if (err) *err = "socket() failed";
        return false;
    }
    int one = 1;
    setsockopt(listen_fd_, SOL_SOCKET, SO_REUSEADDR, &one, sizeof one);
    sockaddr_in addr{};
    addr.sin_family = AF_INET;
    addr.sin_port = htons((uint16_t)port_);
    if (host_ == "0.0.0.0" || host_.empty())
        addr.sin_addr.s_addr = INADDR_ANY;
    else if (inet_pton(AF_INET, host_.c_str(), &addr.sin_addr) != 1) {
        if (err) *err = "bad host " + host_;
        return false;
    }
    if (bind(listen_fd_, (sockaddr*)&addr, sizeof addr) != 0) {
        if (err) *err = "bind " + host_ + ":" + std::to_string(port_) +
                        " failed: " + strerror(errno);
        return false;
    }
    socklen_t alen = sizeof addr;
    getsockname(listen_fd_, (sockaddr*)&addr, &alen);
    bound_port_ = ntohs(addr.sin_port);
    if (listen(listen_fd_, 256) != 0) {
        if (err) *err = "listen failed";
        return false;
    }
    accept_thread_ = std::thread([this] { accept_loop(); });
    return true;
}

void HttpServer::stop() {
    if (stopping_.exchange(true)) return;
    if (listen_fd_ >= 0) {
        ::shutdown(listen_fd_, SHUT_RDWR);
        ::close(listen_fd_);
        listen_fd_ = -1;
    }
    if (accept_thread_.joinable()) accept_thread_.join();
}

void HttpServer::accept_loop() {
    while (!stopping_) {
        sockaddr_in peer{};
        socklen_t plen = sizeof peer;
        int fd = ::accept(listen_fd_, (sockaddr*)&peer, &plen);
        if (fd < 0) {
            if (stopping_) break;
            continue;
        }
        char ip[64] = "unknown";
        inet_ntop(AF_INET, &peer.sin_addr, ip, sizeof ip);
        // shed load above the cap: closing immediately keeps accept
        // responsive for well-behaved clients (serving needs ≪1k
        // concurrent streams; half-open floods go far beyond that)
        if (live_conns_.load() >= 2048) {
            ::close(fd);
            continue;
        }
        live_conns_.fetch_add(1);
        try {
            std::thread(&HttpServer::handle_conn, this, fd,
                        std::string(ip))
                .detach();
        } catch (const std::system_error&) {
            live_conns_.fetch_sub(1);  // thread exhaustion: shed
            ::close(fd);
        }
    }
}

// read until delimiter or cap; returns false on EOF/error
static bool read_headers(int fd, std::string& buf, size_t& header_end) {
    char tmp[8192];
    while (true) {
        auto pos = buf.find("\r\n\r\n");
        if (pos != std::string::npos) {
            header_end = pos + 4;
            return true;
        }
        if (buf.size() > 1 << 20) return false;  // header cap 1 MB
        ssize_t n = ::recv(fd, tmp, sizeof tmp, 0);
        if (n <= 0) return false;
        buf.append(tmp, n);
    }
}

void HttpServer::handle_conn(int fd, std::string peer_ip) {
    struct Live {                      // RAII: cap accounting
        std::atomic<int>& n;
        ~Live() { n.fetch_sub(1); }
    } live{live_conns_};
    int one = 1;
    setsockopt(fd, IPPROTO_TCP, TCP_NODELAY, &one, sizeof one);
    // read timeout: a client trickling headers/body (slowloris) must not
    // pin a connection thread for long — this is a per-recv IDLE bound,
    // not a total-request bound, so legit slow bodies still flow; 30 s
    // (not 120) keeps the half-open thread population small under floods
    timeval rto{30, 0};
    setsockopt(fd, SOL_SOCKET, SO_RCVTIMEO, &rto, sizeof rto);
    std::string buf;
    while (!stopping_) {
        size_t hend = 0;
        if (!read_headers(fd, buf, hend)) break;
        // parse request line + headers
        HttpRequest req;
        req.client_ip = peer_ip;
        {
            std::istringstream hs(buf.substr(0, hend - 2));
            std::string line;
            std::getline(hs, line);
            if (!line.empty() && line.back() == '\r') line.pop_back();
            std::istringstream rl(line);
            std::string target, ver;
            rl >> req.method >> target >> ver;
            auto qpos = target.find('?');
            req.path = target.substr(0, qpos);
            if (qpos != std::string::npos) req.query = target.substr(qpos + 1);
            while (std::getline(hs, line)) {
                if (!line.empty() && line.back() == '\r') line.pop_back();
                if (line.empty()) continue;
                auto c = line.find(':');
                if (c == std::string::npos) continue;
                std::string k = line.substr(0, c);
                size_t vs = c + 1;
                while (vs < line.size() && line[vs] == ' ') vs++;
                req.headers.emplace_back(k, line.substr(vs));
            }
        }
        if (req.method.empty() || req.path.empty()) break;
        buf.erase(0, hend);
        // body: Content-Length, or chunked transfer coding (axum parity —
        // reference clients may stream request bodies)
        auto fill = [&](size_t need) -> bool {   // grow buf to >= need
            while (buf.size() < need) {
                char tmp[65536];
                ssize_t n = ::recv(fd, tmp, sizeof tmp, 0);
                if (n <= 0) return false;
                buf.append(tmp, n);
            }
            return true;
        };
        if (ieq(req.header("Transfer-Encoding"), "chunked")) {
            std::string body;
            bool ok = true;
            for (;;) {
                size_t eol;
                while ((eol = buf.find("\r\n")) == std::string::npos) {
                    // size line is tiny; cap so a client streaming
                    // CRLF-less bytes can't grow the buffer unboundedly
                    if (buf.size() > 4096 || !fill(buf.size() + 1)) {
                        ok = false;
                        break;
                    }
                }
                if (!ok) break;
                const size_t csz = strtoull(buf.c_str(), nullptr, 16);
                buf.erase(0, eol + 2);
                if (csz > BODY_CAP || body.size() + csz > BODY_CAP) {
                    ok = false;  // csz check first: near-SIZE_MAX csz
                    break;       // would overflow the sum
                }
                if (csz == 0) {
                    // consume optional trailers up to the blank line
                    size_t tend;
                    while ((tend = buf.find("\r\n")) == std::string::npos) {
                        if (!fill(buf.size() + 1)) { ok = false; break; }
                    }
                    if (ok) buf.erase(0, tend + 2);
                    break;
                }
                if (!fill(csz + 2)) { ok = false; break; }
                body.append(buf, 0, csz);
                buf.erase(0, csz + 2);   // chunk + CRLF
            }
            if (!ok) {
                HttpConn c(fd);
                c.send(400, {}, "{\"error\":\"bad chunked body\"}");
                break;
            }
            req.body = std::move(body);
        } else {
            size_t clen = 0;
            const std::string cl = req.header("Content-Length");
            if (!cl.empty()) clen = strtoull(cl.c_str(), nullptr, 10);
            if (clen > BODY_CAP) {
                HttpConn c(fd);
                c.send(400, {}, "{\"error\":\"body too large\"}");
                break;
            }
            if (!fill(clen)) { ::close(fd); return; }
            req.body = buf.substr(0, clen);
            buf.erase(0, clen);
        }

        HttpConn conn(fd);
        handler_(req, conn);
        if (!conn.responded())
            conn.send(500, {}, "{\"error\":\"handler sent no response\"}");
        if (!conn.alive()) break;
        const std::string ka = req.header("Connection");
        if (ieq(ka, "close")) break;
    }
    ::close(fd);
}

// ----------------------------------------------------------------- client
static bool parse_url(const std::string& url, std::string& host, int& port,
                      std::string& path) {
    std::string rest = url;
    if (rest.rfind("http://", 0) == 0) rest = rest.substr(7);
    auto slash = rest.find('/');
    std::string hostport = rest.substr(0, slash);
    path = slash == std::string::npos ? "/" : rest.substr(slash);
    auto colon = hostport.find(':');
    port = 80;
    if (colon != std::string::npos) {
        port = atoi(hostport.c_str() + colon + 1);
        host = hostport.substr(0, colon);
    } else {
        host = hostport;
    }
    return !host.empty();
}

static int connect_to(const std::string& host, int port, int timeout_ms,
                      std::string* err) {
    addrinfo hints{}, *res = nullptr;
    hints.ai_family = AF_INET;
    hints.ai_socktype = SOCK_STREAM;
    if (getaddrinfo(host.c_str(), std::to_string(port).c_str(), &hints,
                    &res) != 0 || !res) {
        *err = "resolve failed: " + host;
        return -1;
    }
    int fd = ::socket(res->ai_family, SOCK_STREAM, 0);
    if (fd < 0) {
        freeaddrinfo(res);
        *err = "socket failed";
        return -1;
    }
    timeval tv{timeout_ms / 1000, (timeout_ms % 1000) * 1000};
    setsockopt(fd, SOL_SOCKET, SO_RCVTIMEO, &tv, sizeof tv);
    setsockopt(fd, SOL_SOCKET, SO_SNDTIMEO, &tv, sizeof tv);
    if (connect(fd, res->ai_addr, res->ai_addrlen) != 0) {
        *err = "connect failed: " + host + ":" + std::to_string(port);
        ::close(fd);
        freeaddrinfo(res);
        return -1;
    }
    freeaddrinfo(res);
    int one = 1;
    setsockopt(fd, IPPROTO_TCP, TCP_NODELAY, &one, sizeof one);
    return fd;
}

HttpResponse http_request(
    const std::string& method, const std::string& url,
    const std::vector<std::pair<std::string, std::string>>& headers,
    const std::string& body, int timeout_ms,
    const std::function<bool(const char*, size_t)>& on_chunk,
    const std::function<void(int, const std::vector<std::pair<std::string,
                             std::string>>&)>& on_status) {
    HttpResponse resp;
    std::string host, path;
    int port;
    if (!parse_url(url, host, port, path)) {
        resp.error = "bad url: " + url;
        return resp;
    }
    int fd = connect_to(host, port, timeout_ms, &resp.error);
    if (fd < 0) return resp;

    std::ostringstream req;
    req << method << " " << path << " HTTP/1.1\r\n";
    req << "Host: " << host << ":" << port << "\r\n";
    bool has_ct = false;
    for (const auto& [k, v] : headers) {
        if (ieq(k, "Host") || ieq(k, "Content-Length") ||
            ieq(k, "Transfer-Encoding") || ieq(k, "Connection"))
            continue;
        if (ieq(k, "Content-Type")) has_ct = true;
        req << k << ": " << v << "\r\n";
    }
    if (!body.empty() && !has_ct) req << "Content-Type: application/json\r\n";
    req << "Content-Length: " << body.size() << "\r\n";
    req << "Connection: close\r\n\r\n";
    const std::string head = req.str();
    auto send_all = [&](const char* p, size_t n) {
        while (n) {
            ssize_t w = ::send(fd, p, n, MSG_NOSIGNAL);
            if (w <= 0) return false;
            p += w;
            n -= (size_t)w;
        }
        return true;
    };
    if (!send_all(head.data(), head.size()) ||
        !send_all(body.data(), body.size())) {
        resp.error = "send failed";
        ::close(fd);
        return resp;
    }

    // ---- read response ----
    std::string buf;
    size_t hend = 0;
    if (!read_headers(fd, buf, hend)) {
        resp.error = "no response";
        ::close(fd);
        return resp;
    }
    {
        std::istringstream hs(buf.substr(0, hend - 2));
        std::string line;
        std::getline(hs, line);
        // "HTTP/1.1 200 OK"
        auto sp = line.find(' ');
        resp.status = sp == std::string::npos ? -1
                                              : atoi(line.c_str() + sp + 1);
        while (std::getline(hs, line)) {
            if (!line.empty() && line.back() == '\r') line.pop_back();
            auto c = line.find(':');
            if (c == std::string::npos) continue;
            size_t vs = c + 1;
            while (vs < line.size() && line[vs] == ' ') vs++;
            resp.headers.emplace_back(line.substr(0, c), line.substr(vs));
        }
    }
    buf.erase(0, hend);
    if (on_status) on_status(resp.status, resp.headers);

    auto deliver = [&](const char* p, size_t n) -> bool {
        if (n == 0) return true;
        if (on_chunk) return on_chunk(p, n);
        resp.body.append(p, n);
        return resp.body.size() <= BODY_CAP;
    };

    const std::string te = resp.header("Transfer-Encoding");
    if (ieq(te, "chunked")) {
        // de-chunk
        std::string pending = buf;
        size_t off = 0;
        auto need = [&](size_t n) -> bool {
            while (pending.size() - off < n) {
                char tmp[65536];
                ssize_t r = ::recv(fd, tmp, sizeof tmp, 0);
                if (r <= 0) return false;
                pending.append(tmp, r);
                if (off > (1 << 20)) {
                    pending.erase(0, off);
                    off = 0;
                }
            }
            return true;
        };
        while (true) {
            // read chunk-size line (bounded: a backend streaming bytes
            // with no CRLF must not grow the buffer forever)
            size_t eol;
            while ((eol = pending.find("\r\n", off)) == std::string::npos) {
                if (pending.size() - off > 4096 ||
                    !need(pending.size() - off + 1)) {
                    eol = std::string::npos;
                    break;
                }
            }
            if (eol == std::string::npos) break;
            const size_t csz = strtoull(pending.c_str() + off, nullptr, 16);
            off = eol + 2;
            if (csz == 0) break;
            // reject insane sizes: csz near SIZE_MAX would overflow
            // csz + 2 below and deliver() would read out of bounds
            if (csz > BODY_CAP) break;
            if (!need(csz + 2)) break;
            if (!deliver(pending.data() + off, csz)) break;
            off += csz + 2;  // skip data + CRLF
        }
    } else {
        const std::string cl = resp.header("Content-Length");
        size_t want = cl.empty() ? SIZE_MAX : strtoull(cl.c_str(), nullptr, 10);
        size_t got = buf.size();
        bool ok = deliver(buf.data(), buf.size());
        while (ok && got < want) {
            char tmp[65536];
            ssize_t r = ::recv(fd, tmp, sizeof tmp, 0);
            if (r <= 0) break;
            got += (size_t)r;
            ok = deliver(tmp, (size_t)r);
        }
    }
    ::close(fd);
    return resp;
}

}  // namespace omq
