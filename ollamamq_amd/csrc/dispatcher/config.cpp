// C2: appconf.yaml loader — native mini-YAML for the three-section schema
// (backends / settings / models; reference src/config.rs:36-135 and
// appconf.yaml.example).  Subset parser: 2-level nesting, "- " list items,
// "key: value" scalars, inline comments, quoted strings.  Validation:
// keep_alive >= -1 is an error below -1; a missing file is fine (defaults).
#include "config.h"

#include <fstream>
#include <sstream>

namespace omq {

static std::string strip(const std::string& s) {
    const auto b = s.find_first_not_of(" \t\r\n");
    if (b == std::string::npos) return "";
    const auto e = s.find_last_not_of(" \t\r\n");
    return s.substr(b, e - b + 1);
}

static std::string unquote(std::string v) {
    v = strip(v);
    if (v.size() >= 2 &&
        ((v.front() == '"' && v.back() == '"') ||
         (v.front() == '\'' && v.back() == '\'')))
        v = v.substr(1, v.size() - 2);
    return v;
}

// strip an inline comment (a # not inside quotes)
static std::string decomment(const std::string& line) {
    bool in_s = false, in_d = false;
    for (size_t i = 0; i < line.size(); i++) {
        const char c = line[i];
        if (c == '"' && !in_s) in_d = !in_d;
        else if (c == '\'' && !in_d) in_s = !in_s;
        else if (c == '#' && !in_s && !in_d)
            return line.substr(0, i);
    }
    return line;
}

bool load_config(const std::string& path, AppConfig* out, std::string* err) {
    std::ifstream f(path);
    if (!f) {
        // missing file is fine: defaults (reference main.rs:149-158)
        return true;
    }
    std::string line;
    std::string section;            // backends | settings | models
    ModelConfigEntry* cur_model = nullptr;
    bool in_model_backends = false;
    size_t model_indent = SIZE_MAX;  // indent of the "- name:" model items

    auto parse_i64 = [](const std::string& v, int64_t dflt) {
        try {
            return (int64_t)std::stoll(strip(v));
        } catch (...) {
            return dflt;
        }
    };

    // The parser accepts exactly the documented 3-section subset
    // (appconf.yaml.example).  Anything it cannot represent is a HARD
    // error with a line number — a config that silently parses to
    // something else is worse than one that refuses to load (ADVICE r01:
    // "parses the shipped examples; any real-world appconf fails
    // silently or oddly").
    auto unsupported = [&](int ln, const std::string& what,
                           std::string* e) {
        if (e)
            *e = "line " + std::to_string(ln) + ": unsupported YAML (" +
                 what + ") — appconf.yaml supports plain 2-level " +
                 "mappings, '- ' lists and quoted scalars only " +
                 "(see appconf.yaml.example)";
        return false;
    };

    int lineno = 0;
    while (std::getline(f, line)) {
        lineno++;
        line = decomment(line);
        const std::string t = strip(line);
        if (t.empty()) continue;
        if (t == "---" || t == "...") continue;       // document markers
        // constructs the subset cannot represent -> refuse loudly
        if (t[0] == '&' || t[0] == '*')
            return unsupported(lineno, "anchor/alias", err);
        if (t[0] == '?')
            return unsupported(lineno, "complex mapping key", err);
        if (t.back() == '|' || t.back() == '>')
            return unsupported(lineno, "block scalar", err);
        {   // flow collections / anchors as values — except the empty
            // flow collections, which mean "no entries" and are harmless
            std::string item = t;
            if (item.rfind("- ", 0) == 0) item = strip(item.substr(2));
            if (!item.empty() && (item[0] == '&' || item[0] == '*' ||
                                  item[0] == '{' ||
                                  (item[0] == '[' && item != "[]")))
                return unsupported(lineno,
                                   "anchor/alias/flow list item", err);
            auto c = item.find(':');
            std::string v = c == std::string::npos
                                ? "" : strip(item.substr(c + 1));
            if ((!v.empty() && (v[0] == '{' || v[0] == '[')) &&
                v != "[]" && v != "{}")
                return unsupported(lineno, "flow collection", err);
            if (!v.empty() && (v[0] == '&' || v[0] == '*'))
                return unsupported(lineno, "anchor/alias value", err);
        }
        if (t.rfind("<<", 0) == 0)
            return unsupported(lineno, "merge key", err);
        if (line[0] == '\t')
            return unsupported(lineno, "tab indentation", err);
        const size_t indent = line.find_first_not_of(" \t");

        if (indent == 0 && t.back() == ':') {
            section = t.substr(0, t.size() - 1);
            cur_model = nullptr;
            in_model_backends = false;
            continue;
        }
        if (section == "backends") {
            if (t.rfind("- ", 0) == 0)
                out->backends.push_back(unquote(t.substr(2)));
            continue;
        }
        if (section == "settings") {
            auto c = t.find(':');
            if (c == std::string::npos) continue;
            const std::string key = strip(t.substr(0, c));
            const std::string val = unquote(t.substr(c + 1));
            if (key == "port") out->settings.port = (int)parse_i64(val, 11435);
            else if (key == "host") out->settings.host = val;
            else if (key == "timeout")
                out->settings.timeout_s = parse_i64(val, 300);
            else if (key == "load_keep_alive")
                out->settings.load_keep_alive_s = parse_i64(val, 86400);
            else if (key == "allow_all_routes")
                out->settings.allow_all_routes = (val == "true" || val == "1");
            else if (key == "stuck_timeout")
                out->settings.stuck_timeout_s = parse_i64(val, 60);
            continue;
        }
        if (section == "models") {
            const bool dash = t.rfind("- ", 0) == 0;
            if (dash && model_indent == SIZE_MAX) model_indent = indent;
            if (dash && indent == model_indent &&
                t.find(':') != std::string::npos) {
                // new model entry: "- name: x" or "- key: v"
                out->models.emplace_back();
                cur_model = &out->models.back();
                in_model_backends = false;
                const std::string rest = t.substr(2);
                auto c = rest.find(':');
                const std::string key = strip(rest.substr(0, c));
                const std::string val = unquote(rest.substr(c + 1));
                if (key == "name") cur_model->name = val;
                continue;
            }
            if (!cur_model) continue;
            if (t.rfind("- ", 0) == 0 && in_model_backends) {
                cur_model->backends.push_back(unquote(t.substr(2)));
                continue;
            }
            auto c = t.find(':');
            if (c == std::string::npos) continue;
            const std::string key = strip(t.substr(0, c));
            const std::string val = unquote(t.substr(c + 1));
            in_model_backends = false;
            if (key == "name") cur_model->name = val;
            else if (key == "identifier") cur_model->identifier = val;
            else if (key == "max_ctx")
                cur_model->max_ctx = parse_i64(val, 0);
            else if (key == "keep_alive") {
                cur_model->keep_alive = parse_i64(val, 86400);
                if (cur_model->keep_alive < -1) {
                    if (err)
                        *err = "line " + std::to_string(lineno) +
                               ": keep_alive must be >= -1";
                    return false;
                }
            } else if (key == "max_concurrent_requests")
                cur_model->max_concurrent_requests =
                    (int)parse_i64(val, 1);
            else if (key == "backends")
                in_model_backends = true;   // list follows
            continue;
        }
    }
    return true;
}

}  // namespace omq

namespace omq {

std::string normalize_backend_url(std::string url) {
    url = [](std::string s) {
        const auto b = s.find_first_not_of(" \t\r\n");
        if (b == std::string::npos) return std::string();
        const auto e = s.find_last_not_of(" \t\r\n");
        return s.substr(b, e - b + 1);
    }(url);
    while (!url.empty() && url.back() == '/') url.pop_back();
    if (!url.empty() && url.rfind("http://", 0) != 0 &&
        url.rfind("https://", 0) != 0)
        url = "http://" + url;
    return url;
}

}  // namespace omq
