// C10/C12 implementation — see matching.h for the semantics contract
// (reference src/dispatcher.rs:355-397, src/control.rs:450-510).
#include "matching.h"

#include "core.h"

#include <algorithm>

namespace omq {

std::string lower(const std::string& s) {
    std::string out = s;
    std::transform(out.begin(), out.end(), out.begin(),
                   [](unsigned char c) { return std::tolower(c); });
    return out;
}

std::string strip_tag(const std::string& s) {
    auto pos = s.find(':');
    return pos == std::string::npos ? s : s.substr(0, pos);
}

static bool contains_ci(const std::string& haystack,
                        const std::string& needle) {
    if (needle.empty()) return false;
    return lower(haystack).find(lower(needle)) != std::string::npos;
}

bool smart_model_match_one(const std::string& requested,
                           const std::string& available) {
    if (requested == available) return true;
    const std::string r = lower(strip_tag(requested));
    const std::string a = lower(strip_tag(available));
    return !r.empty() && r == a;
}

bool smart_model_match(const std::string& requested,
                       const std::vector<std::string>& available) {
    for (const auto& a : available)
        if (smart_model_match_one(requested, a)) return true;
    return false;
}

// substring in either direction, case-insensitive — catches publisher
// prefixes and quant suffixes ("unsloth/qwen3.8-27b@q8_0" vs "qwen3.8-27b")
bool fuzzy_model_match(const std::string& requested,
                       const std::vector<std::string>& available) {
    const std::string rb = strip_tag(requested);
    if (rb.empty()) return false;
    for (const auto& a : available) {
        const std::string ab = strip_tag(a);
        if (contains_ci(ab, rb) || contains_ci(rb, ab)) return true;
    }
    return false;
}

bool model_routable(const std::string& requested,
                    const std::vector<std::string>& available) {
    if (requested.empty()) return true;  // no model constraint
    return smart_model_match(requested, available) ||
           fuzzy_model_match(requested, available);
}

std::optional<std::string> resolve_model_name(
    const std::string& requested,
    const std::vector<std::string>& available,
    const std::map<std::string, std::string>& native_display) {
    std::string req = requested;
    // trim whitespace
    const auto b = req.find_first_not_of(" \t\r\n");
    if (b == std::string::npos) return std::nullopt;
    req = req.substr(b, req.find_last_not_of(" \t\r\n") - b + 1);
    if (req.empty()) return std::nullopt;

    // 1. exact
    for (const auto& a : available)
        if (a == req) return a;

    // 2. smart (tag/case-normalized) — sorted for determinism
    std::vector<std::string> sorted = available;
    std::sort(sorted.begin(), sorted.end());
    for (const auto& a : sorted)
        if (smart_model_match_one(req, a)) return a;

    // 3. unique case-insensitive substring; ambiguous => refuse to guess
    std::vector<std::string> subs;
    for (const auto& a : sorted)
        if (contains_ci(a, req)) subs.push_back(a);
    if (subs.size() == 1) return subs[0];
    if (subs.size() > 1) return std::nullopt;

    // 4. LM Studio native keys / display names: exact, then unique substring
    for (const auto& [key, disp] : native_display)
        if (key == req || disp == req) return key;
    std::vector<std::string> nsubs;
    for (const auto& [key, disp] : native_display)
        if (contains_ci(key, req) || contains_ci(disp, req))
            nsubs.push_back(key);
    if (nsubs.size() == 1) return nsubs[0];
    return std::nullopt;
}

const char* api_type_name(ApiType t) {
    switch (t) {
        case ApiType::Ollama: return "ollama";
        case ApiType::OpenAi: return "openai";
        case ApiType::Both: return "both";
        default: return "unknown";
    }
}

}  // namespace omq
