// Minimal dependency-free JSON for the dispatcher (parse + serialize).
// Covers what the wire contract needs: extracting "model" from request
// bodies (reference src/dispatcher.rs:876-882), admin API bodies
// (src/control.rs:1160-1350), probe responses (src/control.rs:128-334) and
// blocklist persistence (src/dispatcher.rs:279-296).
#pragma once

#include <map>
#include <memory>
#include <optional>
#include <string>
#include <vector>

namespace omq {

class Json {
public:
    enum class Type { Null, Bool, Num, Str, Arr, Obj };
    Type type = Type::Null;
    bool b = false;
    double num = 0;
    std::string str;
    std::vector<Json> arr;
    std::vector<std::pair<std::string, Json>> obj;  // insertion-ordered

    Json() = default;
    static Json null() { return Json(); }
    static Json boolean(bool v) { Json j; j.type = Type::Bool; j.b = v; return j; }
    static Json number(double v) { Json j; j.type = Type::Num; j.num = v; return j; }
    static Json string(std::string v) { Json j; j.type = Type::Str; j.str = std::move(v); return j; }
    static Json array() { Json j; j.type = Type::Arr; return j; }
    static Json object() { Json j; j.type = Type::Obj; return j; }

    bool is_null() const { return type == Type::Null; }
    bool is_obj() const { return type == Type::Obj; }
    bool is_arr() const { return type == Type::Arr; }
    bool is_str() const { return type == Type::Str; }
    bool is_num() const { return type == Type::Num; }

    const Json* find(const std::string& key) const {
        if (type != Type::Obj) return nullptr;
        for (const auto& [k, v] : obj)
            if (k == key) return &v;
        return nullptr;
    }
    Json& set(const std::string& key, Json v) {
        for (auto& [k, val] : obj)
            if (k == key) { val = std::move(v); return val; }
        obj.emplace_back(key, std::move(v));
        return obj.back().second;
    }
    std::string get_str(const std::string& key,
                        const std::string& dflt = "") const {
        const Json* j = find(key);
        return (j && j->type == Type::Str) ? j->str : dflt;
    }
    double get_num(const std::string& key, double dflt = 0) const {
        const Json* j = find(key);
        return (j && j->type == Type::Num) ? j->num : dflt;
    }
    bool get_bool(const std::string& key, bool dflt = false) const {
        const Json* j = find(key);
        return (j && j->type == Type::Bool) ? j->b : dflt;
    }

    std::string dump() const;
    // returns nullopt on malformed input
    static std::optional<Json> parse(const std::string& text);
};

}  // namespace omq
