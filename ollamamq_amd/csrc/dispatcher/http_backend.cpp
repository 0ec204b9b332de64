#include "http_backend.h"

#include "http.h"
#include "json.h"
#include "matching.h"

namespace omq {

static const int PROBE_TIMEOUT_MS = 4000;

// Probe sequence (reference src/control.rs:128-334):
//   /api/tags -> Ollama? + available;  /api/ps -> loaded (+ctx);
//   /v1/models -> OpenAI? + available;  /api/v1/models -> LM Studio native;
//   / -> online fallback.  Endpoints in `skip` are skipped unless both
//   primary probes would be skipped.
ProbeResult HttpBackend::probe(const std::set<std::string>& skip) {
    ProbeResult r;
    auto skipped = [&](const std::string& ep) {
        return skip.count(ep) > 0;
    };
    bool skip_tags = skipped("/api/tags");
    bool skip_v1 = skipped("/v1/models");
    if (skip_tags && skip_v1) skip_tags = skip_v1 = false;  // never both

    auto get = [&](const std::string& ep) {
        return http_request("GET", url_ + ep, {}, "", PROBE_TIMEOUT_MS);
    };

    if (!skip_tags) {
        auto resp = get("/api/tags");
        if (resp.status == 200) {
            auto j = Json::parse(resp.body);
            if (j) {
                r.online = true;
                r.api_type = ApiType::Ollama;
                r.good_endpoints.push_back("/api/tags");
                if (const Json* models = j->find("models"))
                    for (const auto& m : models->arr) {
                        const std::string name = m.get_str("name");
                        if (!name.empty()) r.available_models.push_back(name);
                    }
            }
        } else if (resp.status > 0) {
            r.online = true;  // talking HTTP, endpoint rejected
            r.bad_endpoints.push_back("/api/tags");
        }
    }
    if (r.api_type == ApiType::Ollama && !skipped("/api/ps")) {
        auto resp = get("/api/ps");
        if (resp.status == 200) {
            auto j = Json::parse(resp.body);
            if (j) {
                r.good_endpoints.push_back("/api/ps");
                if (const Json* models = j->find("models"))
                    for (const auto& m : models->arr) {
                        const std::string name = m.get_str("name");
                        if (name.empty()) continue;
                        r.loaded_models.push_back(name);
                        double ctx = m.get_num("context_length", 0);
                        if (ctx <= 0)
                            ctx = m.get_num("num_ctx", 0);
                        if (ctx > 0) r.loaded_ctx[name] = (int64_t)ctx;
                    }
            }
        } else if (resp.status > 0) {
            r.bad_endpoints.push_back("/api/ps");
        }
    }
    if (!skip_v1) {
        auto resp = get("/v1/models");
        if (resp.status == 200) {
            auto j = Json::parse(resp.body);
            if (j) {
                r.online = true;
                r.api_type = (r.api_type == ApiType::Ollama)
                                 ? ApiType::Both
                                 : ApiType::OpenAi;
                r.good_endpoints.push_back("/v1/models");
                if (const Json* data = j->find("data"))
                    for (const auto& m : data->arr) {
                        const std::string id = m.get_str("id");
                        if (id.empty()) continue;
                        bool dup = false;
                        for (const auto& a : r.available_models)
                            if (a == id) dup = true;
                        if (!dup) r.available_models.push_back(id);
                    }
            }
        } else if (resp.status > 0) {
            r.online = true;
            r.bad_endpoints.push_back("/v1/models");
        }
    }
    // LM Studio native enumeration (reference src/control.rs:354-448)
    if (r.api_type == ApiType::OpenAi && !skipped("/api/v1/models")) {
        auto resp = get("/api/v1/models");
        if (resp.status == 200) {
            auto j = Json::parse(resp.body);
            if (j && j->find("models")) {
                r.lmstudio = true;
                r.good_endpoints.push_back("/api/v1/models");
                for (const auto& m : j->find("models")->arr) {
                    std::string key = m.get_str("key");
                    if (key.empty()) key = m.get_str("id");
                    if (key.empty()) continue;
                    const std::string disp = m.get_str("display_name");
                    if (!disp.empty()) r.native_display[key] = disp;
                    if (const Json* li = m.find("loaded_instances")) {
                        if (!li->arr.empty()) {
                            const std::string iid = li->arr[0].get_str("id");
                            r.native_instance[key] =
                                iid.empty() ? key : iid;
                            bool dup = false;
                            for (const auto& l : r.loaded_models)
                                if (l == key) dup = true;
                            if (!dup) r.loaded_models.push_back(key);
                            double ctx = 0;
                            if (const Json* cfg = li->arr[0].find("config"))
                                ctx = cfg->get_num("context_length", 0);
                            if (ctx > 0) r.loaded_ctx[key] = (int64_t)ctx;
                        }
                    }
                }
            }
        } else if (resp.status > 0) {
            r.bad_endpoints.push_back("/api/v1/models");
        }
    }
    if (!r.online) {
        auto resp = get("/");
        if (resp.status > 0) r.online = true;
    }
    return r;
}

// Streaming proxy executor (reference src/dispatcher.rs:742-778): forward
// the task, strip Transfer-Encoding/Content-Length from the response
// headers, pump chunks into the responder channel.
int HttpBackend::execute(const Task& task) {
    auto resp_ch = task.resp;
    int status_seen = -1;
    auto on_status = [&](int status,
                         const std::vector<std::pair<std::string,
                                                     std::string>>& hdrs) {
        status_seen = status;
        std::vector<std::pair<std::string, std::string>> fwd;
        for (const auto& [k, v] : hdrs) {
            const std::string lk = lower(k);
            if (lk == "transfer-encoding" || lk == "content-length" ||
                lk == "connection")
                continue;
            fwd.emplace_back(k, v);
        }
        if (resp_ch) resp_ch->send_status(status, std::move(fwd));
    };
    auto on_chunk = [&](const char* p, size_t n) -> bool {
        if (!resp_ch) return true;
        return resp_ch->send_chunk(std::string(p, n));
    };
    auto r = http_request(task.method, url_ + task.path +
                              (task.query.empty() ? "" : "?" + task.query),
                          task.headers, task.body,
                          (int)(timeout_s_ * 1000), on_chunk, on_status);
    if (resp_ch) {
        if (status_seen < 0) {
            resp_ch->send_status(502, {{"Content-Type", "application/json"}});
            resp_ch->send_chunk("{\"error\":\"Backend error: " + r.error +
                                "\"}");
        }
        resp_ch->finish();
    }
    return status_seen;
}

static std::string err_body(const HttpResponse& r) {
    // reference err_from_response (control.rs:939-959): extract "error"
    // from the body, truncate to 300 chars
    std::string msg;
    auto j = Json::parse(r.body);
    if (j) {
        msg = j->get_str("error");
        if (msg.empty() && j->find("error") && j->find("error")->is_obj())
            msg = j->find("error")->get_str("message");
    }
    if (msg.empty()) msg = r.body;
    if (msg.size() > 300) msg = msg.substr(0, 300);
    if (msg.empty()) msg = "HTTP " + std::to_string(r.status);
    return msg;
}

// Ollama load: empty-prompt /api/generate with long keep_alive (+num_ctx);
// LM Studio load: native /api/v1/models/load {model, context_length}
// (reference src/control.rs:762-821)
std::string HttpBackend::load_model(const std::string& model,
                                    int64_t num_ctx, int64_t keep_alive,
                                    const BackendStatus& st) {
    const int64_t ctl_timeout_ms =
        std::max<int64_t>(timeout_s_, 600) * 1000;
    if (st.lmstudio) {
        Json b = Json::object();
        b.set("model", Json::string(model));
        if (num_ctx > 0) b.set("context_length", Json::number((double)num_ctx));
        auto r = http_request("POST", url_ + "/api/v1/models/load", {},
                              b.dump(), (int)ctl_timeout_ms);
        if (r.status == 404)
            return "LM Studio native API not found (needs LM Studio >= "
                   "0.3.6)";
        if (r.status != 200) return err_body(r);
        return "";
    }
    Json b = Json::object();
    b.set("model", Json::string(model));
    b.set("keep_alive",
          Json::number((double)(keep_alive != 0 ? keep_alive
                                                : load_keep_alive_)));
    if (num_ctx > 0) {
        Json opts = Json::object();
        opts.set("num_ctx", Json::number((double)num_ctx));
        b.set("options", std::move(opts));
    }
    auto r = http_request("POST", url_ + "/api/generate", {}, b.dump(),
                          (int)ctl_timeout_ms);
    if (r.status != 200) return err_body(r);
    return "";
}

// Ollama unload: /api/generate keep_alive:0, expects done_reason=="unload";
// LM Studio: native /api/v1/models/unload {instance_id} (instance id from
// probe cache, else fresh fetch) (reference src/control.rs:823-936)
std::string HttpBackend::unload_model(const std::string& model,
                                      const BackendStatus& st) {
    const int64_t ctl_timeout_ms =
        std::max<int64_t>(timeout_s_, 600) * 1000;
    if (st.lmstudio) {
        std::string iid;
        auto it = st.native_instance.find(model);
        if (it != st.native_instance.end()) iid = it->second;
        if (iid.empty()) {
            // fresh fetch (control.rs:823-865)
            auto r = http_request("GET", url_ + "/api/v1/models", {}, "",
                                  PROBE_TIMEOUT_MS);
            auto j = r.status == 200 ? Json::parse(r.body) : std::nullopt;
            if (j && j->find("models"))
                for (const auto& m : j->find("models")->arr) {
                    std::string key = m.get_str("key");
                    if (key.empty()) key = m.get_str("id");
                    if (key != model) continue;
                    if (const Json* li = m.find("loaded_instances"))
                        if (!li->arr.empty())
                            iid = li->arr[0].get_str("id");
                }
        }
        if (iid.empty()) iid = model;
        Json b = Json::object();
        b.set("instance_id", Json::string(iid));
        auto r = http_request("POST", url_ + "/api/v1/models/unload", {},
                              b.dump(), (int)ctl_timeout_ms);
        if (r.status == 404)
            return "LM Studio native API not found (needs LM Studio >= "
                   "0.3.6)";
        if (r.status != 200) return err_body(r);
        return "";
    }
    Json b = Json::object();
    b.set("model", Json::string(model));
    b.set("keep_alive", Json::number(0));
    auto r = http_request("POST", url_ + "/api/generate", {}, b.dump(),
                          (int)ctl_timeout_ms);
    if (r.status != 200) return err_body(r);
    auto j = Json::parse(r.body);
    if (!j || j->get_str("done_reason") != "unload")
        return "backend did not confirm unload";
    return "";
}

bool HttpBackend::supports_control(const BackendStatus& st) const {
    // Ollama always; LM Studio via native API; plain OpenAI: no
    return st.api_type == ApiType::Ollama || st.api_type == ApiType::Both ||
           st.lmstudio;
}

}  // namespace omq
