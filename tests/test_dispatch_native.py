"""Native C++ dispatcher core: matching, resolution, scheduler semantics.

Mirrors the reference's unit tests (reference src/dispatcher.rs:942-984
matching; src/control.rs:1385-1459 resolution; scheduler semantics per
src/dispatcher.rs:494-697) against OUR compiled C++ — the same .so the
server binary links.
"""
import subprocess
import sys

import pytest

from ollamamq_amd.dispatch import load

d = None


def setup_module():
    global d
    try:
        d = load()
    except RuntimeError:
        subprocess.run([sys.executable, "-m", "ollamamq_amd.build"],
                       check=True)
        d = load()


# ---------------------------------------------------------------- matching
def test_smart_match_exact_and_tag():
    assert d.smart_model_match_one("llama3:latest", "llama3:latest")
    assert d.smart_model_match_one("llama3", "llama3:latest")
    assert d.smart_model_match_one("LLAMA3", "llama3:8b")
    assert not d.smart_model_match_one("llama3.1", "llama3:latest")


def test_fuzzy_publisher_and_quant_suffix():
    avail = ["unsloth/qwen3.8-27b@q8_0"]
    assert d.fuzzy_model_match("qwen3.8-27b", avail)
    assert d.fuzzy_model_match("unsloth/qwen3.8-27b@q8_0", ["qwen3.8-27b"])
    assert not d.fuzzy_model_match("mistral", avail)


def test_routable_strict_beats_fuzzy():
    avail = ["llama3:latest", "qwen2:7b"]
    assert d.model_routable("llama3", avail)
    assert d.model_routable("qwen2", avail)
    assert not d.model_routable("gemma", avail)
    assert d.model_routable("", avail)  # no model constraint


def test_resolve_exact_smart_substring_ambiguous():
    avail = ["qwen2.5:7b", "llama3:latest", "llama3.1:8b"]
    assert d.resolve_model_name("llama3:latest", avail) == "llama3:latest"
    assert d.resolve_model_name("qwen2.5", avail) == "qwen2.5:7b"
    assert d.resolve_model_name("qwen", avail) == "qwen2.5:7b"  # unique sub
    assert d.resolve_model_name("llama", avail) is None  # ambiguous
    assert d.resolve_model_name("  ", avail) is None
    assert d.resolve_model_name("nope", avail) is None


def test_resolve_lmstudio_display_name():
    native = {"mistral-7b-v0.3": "Mistral 7B v0.3"}
    assert d.resolve_model_name("Mistral 7B v0.3", [], native) \
        == "mistral-7b-v0.3"
    assert d.resolve_model_name("mistral", [], native) == "mistral-7b-v0.3"


# -------------------------------------------------------------- scheduler
def test_least_connections_round_robin():
    h = d.Harness()
    for i in range(3):
        h.add_backend(f"b{i}", max_conc=1)
    for i in range(3):
        h.enqueue("u", "", "/api/chat")
    picks = [h.schedule()[1] for _ in range(3)]
    assert sorted(picks) == [0, 1, 2]       # least-conn spreads
    assert h.schedule() is None             # all busy (1 in-flight each)
    h.finish(0, True)
    h.enqueue("u")
    assert h.schedule() is not None


def test_fair_share_least_served_first():
    h = d.Harness()
    h.add_backend("b0")
    h.set_processed("heavy", 100)
    h.set_processed("light", 1)
    h.enqueue("heavy")
    h.enqueue("light")
    user, _, _ = h.schedule()
    assert user == "light"


def test_vip_absolute_priority():
    h = d.Harness()
    h.add_backend("b0")
    h.set_processed("vip", 1000)
    h.set_processed("pleb", 0)
    h.set_vip("vip")
    h.enqueue("pleb")
    h.enqueue("vip")
    assert h.schedule()[0] == "vip"


def test_boost_every_second_tick():
    h = d.Harness()
    h.add_backend("b0", max_conc=100)
    h.set_boost("boosted")
    h.set_processed("boosted", 1000)
    h.set_processed("other", 1000)
    for _ in range(6):
        h.enqueue("boosted")
        h.enqueue("other")
    picks = []
    for i in range(12):
        r = h.schedule()
        picks.append(r[0])
    # boost wins on even global counter ticks: roughly alternating, and
    # must get at least 1/3 of the early picks despite equal load
    assert picks.count("boosted") >= 4


def test_offline_and_control_op_excluded():
    h = d.Harness()
    b0 = h.add_backend("b0")
    b1 = h.add_backend("b1")
    h.set_online(b0, False)
    h.enqueue("u")
    assert h.schedule()[1] == b1
    h.finish(0, True)                    # free b1
    h.set_control_op(b1, True)
    h.enqueue("u")
    assert h.schedule() is None          # b0 offline, b1 busy with control
    h.set_control_op(b1, False)
    assert h.schedule()[1] == b1


def test_model_routing_and_loaded_preference():
    h = d.Harness()
    b0 = h.add_backend("b0", available=["llama3:latest"], loaded=[])
    b1 = h.add_backend("b1", available=["llama3:latest"],
                       loaded=["llama3:latest"])
    h.add_backend("b2", available=["qwen2:7b"])
    h.enqueue("u", "llama3")
    # prefers the backend with the model already loaded
    assert h.schedule()[1] == b1
    h.enqueue("u", "llama3")
    assert h.schedule()[1] == b0         # b1 busy now; falls back
    h.enqueue("u", "gemma")
    assert h.schedule() is None          # nobody has it


def test_api_family_routing():
    h = d.Harness()
    b0 = h.add_backend("b0", api="ollama")
    b1 = h.add_backend("b1", api="openai")
    h.enqueue("u", "", "/v1/chat/completions")
    assert h.schedule()[1] == b1
    h.enqueue("u", "", "/api/generate")
    assert h.schedule()[1] == b0


def test_queue_scan_skips_unroutable_head():
    """Head-of-line: the first routable task dispatches even when the queue
    head has no eligible backend (reference scans, src/dispatcher.rs:570)."""
    h = d.Harness()
    h.add_backend("b0", available=["llama3"])
    h.enqueue("u", "gemma")     # head: unroutable
    h.enqueue("u", "llama3")    # second: routable
    r = h.schedule()
    assert r is not None and r[2] == "llama3"
    assert h.queue_len("u") == 1  # gemma still parked


def test_stuck_timeout_503():
    h = d.Harness()
    h.add_backend("b0", available=["llama3"])
    h.set_stuck_timeout(1)      # 1 s
    h.enqueue_aged("u", "gemma", "/api/chat", 5000)
    assert h.schedule() is None
    assert h.dropped("u") == 1
    assert h.queue_len("u") == 0


def test_debug_rejection_log():
    """Per-candidate rejection reasons at debug level (reference logs them
    under RUST_LOG=debug, src/dispatcher.rs:579-615); silent when off."""
    h = d.Harness()
    b0 = h.add_backend("b0", available=["llama3"])
    h.add_backend("b1", available=["qwen2"])
    h.set_online(b0, False)
    h.enqueue("u", "llama3")
    assert h.schedule() is None          # b0 offline, b1 lacks the model
    assert not any(k == "DBG" for k, _ in h.log_events())  # off by default

    h.set_debug_log(True)
    h.enqueue("u", "llama3")
    assert h.schedule() is None
    dbg = [t for k, t in h.log_events() if k == "DBG"]
    assert any("rejected: offline" in t and "backend[0]" in t for t in dbg)
    assert any("rejected: model-not-available" in t and "backend[1]" in t
               for t in dbg)


def test_json_model_extraction():
    assert d.json_get_model('{"model": "llama3", "messages": []}') == "llama3"
    assert d.json_get_model('{"x": 1}') == ""
    assert d.json_get_model('not json') == ""
    # nested / escaped content must not confuse it
    assert d.json_get_model(
        '{"messages":[{"content":"say \\"model\\": x"}],"model":"m1"}'
    ) == "m1"


def test_json_roundtrip():
    s = '{"a":[1,2.5,"x",true,null],"b":{"c":"\\u00e9"}}'
    out = d.json_roundtrip(s)
    assert out is not None
    assert d.json_roundtrip(out) == out   # stable
    assert d.json_roundtrip("{bad") is None
