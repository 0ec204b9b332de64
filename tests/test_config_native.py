"""C2: appconf.yaml parsing — mirrors the reference's config test suite
(reference src/config.rs:137-238: full config, minimal defaults, empty
file valid, keep_alive -1 accepted / < -1 rejected, URL normalization)."""
import os

import pytest

from ollamamq_amd.dispatch import load

d = load()


def write(tmp_path, text):
    p = os.path.join(str(tmp_path), "appconf.yaml")
    with open(p, "w") as f:
        f.write(text)
    return p


def test_full_config(tmp_path):
    p = write(tmp_path, """
# comment
backends:
  - http://10.0.0.1:11434    # Ollama
  - 10.0.0.5:1234

settings:
  port: 12345
  host: 0.0.0.0
  timeout: 120
  load_keep_alive: 3600
  allow_all_routes: true
  stuck_timeout: 30

models:
  - name: "gpt-oss:120b"
    identifier: "my-gpt"
    max_ctx: 128000
    keep_alive: 86400
    max_concurrent_requests: 3
    backends:
      - http://10.0.0.1:11434
      - 10.0.0.5:1234
  - name: "qwen2.5-coder:7b"
    identifier: "coder"
    max_ctx: 32768
""")
    cfg = d.load_config(p)
    assert cfg["backends"] == ["http://10.0.0.1:11434", "10.0.0.5:1234"]
    s = cfg["settings"]
    assert s["port"] == 12345 and s["host"] == "0.0.0.0"
    assert s["timeout"] == 120 and s["load_keep_alive"] == 3600
    assert s["allow_all_routes"] is True and s["stuck_timeout"] == 30
    m = cfg["models"]
    assert len(m) == 2
    assert m[0]["name"] == "gpt-oss:120b"
    assert m[0]["identifier"] == "my-gpt"
    assert m[0]["max_ctx"] == 128000
    assert m[0]["max_concurrent_requests"] == 3
    assert m[0]["backends"] == ["http://10.0.0.1:11434", "10.0.0.5:1234"]
    assert m[1]["backends"] == []


def test_minimal_defaults(tmp_path):
    p = write(tmp_path, "settings:\n  port: 1\n")
    cfg = d.load_config(p)
    assert cfg["settings"]["port"] == 1
    assert cfg["settings"]["timeout"] == 300         # default
    assert cfg["settings"]["load_keep_alive"] == 86400
    assert cfg["backends"] == [] and cfg["models"] == []


def test_empty_file_valid(tmp_path):
    p = write(tmp_path, "\n# nothing\n")
    cfg = d.load_config(p)
    assert cfg["settings"]["port"] == 11435


def test_missing_file_defaults(tmp_path):
    cfg = d.load_config(os.path.join(str(tmp_path), "nope.yaml"))
    assert cfg["settings"]["port"] == 11435


def test_keep_alive_minus_one_ok(tmp_path):
    p = write(tmp_path, "models:\n  - name: m\n    keep_alive: -1\n")
    assert d.load_config(p)["models"][0]["keep_alive"] == -1


def test_keep_alive_below_minus_one_rejected(tmp_path):
    p = write(tmp_path, "models:\n  - name: m\n    keep_alive: -2\n")
    with pytest.raises(RuntimeError, match="keep_alive"):
        d.load_config(p)


def test_url_normalization():
    assert d.normalize_backend_url("10.0.0.1:11434/") == \
        "http://10.0.0.1:11434"
    assert d.normalize_backend_url("http://x/") == "http://x"
    assert d.normalize_backend_url(" https://y ") == "https://y"


def test_adversarial_yaml_refused(tmp_path):
    """Constructs outside the documented subset must refuse to load with
    a line-numbered error, never silently misparse."""
    bad = [
        ("anchors", "backends:\n  - &b http://x\n"),
        ("alias value", "models:\n  - name: *b\n"),
        ("block scalar", "settings:\n  host: |\n    multi\n    line\n"),
        ("flow mapping", "models:\n  - {name: x, max_ctx: 1}\n"),
        ("flow list", "backends: [http://a, http://b]\n"),
        ("tabs", "settings:\n\tport: 1\n"),
        ("merge key", "settings:\n  <<: *base\n"),
    ]
    for label, text in bad:
        p = write(tmp_path, text)
        with pytest.raises(RuntimeError) as ei:
            d.load_config(p)
        msg = str(ei.value)
        assert "line " in msg and "unsupported" in msg, (label, msg)


def test_benign_yaml_still_parses(tmp_path):
    """Doc markers, comments, quotes, empty flow lists and unknown keys
    stay accepted (serde_yaml ignores unknown fields; reference
    config.rs:137-238)."""
    p = write(tmp_path,
              "---\n"
              "# comment\n"
              "backends:\n"
              "  - \"http://a:1\"   # inline comment\n"
              "settings:\n"
              "  port: 1234\n"
              "  future_knob: whatever\n"
              "models: []\n"
              "...\n")
    cfg = d.load_config(p)
    assert cfg["backends"] == ["http://a:1"]
    assert cfg["settings"]["port"] == 1234
