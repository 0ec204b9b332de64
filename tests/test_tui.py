"""TUI smoke over a real PTY (C19): the dashboard must render its
panels, respond to keys (help overlay, panel focus, expand), and exit
cleanly on 'q'."""
import os
import pty
import subprocess
import sys
import time

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
BIN = os.path.join(REPO, "ollamamq_amd", "csrc", "dispatcher",
                   "ollamamq-server")


def test_tui_renders_and_quits(tmp_path):
    if not os.path.exists(BIN):
        subprocess.run([sys.executable, "-m", "ollamamq_amd.build"],
                       check=True)
    master, slave = pty.openpty()
    p = subprocess.Popen(
        [BIN, "-p", "0", "-c", os.path.join(str(tmp_path), "absent.yaml")],
        stdin=slave, stdout=slave, stderr=subprocess.DEVNULL,
        cwd=str(tmp_path))
    os.close(slave)
    out = b""
    deadline = time.time() + 30
    try:
        # let a few frames render, poke some keys, then quit
        sent = False
        while time.time() < deadline:
            try:
                chunk = os.read(master, 65536)
            except OSError:
                break
            out += chunk
            if not sent and b"Backends" in out:
                os.write(master, b"?")      # help overlay
                time.sleep(0.3)
                os.write(master, b"\t")     # panel focus
                time.sleep(0.3)
                os.write(master, b"a")      # show-all models
                time.sleep(0.3)
                os.write(master, b"q")      # quit
                sent = True
            if sent and p.poll() is not None:
                break
        assert sent, f"TUI never rendered Backends panel: {out[-500:]!r}"
        assert p.wait(timeout=10) == 0, "TUI did not exit cleanly on q"
    finally:
        os.close(master)
        if p.poll() is None:
            p.terminate()
    text = out.decode(errors="replace")
    assert "Backends" in text and "Users" in text and "Logs" in text
    assert "Keys" in text, "help overlay never rendered"


def test_tui_model_cursor_direct_load(tmp_path):
    """Reference tui.rs:607-636,345-387: Tab walks the expanded backend's
    sorted model list and L loads the highlighted model directly (no
    typed-name input mode)."""
    sys.path.insert(0, os.path.join(REPO, "tests"))
    from mocks import MockFleet
    fleet = MockFleet()
    try:
        master, slave = pty.openpty()
        p = subprocess.Popen(
            [BIN, "-p", "0", "-o", fleet.ollama_url,
             "-c", os.path.join(str(tmp_path), "absent.yaml")],
            stdin=slave, stdout=slave, stderr=subprocess.DEVNULL,
            cwd=str(tmp_path))
        os.close(slave)
        out = b""
        deadline = time.time() + 40
        stage = 0
        try:
            while time.time() < deadline:
                try:
                    chunk = os.read(master, 65536)
                except OSError:
                    break
                out += chunk
                # wait for the probe to mark the mock backend online
                if stage == 0 and b"\xe2\x97\x8f" in out:
                    # backend online; expand + cursor + L
                    os.write(master, b"\t")    # focus Backends
                    time.sleep(0.3)
                    os.write(master, b" ")     # expand selected backend
                    time.sleep(0.5)
                    os.write(master, b"\t")    # model cursor -> first
                    time.sleep(0.3)
                    os.write(master, b"L")     # direct load
                    stage = 1
                    t_fire = time.time()
                if stage == 1 and (b"accepted" in out
                                   or time.time() - t_fire > 8):
                    os.write(master, b"q")
                    stage = 2
                if stage == 2 and p.poll() is not None:
                    break
            assert stage >= 2, f"never fired cursor-load: {out[-600:]!r}"
            p.wait(timeout=10)
        finally:
            os.close(master)
            if p.poll() is None:
                p.terminate()
        text = out.decode(errors="replace")
        assert "accepted" in text, text[-800:]
        # the direct load reached the mock backend's generate endpoint
        deadline = time.time() + 10
        while time.time() < deadline:
            if fleet.recorder.of(path="/api/generate"):
                break
            time.sleep(0.3)
        assert fleet.recorder.of(path="/api/generate"), \
            "cursor-load never hit the backend"
    finally:
        fleet.stop()


def test_tui_arrow_keys_navigate_not_quit(tmp_path):
    """Arrow keys send ESC [ A/B: they must navigate (reference tui.rs
    j/k/up-down parity), NOT be mistaken for a bare Esc quit; bare Esc
    still quits."""
    master, slave = pty.openpty()
    p = subprocess.Popen(
        [BIN, "-p", "0", "-c", os.path.join(str(tmp_path), "absent.yaml")],
        stdin=slave, stdout=slave, stderr=subprocess.DEVNULL,
        cwd=str(tmp_path))
    os.close(slave)
    out = b""
    deadline = time.time() + 30
    stage = 0
    try:
        while time.time() < deadline:
            try:
                out += os.read(master, 65536)
            except OSError:
                break
            if stage == 0 and b"Backends" in out:
                os.write(master, b"\x1b[B\x1b[A\x1b[Z")  # down, up, S-Tab
                stage, out = 1, b""
            elif stage == 1 and b"Backends" in out:
                # still alive and rendering after the arrow keys
                assert p.poll() is None, "arrow keys quit the TUI"
                os.write(master, b"\x1b")               # bare Esc quits
                stage = 2
            elif stage == 2 and p.poll() is not None:
                break
        assert stage == 2, f"stage={stage}: {out[-300:]!r}"
        assert p.wait(timeout=10) == 0, "Esc did not exit cleanly"
    finally:
        os.close(master)
        if p.poll() is None:
            p.terminate()


def test_tui_queues_panel(tmp_path):
    """The Queues panel renders per-user load bars (reference
    tui.rs:1124-1163)."""
    master, slave = pty.openpty()
    p = subprocess.Popen(
        [BIN, "-p", "0", "-c", os.path.join(str(tmp_path), "absent.yaml")],
        stdin=slave, stdout=slave, stderr=subprocess.DEVNULL,
        cwd=str(tmp_path))
    os.close(slave)
    out = b""
    deadline = time.time() + 20
    try:
        while time.time() < deadline:
            try:
                out += os.read(master, 65536)
            except OSError:
                break
            if b"Queues" in out:
                os.write(master, b"q")
                break
        p.wait(timeout=10)
    finally:
        os.close(master)
        if p.poll() is None:
            p.terminate()
    assert b"Queues" in out, out[-400:]
