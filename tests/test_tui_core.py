"""Event-loop TUI models (tui/core.py): pure update/view driven by
synthetic messages — the same way bubbletea models are unit-tested
(reference internal/tui/*)."""
import io

from runbooks_amd.tui.core import (
    ChecklistModel,
    DataMsg,
    KeyMsg,
    LogViewModel,
    Program,
    QuitMsg,
    SelectModel,
    TickMsg,
    decode_key,
)


def drain(program):
    quits = []
    while not program.msgs.empty():
        m = program.msgs.get()
        if isinstance(m, QuitMsg):
            quits.append(m)
    return quits


def test_decode_key_sequences():
    assert decode_key(b"\x1b[A").key == "up"
    assert decode_key(b"\x1b[B").key == "down"
    assert decode_key(b"\r").key == "enter"
    assert decode_key(b"\x03").key == "ctrl+c"
    assert decode_key(b"j").key == "j"
    assert decode_key(b"\x1b").key == "esc"


def test_select_model_navigation_and_choice():
    m = SelectModel("pick a manifest", ["Model/llama", "Server/llama",
                                       "Notebook/dev"])
    p = Program(m, out=io.StringIO())
    p.step(KeyMsg("down"))
    p.step(KeyMsg("down"))
    p.step(KeyMsg("up"))
    assert p.model.cursor == 1
    assert "❯ Server/llama" in p.model.view()
    p.step(KeyMsg("enter"))
    assert p.model.chosen == 1
    import time
    time.sleep(0.05)          # command thread posts QuitMsg
    assert drain(p)


def test_select_model_bounds():
    m = SelectModel("t", ["a"])
    p = Program(m, out=io.StringIO())
    p.step(KeyMsg("up"))
    p.step(KeyMsg("down"))
    assert p.model.cursor == 0


def test_checklist_transitions_to_done():
    m = ChecklistModel("waiting for server")
    p = Program(m, out=io.StringIO())
    p.step(DataMsg("conditions", [("Built", True, ""),
                                  ("Deployed", False, "0/1 replicas")]))
    v = p.model.view()
    assert "✓ Built" in v and "Deployed" in v and "0/1" in v
    assert not p.model.done
    p.step(TickMsg(3))
    p.step(DataMsg("conditions", [("Built", True, ""),
                                  ("Deployed", True, "")]))
    assert p.model.done
    assert "ready" in p.model.view()


def test_logview_scrolls_and_caps():
    m = LogViewModel("pod logs", height=3)
    p = Program(m, out=io.StringIO())
    for i in range(10):
        p.step(DataMsg("log", f"line-{i}"))
    v = p.model.view()
    assert "line-9" in v and "line-7" in v and "line-6" not in v


def test_program_render_repaints(capsys=None):
    out = io.StringIO()
    m = SelectModel("t", ["a", "b"])
    p = Program(m, out=out)
    p._render()
    first = out.getvalue()
    assert "❯ a" in first
    p.step(KeyMsg("down"))
    p._render()
    assert "\x1b[" in out.getvalue()[len(first):]  # repaint moved cursor
    assert "❯ b" in out.getvalue()
