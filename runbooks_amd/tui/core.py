"""Interactive event-loop TUI runtime (the bubbletea architecture,
natively).

Parity: the reference's TUI layer is charmbracelet/bubbletea
(internal/tui/notebook.go:19-58, serve.go, run.go, apply.go, get.go,
delete.go composed from sub-models in internal/tui/common.go:70-310).
This module reproduces the ARCHITECTURE, which is what makes that layer
what it is: a Model with pure `update(msg) -> (model, commands)` and
`view() -> str` functions driven by an event loop that owns the
terminal — so every interaction is unit-testable by feeding messages,
no terminal required (exactly how bubbletea models are tested).

Runtime: raw-mode stdin reader thread -> message queue -> update ->
re-render (ANSI, diff-free full repaint like bubbletea's standard
renderer). Commands returned by update() run on worker threads and feed
their results back as messages.
"""
from __future__ import annotations

import os
import queue
import sys
import threading
from dataclasses import dataclass
from typing import Callable, List, Optional, Tuple

# ---------------------------------------------------------------------------
# Messages (bubbletea tea.Msg analog)
# ---------------------------------------------------------------------------


@dataclass(frozen=True)
class KeyMsg:
    key: str          # "up", "down", "enter", "q", "ctrl+c", plain chars


@dataclass(frozen=True)
class TickMsg:
    n: int


@dataclass(frozen=True)
class QuitMsg:
    pass


@dataclass(frozen=True)
class DataMsg:
    """Carrier for command results (watch events, upload progress...)."""
    kind: str
    payload: object = None


Cmd = Callable[[], Optional[object]]   # returns a Msg or None


class Model:
    """Base interface (tea.Model analog): override update/view."""

    def init(self) -> List[Cmd]:
        return []

    def update(self, msg) -> Tuple["Model", List[Cmd]]:
        return self, []

    def view(self) -> str:
        return ""


# ---------------------------------------------------------------------------
# Key decoding (raw bytes -> KeyMsg)
# ---------------------------------------------------------------------------

_ESC_SEQS = {
    b"[A": "up", b"[B": "down", b"[C": "right", b"[D": "left",
    b"[H": "home", b"[F": "end",
}


def decode_key(data: bytes) -> Optional[KeyMsg]:
    if not data:
        return None
    if data == b"\x03":
        return KeyMsg("ctrl+c")
    if data in (b"\r", b"\n"):
        return KeyMsg("enter")
    if data in (b"\x7f", b"\x08"):
        return KeyMsg("backspace")
    if data == b"\t":
        return KeyMsg("tab")
    if data == b"\x1b":
        return KeyMsg("esc")
    if data.startswith(b"\x1b"):
        return KeyMsg(_ESC_SEQS.get(data[1:], "esc"))
    try:
        return KeyMsg(data.decode())
    except UnicodeDecodeError:
        return None


# ---------------------------------------------------------------------------
# Program (tea.Program analog)
# ---------------------------------------------------------------------------


class Program:
    """Owns the terminal: raw mode, repaint, message pump.

    `run()` blocks until the model emits QuitMsg (or ctrl+c). For tests,
    `step(msg)` advances the model synchronously with no terminal."""

    def __init__(self, model: Model, out=None, fps: int = 30):
        self.model = model
        self.out = out if out is not None else sys.stdout
        self.msgs: "queue.Queue" = queue.Queue()
        self._quit = threading.Event()
        self._last_lines = 0
        self.fps = fps

    # -- test/synchronous path ---------------------------------------------
    def step(self, msg) -> Model:
        self.model, cmds = self.model.update(msg)
        for cmd in cmds:
            self._spawn(cmd)
        return self.model

    # -- event loop ---------------------------------------------------------
    def _spawn(self, cmd: Cmd) -> None:
        def runner():
            try:
                out = cmd()
            except Exception as e:  # surface as a message, don't kill loop
                out = DataMsg("error", repr(e))
            if out is not None:
                self.msgs.put(out)
        threading.Thread(target=runner, daemon=True).start()

    def _reader(self) -> None:
        fd = sys.stdin.fileno()
        while not self._quit.is_set():
            try:
                data = os.read(fd, 8)
            except OSError:
                return
            msg = decode_key(data)
            if msg is not None:
                self.msgs.put(msg)

    def _render(self) -> None:
        view = self.model.view()
        lines = view.split("\n")
        buf = []
        if self._last_lines:
            buf.append(f"\x1b[{self._last_lines}F")   # cursor up N
        for ln in lines:
            buf.append("\x1b[2K" + ln + "\n")         # clear + write
        if self._last_lines > len(lines):
            for _ in range(self._last_lines - len(lines)):
                buf.append("\x1b[2K\n")
            buf.append(f"\x1b[{self._last_lines - len(lines)}F")
        self.out.write("".join(buf))
        self.out.flush()
        self._last_lines = len(lines)

    def run(self) -> Model:
        import termios
        import tty
        fd = sys.stdin.fileno()
        old = termios.tcgetattr(fd)
        tty.setcbreak(fd)
        reader = threading.Thread(target=self._reader, daemon=True)
        reader.start()
        try:
            for cmd in self.model.init():
                self._spawn(cmd)
            self._render()
            while not self._quit.is_set():
                try:
                    msg = self.msgs.get(timeout=1.0 / self.fps)
                except queue.Empty:
                    continue
                if isinstance(msg, QuitMsg) or (
                        isinstance(msg, KeyMsg) and msg.key == "ctrl+c"):
                    break
                self.model, cmds = self.model.update(msg)
                for cmd in cmds:
                    self._spawn(cmd)
                if any(isinstance(m, QuitMsg) for m in self._drain_quits()):
                    break
                self._render()
            self._render()
        finally:
            termios.tcsetattr(fd, termios.TCSADRAIN, old)
            self.out.write("\n")
            self.out.flush()
        return self.model

    def _drain_quits(self):
        out = []
        while True:
            try:
                m = self.msgs.get_nowait()
            except queue.Empty:
                return out
            if isinstance(m, QuitMsg):
                out.append(m)
            else:
                self.msgs.put(m)
                return out


# ---------------------------------------------------------------------------
# Reusable sub-models (reference internal/tui/common.go analogs)
# ---------------------------------------------------------------------------


@dataclass
class SelectModel(Model):
    """Arrow-key list picker (reference manifests.go:130-262)."""
    title: str
    items: List[str]
    cursor: int = 0
    chosen: Optional[int] = None

    def update(self, msg):
        if isinstance(msg, KeyMsg):
            if msg.key in ("up", "k") and self.cursor > 0:
                self.cursor -= 1
            elif msg.key in ("down", "j") and self.cursor < len(self.items) - 1:
                self.cursor += 1
            elif msg.key == "enter":
                self.chosen = self.cursor
                return self, [lambda: QuitMsg()]
            elif msg.key in ("q", "esc"):
                return self, [lambda: QuitMsg()]
        return self, []

    def view(self) -> str:
        rows = [self.title]
        for i, it in enumerate(self.items):
            mark = "❯" if i == self.cursor else " "
            rows.append(f" {mark} {it}")
        rows.append("  ↑/↓ move · enter select · q quit")
        return "\n".join(rows)


@dataclass
class ChecklistModel(Model):
    """Live readiness checklist (reference readiness.go:15-100): each
    DataMsg("conditions", [...]) repaints; done when every condition is
    True."""
    title: str
    conditions: Tuple[Tuple[str, bool, str], ...] = ()
    done: bool = False
    frame: int = 0
    SPIN = "⠋⠙⠹⠸⠼⠴⠦⠧⠇⠏"

    def update(self, msg):
        if isinstance(msg, TickMsg):
            self.frame = msg.n
        elif isinstance(msg, DataMsg) and msg.kind == "conditions":
            self.conditions = tuple(msg.payload)
            if self.conditions and all(ok for _, ok, _ in self.conditions):
                self.done = True
                return self, [lambda: QuitMsg()]
        elif isinstance(msg, KeyMsg) and msg.key in ("q", "ctrl+c"):
            return self, [lambda: QuitMsg()]
        return self, []

    def view(self) -> str:
        rows = [self.title]
        spin = self.SPIN[self.frame % len(self.SPIN)]
        for name, ok, detail in self.conditions:
            mark = "✓" if ok else spin
            suffix = f"  {detail}" if detail else ""
            rows.append(f" {mark} {name}{suffix}")
        if self.done:
            rows.append(" ready")
        return "\n".join(rows)


@dataclass
class LogViewModel(Model):
    """Scrolling pod-log pane (reference pods.go:23-231): DataMsg("log",
    line) appends; keeps the last `height` lines; q quits."""
    title: str
    height: int = 12
    lines: Tuple[str, ...] = ()

    def update(self, msg):
        if isinstance(msg, DataMsg) and msg.kind == "log":
            self.lines = tuple(list(self.lines) + [str(msg.payload)])[-200:]
        elif isinstance(msg, KeyMsg) and msg.key in ("q", "esc"):
            return self, [lambda: QuitMsg()]
        return self, []

    def view(self) -> str:
        shown = self.lines[-self.height:]
        return "\n".join([self.title, *[f" │ {ln}" for ln in shown],
                          "  q quit"])
