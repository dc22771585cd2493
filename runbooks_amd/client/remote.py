"""Native pod exec / cp / port-forward over Kubernetes WebSocket
streams (no kubectl subprocess).

Parity: reference internal/client/sync.go:125-176 (SPDY exec via
client-go remotecommand) and internal/client/port_forward.go:21-46
(SPDY port-forward). The modern K8s API serves the SAME channel
protocol over WebSocket (`v4.channel.k8s.io`), which is what this
module speaks via aiohttp:

- exec:        GET  .../pods/{name}/exec?command=...      channels:
               0 stdin, 1 stdout, 2 stderr, 3 error-status (JSON)
- port-forward GET  .../pods/{name}/portforward?ports=N   channels per
               port: even = data, odd = error; the FIRST frame on each
               channel carries the port number as 2 bytes LE.

Auth/TLS reuse the dynamic client's config (bearer token + CA bundle,
`runbooks_amd/k8s/client.py`).
"""
from __future__ import annotations

import asyncio
import json
import socket
import threading
from typing import Callable, Iterable, Optional

import aiohttp

CHANNEL_STDIN = 0
CHANNEL_STDOUT = 1
CHANNEL_STDERR = 2
CHANNEL_ERROR = 3

SUBPROTOCOL = "v4.channel.k8s.io"


def _ws_base(host: str) -> str:
    if host.startswith("https://"):
        return "wss://" + host[len("https://"):]
    if host.startswith("http://"):
        return "ws://" + host[len("http://"):]
    return "wss://" + host


class ExecResult:
    def __init__(self):
        self.stdout = bytearray()
        self.stderr = bytearray()
        self.status: dict = {}

    @property
    def returncode(self) -> int:
        if self.status.get("status") == "Success":
            return 0
        for c in self.status.get("details", {}).get("causes", []):
            if c.get("reason") == "ExitCode":
                return int(c.get("message", 1))
        return 1 if self.status else 0


async def _exec_async(host: str, token: str, ssl_ctx, namespace: str,
                      pod: str, command: Iterable[str],
                      container: Optional[str], stdin: Optional[bytes],
                      on_stdout: Optional[Callable[[bytes], None]],
                      on_stderr: Optional[Callable[[bytes], None]],
                      ) -> ExecResult:
    params = [("stdout", "true"), ("stderr", "true")]
    if stdin is not None:
        params.append(("stdin", "true"))
    if container:
        params.append(("container", container))
    params += [("command", c) for c in command]
    url = (f"{_ws_base(host)}/api/v1/namespaces/{namespace}"
           f"/pods/{pod}/exec")
    headers = {}
    if token:
        headers["Authorization"] = f"Bearer {token}"
    res = ExecResult()
    async with aiohttp.ClientSession() as sess:
        async with sess.ws_connect(url, params=params, headers=headers,
                                   ssl=ssl_ctx if ssl_ctx is not None
                                   else True,
                                   protocols=(SUBPROTOCOL,)) as ws:
            if stdin is not None:
                # chunk stdin; channel byte 0 prefixes each frame
                view = memoryview(stdin)
                for off in range(0, len(view), 1 << 20):
                    await ws.send_bytes(
                        bytes([CHANNEL_STDIN]) + bytes(view[off:off + (1 << 20)]))
                # half-close: servers treat close of stdin via close_send;
                # the WS protocol has no stdin EOF frame, so we rely on
                # the command reading until the socket closes OR finishing
                # on its own (tar/cat do).
            async for msg in ws:
                if msg.type != aiohttp.WSMsgType.BINARY or not msg.data:
                    continue
                ch, payload = msg.data[0], msg.data[1:]
                if ch == CHANNEL_STDOUT:
                    res.stdout.extend(payload)
                    if on_stdout:
                        on_stdout(payload)
                elif ch == CHANNEL_STDERR:
                    res.stderr.extend(payload)
                    if on_stderr:
                        on_stderr(payload)
                elif ch == CHANNEL_ERROR:
                    try:
                        res.status = json.loads(payload.decode())
                    except json.JSONDecodeError:
                        res.status = {"status": "Failure",
                                      "message": payload.decode("utf-8",
                                                                "replace")}
    return res


def pod_exec(client, namespace: str, pod: str, command: Iterable[str],
             container: Optional[str] = None,
             stdin: Optional[bytes] = None,
             on_stdout: Optional[Callable[[bytes], None]] = None,
             on_stderr: Optional[Callable[[bytes], None]] = None
             ) -> ExecResult:
    """Blocking exec in `pod` via the API server's WebSocket channel
    protocol. `client` is an HTTPKubeClient (host/token/_ctx reused)."""
    return asyncio.run(_exec_async(
        client.host, client.token, client._ctx, namespace, pod,
        list(command), container, stdin, on_stdout, on_stderr))


def cp_to_pod_native(client, namespace: str, pod: str, data: bytes,
                     dst_path: str, container: Optional[str] = None
                     ) -> ExecResult:
    """Write `data` to dst_path inside the pod (sh redirection via exec
    stdin — the same trick kubectl cp plays with tar)."""
    return pod_exec(client, namespace, pod,
                    ["sh", "-c", f"cat > {dst_path}"],
                    container=container, stdin=data)


def cp_from_pod_native(client, namespace: str, pod: str, src_path: str,
                       container: Optional[str] = None) -> bytes:
    res = pod_exec(client, namespace, pod, ["cat", src_path],
                   container=container)
    if res.returncode != 0:
        raise RuntimeError(f"cp_from_pod {src_path}: {res.status}")
    return bytes(res.stdout)


class PortForward:
    """Local TCP listener forwarding each connection to `remote_port` of
    the pod over its own WebSocket stream (the API server's WS
    port-forward carries one port pair per session; kubectl's SPDY
    multiplexes instead — one WS per TCP connection is the browser-
    client pattern and keeps the framing trivial)."""

    def __init__(self, client, namespace: str, name: str, remote_port: int,
                 local_port: int = 0, resource: str = "pod"):
        self.client = client
        self.namespace = namespace
        self.name = name
        self.remote_port = remote_port
        self.resource = resource
        self._sock = socket.socket()
        self._sock.setsockopt(socket.SOL_SOCKET, socket.SO_REUSEADDR, 1)
        self._sock.bind(("127.0.0.1", local_port))
        self._sock.listen(8)
        self.local_port = self._sock.getsockname()[1]
        self._stop = threading.Event()
        self._thread = threading.Thread(target=self._serve, daemon=True)

    def start(self) -> "PortForward":
        self._thread.start()
        return self

    def stop(self) -> None:
        self._stop.set()
        try:
            self._sock.close()
        except OSError:
            pass

    # -- internals ---------------------------------------------------------
    def _url(self) -> str:
        c = self.client
        return (f"{_ws_base(c.host)}/api/v1/namespaces/{self.namespace}"
                f"/pods/{self.name}/portforward")

    def _serve(self) -> None:
        while not self._stop.is_set():
            try:
                conn, _ = self._sock.accept()
            except OSError:
                return
            threading.Thread(target=self._handle, args=(conn,),
                             daemon=True).start()

    def _handle(self, conn: socket.socket) -> None:
        try:
            asyncio.run(self._pipe(conn))
        finally:
            try:
                conn.close()
            except OSError:
                pass

    async def _pipe(self, conn: socket.socket) -> None:
        c = self.client
        headers = {}
        if c.token:
            headers["Authorization"] = f"Bearer {c.token}"
        loop = asyncio.get_running_loop()
        conn.setblocking(False)
        async with aiohttp.ClientSession() as sess:
            async with sess.ws_connect(
                    self._url(), params={"ports": str(self.remote_port)},
                    headers=headers,
                    ssl=c._ctx if c._ctx is not None else True,
                    protocols=(SUBPROTOCOL,)) as ws:
                seen_first = {0: False, 1: False}

                async def ws_to_sock():
                    async for msg in ws:
                        if msg.type != aiohttp.WSMsgType.BINARY or \
                                not msg.data:
                            continue
                        ch, payload = msg.data[0], msg.data[1:]
                        if not seen_first.get(ch, True):
                            # first frame per channel: 2-byte LE port id
                            seen_first[ch] = True
                            payload = payload[2:]
                        if ch == 0 and payload:
                            await loop.sock_sendall(conn, payload)
                        elif ch == 1 and payload:
                            raise RuntimeError(
                                f"port-forward error: {payload.decode()}")

                async def sock_to_ws():
                    while True:
                        data = await loop.sock_recv(conn, 1 << 16)
                        if not data:
                            await ws.close()
                            return
                        await ws.send_bytes(b"\x00" + data)

                done, pending = await asyncio.wait(
                    [asyncio.ensure_future(ws_to_sock()),
                     asyncio.ensure_future(sock_to_ws())],
                    return_when=asyncio.FIRST_COMPLETED)
                for p in pending:
                    p.cancel()
                for d in done:
                    exc = d.exception()
                    if exc and not isinstance(exc, asyncio.CancelledError):
                        raise exc
