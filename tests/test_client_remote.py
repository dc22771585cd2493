"""Native WebSocket exec / cp / port-forward (client/remote.py) against
a hermetic fake API server speaking the v4.channel.k8s.io protocol —
the same layering the reference uses for its SPDY client (client-go
remotecommand against envtest/testserver)."""
import asyncio
import json
import socket
import threading
import time

import pytest
from aiohttp import WSMsgType, web

from runbooks_amd.client import remote


class FakeClient:
    def __init__(self, host):
        self.host = host
        self.token = "test-token"
        self._ctx = None  # plain ws:// in tests


@pytest.fixture()
def fake_apiserver():
    """aiohttp server with /exec and /portforward WebSocket endpoints."""
    state = {"exec_requests": [], "auth": []}

    async def exec_handler(request):
        state["auth"].append(request.headers.get("Authorization"))
        cmd = request.query.getall("command")
        state["exec_requests"].append(cmd)
        ws = web.WebSocketResponse(protocols=(remote.SUBPROTOCOL,))
        await ws.prepare(request)
        stdin = bytearray()
        if cmd[:2] == ["sh", "-c"] and cmd[2].startswith("cat > "):
            # cp-to-pod flavor: drain stdin until close, store it
            async for msg in ws:
                if msg.type == WSMsgType.BINARY and msg.data and \
                        msg.data[0] == remote.CHANNEL_STDIN:
                    stdin.extend(msg.data[1:])
                    if b"<EOF>" in stdin:
                        break
            state[cmd[2][len("cat > "):]] = bytes(stdin)
            await ws.send_bytes(bytes([remote.CHANNEL_ERROR]) + json.dumps(
                {"status": "Success"}).encode())
            await ws.close()
            return ws
        if cmd[0] == "cat":
            await ws.send_bytes(bytes([remote.CHANNEL_STDOUT]) +
                                b"file-contents-1")
            await ws.send_bytes(bytes([remote.CHANNEL_STDOUT]) + b"-part2")
        elif cmd[0] == "false":
            await ws.send_bytes(
                bytes([remote.CHANNEL_ERROR]) + json.dumps({
                    "status": "Failure", "reason": "NonZeroExitCode",
                    "details": {"causes": [
                        {"reason": "ExitCode", "message": "3"}]}}).encode())
            await ws.close()
            return ws
        else:
            await ws.send_bytes(bytes([remote.CHANNEL_STDOUT]) + b"out!")
            await ws.send_bytes(bytes([remote.CHANNEL_STDERR]) + b"err!")
        await ws.send_bytes(bytes([remote.CHANNEL_ERROR]) + json.dumps(
            {"status": "Success"}).encode())
        await ws.close()
        return ws

    async def pf_handler(request):
        port = int(request.query["ports"])
        ws = web.WebSocketResponse(protocols=(remote.SUBPROTOCOL,))
        await ws.prepare(request)
        pb = port.to_bytes(2, "little")
        # announce the port on both channels (data=0, error=1)
        await ws.send_bytes(b"\x00" + pb)
        await ws.send_bytes(b"\x01" + pb)
        async for msg in ws:
            if msg.type == WSMsgType.BINARY and msg.data and \
                    msg.data[0] == 0:
                # echo server: reply with upper-cased payload
                await ws.send_bytes(b"\x00" + msg.data[1:].upper())
        return ws

    app = web.Application()
    app.router.add_get(
        "/api/v1/namespaces/{ns}/pods/{pod}/exec", exec_handler)
    app.router.add_get(
        "/api/v1/namespaces/{ns}/pods/{pod}/portforward", pf_handler)

    loop = asyncio.new_event_loop()
    runner = web.AppRunner(app)

    async def _start():
        await runner.setup()
        site = web.TCPSite(runner, "127.0.0.1", 0)
        await site.start()
        return runner.addresses[0][1]

    holder = {}

    def run():
        asyncio.set_event_loop(loop)
        holder["port"] = loop.run_until_complete(_start())
        loop.run_forever()

    t = threading.Thread(target=run, daemon=True)
    t.start()
    for _ in range(100):
        if "port" in holder:
            break
        time.sleep(0.02)
    state["host"] = f"http://127.0.0.1:{holder['port']}"
    yield state
    loop.call_soon_threadsafe(loop.stop)


def test_pod_exec_streams_and_status(fake_apiserver):
    c = FakeClient(fake_apiserver["host"])
    outs = []
    res = remote.pod_exec(c, "default", "pod-1", ["echo", "hi"],
                          on_stdout=lambda b: outs.append(b))
    assert bytes(res.stdout) == b"out!"
    assert bytes(res.stderr) == b"err!"
    assert outs == [b"out!"]
    assert res.returncode == 0
    assert fake_apiserver["auth"][-1] == "Bearer test-token"
    assert fake_apiserver["exec_requests"][-1] == ["echo", "hi"]


def test_pod_exec_exit_code(fake_apiserver):
    c = FakeClient(fake_apiserver["host"])
    res = remote.pod_exec(c, "default", "pod-1", ["false"])
    assert res.returncode == 3


def test_cp_roundtrip(fake_apiserver):
    c = FakeClient(fake_apiserver["host"])
    res = remote.cp_to_pod_native(c, "ns", "pod-1",
                                  b"payload-bytes<EOF>", "/tmp/x.bin")
    assert res.returncode == 0
    assert fake_apiserver["/tmp/x.bin"] == b"payload-bytes<EOF>"
    data = remote.cp_from_pod_native(c, "ns", "pod-1", "/tmp/y.txt")
    assert data == b"file-contents-1-part2"


def test_port_forward_echo(fake_apiserver):
    c = FakeClient(fake_apiserver["host"])
    pf = remote.PortForward(c, "ns", "pod-1", remote_port=8080).start()
    try:
        s = socket.create_connection(("127.0.0.1", pf.local_port), timeout=5)
        s.sendall(b"hello-through-the-tunnel")
        s.settimeout(5)
        got = s.recv(1 << 16)
        assert got == b"HELLO-THROUGH-THE-TUNNEL"
        s.close()
    finally:
        pf.stop()
