"""In-process fake Pulsar WebSocket proxy (aiohttp server)."""

from __future__ import annotations

import asyncio
import base64
import json
import threading


class FakePulsar:
    def __init__(self):
        #: topic path -> list of payload bytes
        self.messages: dict[str, list[bytes]] = {}
        #: topic path -> asyncio.Queue per consumer
        self._queues: dict[str, list] = {}
        self.lock = threading.Lock()
        self.loop = asyncio.new_event_loop()
        self.port = None
        self._started = threading.Event()
        self.thread = threading.Thread(target=self._serve, daemon=True)

    def _serve(self):
        asyncio.set_event_loop(self.loop)
        from aiohttp import web

        broker = self

        async def producer_ws(request):
            topic = request.match_info["topic"]
            ws = web.WebSocketResponse()
            await ws.prepare(request)
            async for msg in ws:
                if msg.type.name != "TEXT":
                    break
                d = json.loads(msg.data)
                payload = base64.b64decode(d.get("payload", ""))
                with broker.lock:
                    broker.messages.setdefault(topic, []).append(payload)
                    queues = list(broker._queues.get(topic, []))
                for q in queues:
                    q.put_nowait(payload)
                await ws.send_str(json.dumps(
                    {"result": "ok", "messageId": f"m{len(broker.messages[topic])}"}
                ))
            return ws

        async def consumer_ws(request):
            topic = request.match_info["topic"]
            ws = web.WebSocketResponse()
            await ws.prepare(request)
            q: asyncio.Queue = asyncio.Queue()
            with broker.lock:
                for p in broker.messages.get(topic, []):
                    q.put_nowait(p)
                broker._queues.setdefault(topic, []).append(q)
            try:
                i = 0
                while True:
                    payload = await q.get()
                    i += 1
                    await ws.send_str(json.dumps({
                        "messageId": f"c{i}",
                        "payload": base64.b64encode(payload).decode(),
                        "properties": {},
                    }))
                    ack = await ws.receive()
                    if ack.type.name != "TEXT":
                        break
            finally:
                with broker.lock:
                    if q in broker._queues.get(topic, []):
                        broker._queues[topic].remove(q)
            return ws

        async def consumer_dispatch(request):
            # consumer path ends with /<subscription>; strip it
            full = request.match_info["topic_and_sub"]
            request.match_info["topic"] = full.rsplit("/", 1)[0]
            return await consumer_ws(request)

        app = web.Application()
        app.router.add_get("/ws/v2/producer/{topic:.+}", producer_ws)
        app.router.add_get(
            "/ws/v2/consumer/{topic_and_sub:.+}", consumer_dispatch
        )

        runner = web.AppRunner(app)
        self.loop.run_until_complete(runner.setup())
        site = web.TCPSite(runner, "127.0.0.1", 0)
        self.loop.run_until_complete(site.start())
        self.port = site._server.sockets[0].getsockname()[1]
        self._started.set()
        self.loop.run_forever()

    @property
    def url(self) -> str:
        return f"http://127.0.0.1:{self.port}"

    def start(self) -> "FakePulsar":
        self.thread.start()
        self._started.wait(10)
        return self

    def stop(self) -> None:
        self.loop.call_soon_threadsafe(self.loop.stop)
