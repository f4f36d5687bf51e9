"""WebSocket stats mesh — server side.

Parity surface: /root/reference/stats_server.py (asyncio WebSocket server:
worker register/broadcast/history, JSON persistence, 30 s staleness marking).
Built on aiohttp (the image has no ``websockets`` package).

Protocol (JSON messages):
  {"type": "register", "worker_id": W, "info": {...}}
  {"type": "stats",    "worker_id": W, "stats": {...}}
  {"type": "heartbeat","worker_id": W}
  {"type": "get_history", "worker_id": W|null}    -> {"type":"history", ...}
  {"type": "get_workers"}                          -> {"type":"workers", ...}
Every stats message is re-broadcast to all other connected clients.
"""
from __future__ import annotations

import asyncio
import json
import time
from pathlib import Path
from typing import Any, Dict, List, Optional

try:
    from aiohttp import WSMsgType, web

    HAVE_AIOHTTP = True
except Exception:  # pragma: no cover
    HAVE_AIOHTTP = False

STALE_AFTER_S = 30.0


class StatsServer:
    def __init__(self, host: str = "127.0.0.1", port: int = 8765,
                 persist_path: Optional[str] = None, history_limit: int = 1000):
        if not HAVE_AIOHTTP:  # pragma: no cover
            raise RuntimeError("aiohttp is required for StatsServer")
        self.host, self.port = host, port
        self.history_limit = history_limit
        self.persist_path = Path(persist_path) if persist_path else None
        self.workers: Dict[str, Dict[str, Any]] = {}
        self.history: Dict[str, List[Dict[str, Any]]] = {}
        self._clients: set = set()
        self._runner: Optional[web.AppRunner] = None
        if self.persist_path and self.persist_path.exists():
            try:
                saved = json.loads(self.persist_path.read_text())
                self.workers = saved.get("workers", {})
                self.history = saved.get("history", {})
            except Exception:
                pass

    # ---- message handling ----
    async def _handle_message(self, ws, data: Dict[str, Any]) -> None:
        mtype = data.get("type")
        now = time.time()
        wid = data.get("worker_id")
        if mtype == "register":
            self.workers[wid] = {"info": data.get("info", {}), "registered_at": now,
                                 "last_seen": now, "active": True}
            self.history.setdefault(wid, [])
            await ws.send_json({"type": "registered", "worker_id": wid})
        elif mtype == "stats":
            entry = {"time": now, **data.get("stats", {})}
            self.history.setdefault(wid, []).append(entry)
            if len(self.history[wid]) > self.history_limit:
                self.history[wid] = self.history[wid][-self.history_limit:]
            if wid in self.workers:
                self.workers[wid]["last_seen"] = now
                self.workers[wid]["active"] = True
            await self._broadcast({"type": "stats", "worker_id": wid, "stats": entry},
                                  exclude=ws)
        elif mtype == "heartbeat":
            if wid in self.workers:
                self.workers[wid]["last_seen"] = now
                self.workers[wid]["active"] = True
            await ws.send_json({"type": "heartbeat_ack", "time": now})
        elif mtype == "get_history":
            hist = self.history.get(wid) if wid else self.history
            await ws.send_json({"type": "history", "worker_id": wid, "history": hist})
        elif mtype == "get_workers":
            self._mark_stale()
            await ws.send_json({"type": "workers", "workers": self.workers})

    def _mark_stale(self) -> None:
        now = time.time()
        for w in self.workers.values():
            if now - w.get("last_seen", 0) > STALE_AFTER_S:
                w["active"] = False

    async def _broadcast(self, msg: Dict[str, Any], exclude=None) -> None:
        dead = []
        for c in self._clients:
            if c is exclude:
                continue
            try:
                await c.send_json(msg)
            except Exception:
                dead.append(c)
        for c in dead:
            self._clients.discard(c)

    async def _ws_handler(self, request):
        ws = web.WebSocketResponse()
        await ws.prepare(request)
        self._clients.add(ws)
        try:
            async for msg in ws:
                if msg.type == WSMsgType.TEXT:
                    try:
                        await self._handle_message(ws, json.loads(msg.data))
                    except (json.JSONDecodeError, KeyError):
                        await ws.send_json({"type": "error", "error": "bad message"})
                elif msg.type == WSMsgType.ERROR:
                    break
        finally:
            self._clients.discard(ws)
        return ws

    async def _http_status(self, request):
        self._mark_stale()
        return web.json_response({"workers": self.workers,
                                  "n_clients": len(self._clients)})

    def persist(self) -> None:
        if self.persist_path:
            self.persist_path.write_text(
                json.dumps({"workers": self.workers, "history": self.history}))

    # ---- lifecycle ----
    async def start(self) -> None:
        app = web.Application()
        app.router.add_get("/ws", self._ws_handler)
        app.router.add_get("/status", self._http_status)
        self._runner = web.AppRunner(app)
        await self._runner.setup()
        site = web.TCPSite(self._runner, self.host, self.port)
        await site.start()

    async def stop(self) -> None:
        self.persist()
        if self._runner:
            await self._runner.cleanup()

    def run_forever(self) -> None:  # pragma: no cover
        async def _main():
            await self.start()
            print(f"StatsServer on ws://{self.host}:{self.port}/ws")
            while True:
                await asyncio.sleep(10)
                self.persist()

        asyncio.run(_main())


def main(argv=None) -> None:  # pragma: no cover
    import argparse

    p = argparse.ArgumentParser(description="Training stats WebSocket server")
    p.add_argument("--host", default="127.0.0.1")
    p.add_argument("--port", type=int, default=8765)
    p.add_argument("--persist", default=None)
    args = p.parse_args(argv)
    StatsServer(args.host, args.port, persist_path=args.persist).run_forever()


if __name__ == "__main__":  # pragma: no cover
    main()
