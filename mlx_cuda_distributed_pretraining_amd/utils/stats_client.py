"""WebSocket stats mesh — client side + per-rank metrics collection.

Parity surface: /root/reference/stats_client.py (threaded StatsClient with
reconnect + message buffering, heartbeats every 10 s; WorkerMetricsCollector
aggregating GPU util/memory/throughput). GPU metrics here come from
torch.cuda / rocm-smi instead of Metal.
"""
from __future__ import annotations

import asyncio
import json
import queue
import subprocess
import threading
import time
from typing import Any, Dict, Optional

try:
    import aiohttp

    HAVE_AIOHTTP = True
except Exception:  # pragma: no cover
    HAVE_AIOHTTP = False

HEARTBEAT_INTERVAL_S = 10.0
RECONNECT_DELAY_S = 5.0


def collect_gpu_metrics(device: int = 0) -> Dict[str, Any]:
    """Best-effort GPU utilization/memory snapshot (rocm-smi or torch.cuda)."""
    out: Dict[str, Any] = {}
    try:
        import torch

        if torch.cuda.is_available():
            free, total = torch.cuda.mem_get_info(device)
            out["gpu_mem_used_gb"] = (total - free) / 2**30
            out["gpu_mem_total_gb"] = total / 2**30
            out["gpu_mem_allocated_gb"] = torch.cuda.memory_allocated(device) / 2**30
    except Exception:
        pass
    try:
        r = subprocess.run(
            ["rocm-smi", "--showuse", "--json"], capture_output=True, text=True, timeout=5
        )
        if r.returncode == 0:
            data = json.loads(r.stdout)
            for card, vals in data.items():
                if str(device) in card or card.endswith(str(device)):
                    use = vals.get("GPU use (%)")
                    if use is not None:
                        out["gpu_util_pct"] = float(use)
    except Exception:
        pass
    try:
        import psutil

        out["cpu_pct"] = psutil.cpu_percent(interval=None)
        out["ram_used_gb"] = psutil.virtual_memory().used / 2**30
    except Exception:
        pass
    return out


class WorkerMetricsCollector:
    """Rolling aggregation of training metrics for one worker/rank
    (reference stats_client.py WorkerMetricsCollector)."""

    def __init__(self, worker_id: str, device: int = 0):
        self.worker_id = worker_id
        self.device = device
        self.step = 0
        self.tokens_total = 0
        self._tps_window: list = []

    def update(self, step: int, loss: float, tokens: int, elapsed_s: float) -> Dict[str, Any]:
        self.step = step
        self.tokens_total += tokens
        tps = tokens / max(elapsed_s, 1e-9)
        self._tps_window.append(tps)
        if len(self._tps_window) > 50:
            self._tps_window = self._tps_window[-50:]
        stats = {
            "step": step,
            "loss": float(loss),
            "tokens_total": self.tokens_total,
            "tokens_per_sec": tps,
            "tokens_per_sec_avg": sum(self._tps_window) / len(self._tps_window),
            **collect_gpu_metrics(self.device),
        }
        return stats


class StatsClient:
    """Threaded WebSocket client: buffers messages while disconnected,
    reconnects with a fixed delay, sends heartbeats."""

    def __init__(self, url: str = "ws://127.0.0.1:8765/ws", worker_id: str = "rank0",
                 info: Optional[Dict[str, Any]] = None, buffer_limit: int = 1000):
        if not HAVE_AIOHTTP:  # pragma: no cover
            raise RuntimeError("aiohttp is required for StatsClient")
        self.url = url
        self.worker_id = worker_id
        self.info = info or {}
        self._out: "queue.Queue[Dict[str, Any]]" = queue.Queue(maxsize=buffer_limit)
        self._stop = threading.Event()
        self._thread: Optional[threading.Thread] = None
        self.connected = threading.Event()

    # ---- public API (thread-safe) ----
    def start(self) -> None:
        self._thread = threading.Thread(target=self._run, daemon=True)
        self._thread.start()

    def stop(self, timeout: float = 5.0) -> None:
        self._stop.set()
        if self._thread:
            self._thread.join(timeout)

    def send_stats(self, stats: Dict[str, Any]) -> None:
        self._enqueue({"type": "stats", "worker_id": self.worker_id, "stats": stats})

    def _enqueue(self, msg: Dict[str, Any]) -> None:
        try:
            self._out.put_nowait(msg)
        except queue.Full:  # drop oldest
            try:
                self._out.get_nowait()
                self._out.put_nowait(msg)
            except queue.Empty:
                pass

    # ---- connection loop ----
    def _run(self) -> None:
        asyncio.run(self._loop())

    async def _loop(self) -> None:
        while not self._stop.is_set():
            try:
                async with aiohttp.ClientSession() as session:
                    async with session.ws_connect(self.url, heartbeat=None) as ws:
                        await ws.send_json({"type": "register",
                                            "worker_id": self.worker_id,
                                            "info": self.info})
                        self.connected.set()
                        last_hb = time.time()
                        while True:
                            # drain the buffer before honoring stop(): the
                            # trainer enqueues its final step metrics moments
                            # before stopping, and dropping them loses the
                            # tail of every short run
                            try:
                                msg = self._out.get_nowait()
                                await ws.send_json(msg)
                                continue
                            except queue.Empty:
                                if self._stop.is_set():
                                    break
                                await asyncio.sleep(0.05)
                            if time.time() - last_hb > HEARTBEAT_INTERVAL_S:
                                await ws.send_json({"type": "heartbeat",
                                                    "worker_id": self.worker_id})
                                last_hb = time.time()
            except Exception:
                self.connected.clear()
                await asyncio.sleep(RECONNECT_DELAY_S)
        self.connected.clear()
