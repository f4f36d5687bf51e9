"""Observability layer: log parsing, plotting, monitoring, stats mesh."""
import asyncio
import json
import time

import pytest

from mlx_cuda_distributed_pretraining_amd.utils.log_parse import (
    parse_log_file, parse_log_line,
)

SAMPLE = (
    "Step 40: loss=2.345e+00 | ppl=10.43 | val_loss=2.100e+00 | val_ppl=8.17 "
    "| tok/s=123.45K | toks=81920 | lr=1.0e-03 | grad_norm=1.5e+00"
)


def test_parse_log_line_full():
    rec = parse_log_line(SAMPLE)
    assert rec.step == 40
    assert rec.loss == pytest.approx(2.345)
    assert rec.val_loss == pytest.approx(2.1)
    assert rec.ppl == pytest.approx(10.43)
    assert rec.val_ppl == pytest.approx(8.17)
    assert rec.tokens_per_sec == pytest.approx(123450.0)
    assert rec.toks == 81920
    assert rec.lr == pytest.approx(1e-3)
    assert rec.grad_norm == pytest.approx(1.5)


def test_parse_log_line_ignores_noise():
    assert parse_log_line("random info line") is None
    rec = parse_log_line("Step 3: loss=1.000e+00 | ppl=2.72 | tok/s=5.00K | toks=100 | lr=1.0e-04")
    assert rec.val_loss is None and rec.step == 3


def _write_log(tmp_path, n=20):
    run = tmp_path / "run"
    run.mkdir(parents=True)
    with open(run / "log.txt", "w") as f:
        f.write("2026-01-01 starting run\n")
        for i in range(n):
            loss = 3.0 - i * 0.1
            line = (f"Step {i}: loss={loss:.3e} | ppl={2.718**loss:.2f} "
                    f"| tok/s=100.00K | toks={i*1000} | lr=1.0e-03")
            if i % 5 == 4:
                line += f" | val_loss={loss + 0.1:.3e} | val_ppl=9.99"
            f.write(line + "\n")
    return run


def test_parse_log_file_and_plot(tmp_path):
    run = _write_log(tmp_path)
    records = parse_log_file(run)
    assert len(records) == 20
    assert records[-1].step == 19

    from mlx_cuda_distributed_pretraining_amd.utils.plotting import (
        plot_run, records_to_csv,
    )

    fig = plot_run(run)
    assert fig is not None
    assert (run / "loss_curve.png").exists()
    records_to_csv(records, tmp_path / "out.csv")
    lines = (tmp_path / "out.csv").read_text().splitlines()
    assert len(lines) == 21  # header + 20


def test_compare_runs(tmp_path):
    r1 = _write_log(tmp_path / "a")
    r2 = _write_log(tmp_path / "b")
    from mlx_cuda_distributed_pretraining_amd.utils.plotting import compare_runs

    out = tmp_path / "cmp.png"
    compare_runs([str(r1), str(r2)], out_path=str(out))
    assert out.exists()


def test_training_monitor_incremental(tmp_path):
    run = tmp_path / "run"
    run.mkdir()
    log = run / "log.txt"
    log.write_text("Step 0: loss=3.000e+00 | ppl=20.09 | tok/s=1.00K | toks=100 | lr=1.0e-03\n")

    from mlx_cuda_distributed_pretraining_amd.utils.monitoring import TrainingMonitor

    mon = TrainingMonitor(run)
    new = mon.poll()
    assert len(new) == 1
    with open(log, "a") as f:
        f.write("Step 1: loss=2.500e+00 | ppl=12.18 | val_loss=2.600e+00 | tok/s=2.00K | toks=200 | lr=1.0e-03\n")
    new = mon.poll()
    assert len(new) == 1 and new[0].step == 1
    s = mon.summary()
    assert s["steps_seen"] == 2
    assert s["last_loss"] == pytest.approx(2.5)
    assert s["best_val_loss"] == pytest.approx(2.6)
    assert s["total_tokens"] == 200


@pytest.mark.timeout(60)
def test_stats_mesh_roundtrip(tmp_path):
    """Server + threaded client: register, stats, heartbeat, history query."""
    from mlx_cuda_distributed_pretraining_amd.utils.stats_client import StatsClient
    from mlx_cuda_distributed_pretraining_amd.utils.stats_server import StatsServer

    persist = tmp_path / "stats.json"

    async def scenario():
        server = StatsServer(port=18765, persist_path=str(persist))
        await server.start()
        client = StatsClient("ws://127.0.0.1:18765/ws", worker_id="w0",
                             info={"gpu": "MI355X"})
        client.start()
        # poll asynchronously — a blocking Event.wait would stall the server's
        # event loop (it runs in this same loop)
        for _ in range(200):
            if client.connected.is_set():
                break
            await asyncio.sleep(0.05)
        assert client.connected.is_set(), "client never connected"
        client.send_stats({"step": 1, "loss": 2.0})
        for _ in range(100):
            if server.history.get("w0"):
                break
            await asyncio.sleep(0.05)
        assert server.history["w0"], "stats never arrived"
        assert server.history["w0"][0]["loss"] == 2.0
        assert server.workers["w0"]["info"]["gpu"] == "MI355X"
        client.stop()
        await server.stop()

    asyncio.run(scenario())
    saved = json.loads(persist.read_text())
    assert "w0" in saved["workers"]


def test_metrics_collector():
    from mlx_cuda_distributed_pretraining_amd.utils.stats_client import (
        WorkerMetricsCollector,
    )

    c = WorkerMetricsCollector("rank0")
    s1 = c.update(step=1, loss=2.0, tokens=1000, elapsed_s=0.5)
    s2 = c.update(step=2, loss=1.9, tokens=1000, elapsed_s=0.5)
    assert s2["tokens_total"] == 2000
    assert s1["tokens_per_sec"] == pytest.approx(2000.0)
    assert s2["tokens_per_sec_avg"] == pytest.approx(2000.0)


@pytest.mark.timeout(120)
def test_trainer_streams_to_stats_mesh(tmp_path):
    """End-to-end: a Trainer with logging.stats_url streams per-step metrics
    to a live StatsServer."""
    import threading
    from pathlib import Path

    from mlx_cuda_distributed_pretraining_amd.core.config import Config
    from mlx_cuda_distributed_pretraining_amd.core.trainer import Trainer
    from mlx_cuda_distributed_pretraining_amd.utils.stats_server import StatsServer

    server = StatsServer(port=18766)
    loop_box = {}

    def run_server():
        async def main():
            await server.start()
            while not loop_box.get("stop"):
                await asyncio.sleep(0.1)
            await server.stop()

        asyncio.run(main())

    t = threading.Thread(target=run_server, daemon=True)
    t.start()
    # wait for the server to actually accept connections (a fixed sleep is
    # flaky under full-suite load: the client connects once, silently)
    import socket

    for _ in range(100):
        try:
            with socket.create_connection(("127.0.0.1", 18766), timeout=0.2):
                break
        except OSError:
            time.sleep(0.1)

    repo = Path(__file__).resolve().parent.parent
    cfg = Config.from_yaml(repo / "configs" / "model-config-sample.yaml")
    cfg.name = "stats-run"
    cfg.overwrite = True
    cfg.data.synthetic = True
    cfg.training.hyperparameters["iters"] = 3
    cfg.training.hyperparameters["batch_size"] = 2
    cfg.data.preprocessing["max_context_size"] = 32
    cfg.logging.steps = {"logging_interval": 1, "checkpoint_interval": 0,
                         "validation_interval": 0}
    cfg.logging.stats_url = "ws://127.0.0.1:18766/ws"
    trainer = Trainer(cfg, runs_root=str(tmp_path / "runs"))
    trainer.train()
    for _ in range(50):
        if len(server.history.get("rank0", [])) >= 3:
            break
        time.sleep(0.1)
    loop_box["stop"] = True
    t.join(5)
    hist = server.history.get("rank0", [])
    assert len(hist) >= 3
    assert hist[-1]["step"] == 3 and "loss" in hist[-1]
