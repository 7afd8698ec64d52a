"""CLI smoke tests: run_dht + run_server boot a servable worker
(mirror the reference's out-of-band swarm bring-up, but self-contained)."""
import os
import signal
import subprocess
import sys
import time

import pytest
import torch


def test_cli_dht_and_server_serve_requests(tmp_path):
    env = dict(os.environ)
    env["PYTHONPATH"] = os.getcwd()
    dht = subprocess.Popen(
        [sys.executable, "-m", "bloombee_amd.cli.run_dht", "--port", "0"],
        stdout=subprocess.PIPE, stderr=subprocess.STDOUT, text=True, env=env)
    try:
        # parse the announced endpoint from the log line
        ep = None
        t0 = time.monotonic()
        while time.monotonic() - t0 < 30:
            line = dht.stdout.readline()
            if "DHT bootstrap node at" in line:
                part = line.split("DHT bootstrap node at")[1].split("—")[0]
                host, port = part.strip().rsplit(":", 1)
                ep = (host, int(port))
                break
        assert ep is not None, "bootstrap endpoint not announced"

        srv = subprocess.Popen(
            [sys.executable, "-m", "bloombee_amd.cli.run_server", "llama-tiny",
             "--initial-peers", f"{ep[0]}:{ep[1]}", "--block-indices", "0:4",
             "--device", "cpu", "--throughput", "1.0",
             "--attn-cache-tokens", "8192"],
            stdout=subprocess.PIPE, stderr=subprocess.STDOUT, text=True, env=env)
        try:
            from bloombee_amd.client import ClientConfig
            from bloombee_amd.models.auto import AutoDistributedModelForCausalLM

            cfg = ClientConfig(initial_peers=[ep])
            model = None
            t0 = time.monotonic()
            last = None
            while time.monotonic() - t0 < 60:
                try:
                    model = AutoDistributedModelForCausalLM.from_pretrained(
                        "llama-tiny", client_config=cfg, seed=0)
                    model.remote.manager.make_sequence()
                    break
                except Exception as e:  # noqa: BLE001
                    last = e
                    if model is not None:
                        model.remote.manager.shutdown()
                        model = None
                    time.sleep(1.0)
            assert model is not None, f"swarm never became routable: {last}"
            ids = torch.randint(0, 1000, (1, 5),
                                generator=torch.Generator().manual_seed(0))
            out = model.generate(ids, max_new_tokens=3)
            assert out.shape == (1, 8)
            model.remote.manager.shutdown()
        finally:
            srv.send_signal(signal.SIGINT)
            try:
                srv.wait(timeout=10)
            except subprocess.TimeoutExpired:
                srv.kill()
    finally:
        dht.terminate()
        try:
            dht.wait(timeout=5)
        except subprocess.TimeoutExpired:
            dht.kill()


def test_swarm_health_report():
    """Health monitor: coverage + per-server rows against a live 2-server
    loopback swarm (ref health website / DHT models registry)."""
    import torch

    from bloombee_amd.cli.health import swarm_health
    from bloombee_amd.net.dht import Dht
    from bloombee_amd.server import Server

    boot = Dht()
    s1 = Server("llama-tiny", initial_peers=[boot.endpoint],
                block_indices=(0, 2), device="cpu", seed=0,
                kv_max_tokens=1 << 12, update_period=5.0)
    s2 = Server("llama-tiny", initial_peers=[boot.endpoint],
                block_indices=(2, 4), device="cpu", seed=0,
                kv_max_tokens=1 << 12, update_period=5.0)
    s1.run_in_background()
    s2.run_in_background()
    try:
        h = swarm_health("llama-tiny", [boot.endpoint])
        assert h["complete"], h
        assert h["blocks"] == [1, 1, 1, 1]
        assert len(h["servers"]) == 2
        for s in h["servers"].values():
            assert s["rtt_ms"] is not None and s["rtt_ms"] < 2000
            assert s["cache_tokens_left"] is not None
    finally:
        s1.shutdown()
        s2.shutdown()
        boot.shutdown()


def test_identity_persistence_and_batch_cap(tmp_path):
    """--identity-path keeps the peer id stable across restarts; sessions
    above --max-batch-size are rejected."""
    import pytest
    import torch

    from bloombee_amd.net.dht import Dht
    from bloombee_amd.server import Server

    ident = tmp_path / "id"
    boot = Dht()
    s1 = Server("llama-tiny", initial_peers=[boot.endpoint],
                block_indices=(0, 4), device="cpu", seed=0,
                kv_max_tokens=1 << 12, identity_path=str(ident),
                max_batch_size=4)
    pid = s1.peer_id
    with pytest.raises(ValueError, match="max_batch_size"):
        s1.backend.open_session("big", batch_size=8, max_length=16)
    s1.backend.open_session("ok", batch_size=4, max_length=16)
    s1.backend.close_session("ok")
    s1.shutdown()

    s2 = Server("llama-tiny", initial_peers=[boot.endpoint],
                block_indices=(0, 4), device="cpu", seed=0,
                kv_max_tokens=1 << 12, identity_path=str(ident))
    assert s2.peer_id == pid
    s2.shutdown()
    boot.shutdown()


@pytest.mark.timeout(300)
def test_bench_swarm_mode_cpu_contract():
    """bench.py --mode swarm runs the REAL serving stack (Server + session +
    dist channels) and prints the driver's JSON contract line."""
    import json
    import subprocess
    import sys

    r = subprocess.run(
        [sys.executable, "bench.py", "--mode", "swarm", "--model",
         "llama-tiny", "--gpus", "1", "--steps", "3", "--warmup", "1",
         "--batch-per-gpu", "2", "--prompt", "16", "--device", "cpu"],
        capture_output=True, text=True, timeout=240,
        cwd=os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
    assert r.returncode == 0, r.stderr[-2000:]
    line = [l for l in r.stdout.splitlines() if l.startswith("{")][-1]
    out = json.loads(line)
    assert out["unit"] == "tokens/s" and out["value"] > 0
    assert "serving stack" in out["metric"]
    assert out["config"]["parallelism"] == "swarm-pp1"
