"""Driver-contract guard: `python bench.py` must emit exactly one JSON
line with the agreed schema (the round driver parses this)."""
import json
import subprocess
import sys
from pathlib import Path

REPO = Path(__file__).resolve().parent.parent


def test_bench_json_contract():
    out = subprocess.run(
        [sys.executable, str(REPO / "bench.py"), "--steps", "3",
         "--warmup", "1", "--device", "cpu", "--msg-bytes", "2097152"],
        capture_output=True, text=True, timeout=180, cwd=REPO,
    )
    assert out.returncode == 0, out.stderr[-800:]
    json_lines = [l for l in out.stdout.splitlines() if l.startswith("{")]
    assert len(json_lines) == 1, out.stdout
    d = json.loads(json_lines[0])
    for key in ("metric", "value", "unit", "n_gpus", "steps", "warmup",
                "ms_per_step", "higher_is_better", "scaling", "vs_baseline",
                "dtype", "data", "config"):
        assert key in d, f"missing {key}"
    assert d["n_gpus"] == 1 and d["steps"] == 3 and d["warmup"] == 1
    assert d["unit"] == "GB/s" and d["higher_is_better"] is True
    assert d["scaling"] == "weak" and d["data"] == "synthetic"
    assert d["value"] > 0 and d["ms_per_step"] > 0
    assert "model" in d["config"] and "parallelism" in d["config"]


def test_bench_two_rank_cpu_torchrun():
    """The driver launches N>1 via torch.distributed.run (gloo rendezvous
    for worker-address exchange); validate that whole path on CPU so a
    contract break never first appears on the 8-GPU round-end node."""
    out = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
         "--master-port", "29871", str(REPO / "bench.py"), "--gpus", "2",
         "--steps", "2", "--warmup", "1", "--device", "cpu",
         "--msg-bytes", "1048576", "--lat-iters", "20"],
        capture_output=True, text=True, timeout=300, cwd=REPO,
    )
    assert out.returncode == 0, (out.stdout[-500:], out.stderr[-1200:])
    json_lines = [l for l in out.stdout.splitlines() if l.startswith("{")]
    assert len(json_lines) == 1, out.stdout
    d = json.loads(json_lines[0])
    assert d["n_gpus"] == 2
    assert d["config"]["endpoints"] == 2
    assert d["value"] > 0
    assert d["config"]["pingpong_64B_half_rtt_us"] > 0


def test_bench_cli_two_process_roles(tmp_path):
    """role=server + role=client in separate processes over localhost
    (socket mode), one scenario, JSON report written by the client."""
    port = "18441"
    srv = subprocess.Popen(
        [sys.executable, "-m", "starway_amd.bench", "--role", "server",
         "--addr", "127.0.0.1", "--port", port],
        cwd=REPO, stdout=subprocess.PIPE, stderr=subprocess.STDOUT,
        text=True)
    try:
        out_file = tmp_path / "r.json"
        cli = subprocess.run(
            [sys.executable, "-m", "starway_amd.bench", "--role", "client",
             "--server-host", "127.0.0.1", "--port", port,
             "--scenarios", "pingpong-flag", "--flag-iterations", "50",
             "--flag-warmup", "5", "--output", str(out_file)],
            capture_output=True, text=True, timeout=120, cwd=REPO)
        assert cli.returncode == 0, cli.stdout[-800:] + cli.stderr[-400:]
        report = json.loads(out_file.read_text())
        assert report["scenarios"][0]["metrics"]["avg_rtt_us"] > 0
        assert srv.wait(timeout=60) == 0
    finally:
        if srv.poll() is None:
            srv.kill()
            srv.wait()


def test_bench_cli_worker_address_mode(tmp_path):
    """Hex worker-address handshake through the CLI (reference C20
    surface): server prints the blob, client connects with it."""
    srv = subprocess.Popen(
        [sys.executable, "-u", "-m", "starway_amd.bench", "--role",
         "server", "--listen-mode", "worker"],
        cwd=REPO, stdout=subprocess.PIPE, stderr=subprocess.STDOUT,
        text=True)
    try:
        import time as _t
        blob_hex = None
        deadline = _t.time() + 60
        while _t.time() < deadline and blob_hex is None:
            line = srv.stdout.readline()
            if "worker address:" in line:
                blob_hex = line.rsplit(":", 1)[1].strip()
        assert blob_hex, "server never printed its worker address"
        out_file = tmp_path / "r.json"
        cli = subprocess.run(
            [sys.executable, "-m", "starway_amd.bench", "--role", "client",
             "--connect-mode", "worker", "--worker-address", blob_hex,
             "--scenarios", "small-messages", "--small-iterations", "3",
             "--small-warmup", "1", "--output", str(out_file)],
            capture_output=True, text=True, timeout=120, cwd=REPO)
        assert cli.returncode == 0, cli.stdout[-800:] + cli.stderr[-400:]
        report = json.loads(out_file.read_text())
        assert report["scenarios"][0]["metrics"]["messages_per_second"] > 0
        assert srv.wait(timeout=60) == 0
    finally:
        if srv.poll() is None:
            srv.kill()
            srv.wait()
