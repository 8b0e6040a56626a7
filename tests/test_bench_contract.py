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
