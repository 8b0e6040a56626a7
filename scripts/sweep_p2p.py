#!/usr/bin/env python3
"""Message-size sweep for the BASELINE table: tagged pingpong GB/s and p50
half-RTT from 64 B to 256 MB.

Modes:
  * single process (default): Server+Client loopback in one process; on a
    GPU host buffers are HIP device tensors (same-GPU HBM copy path), else
    CPU numpy over localhost TCP.
  * --cross-process: the server lives in a spawned subprocess => the full
    worker-address handshake + hipIpc import path, still on one GPU when
    only one is visible.

Writes a JSON + text table to --out (default gpurun_out/sweep.json).
"""
from __future__ import annotations

import os

# ROCm multiplexes streams onto GPU_MAX_HW_QUEUES hardware queues
# (default 4); oversubscription time-slices co-mapped streams at ~ms
# granularity. Must be set before the FIRST HIP init in the process
# (torch's or ours) — see ROUND2_NOTES.md "hardware-queue starvation".
os.environ.setdefault("GPU_MAX_HW_QUEUES", "16")

import argparse
import asyncio
import json
import multiprocessing as mp
import os
import sys
import time
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))

import numpy as np  # noqa: E402

SIZES = [64, 1024, 64 * 1024, 1 << 20, 16 << 20, 256 << 20]
FULL = (1 << 64) - 1


def pick_iters(size: int) -> tuple[int, int]:
    if size <= 64 * 1024:
        return 200, 1000
    if size <= (16 << 20):
        return 20, 100
    return 5, 20


def _alloc(n, device, fill=None):
    if device == "cpu":
        a = np.empty(n, dtype=np.uint8)
        if fill is not None:
            a.fill(fill)
        return a
    import torch

    t = torch.empty(n, dtype=torch.uint8, device="cuda")
    if fill is not None:
        t.fill_(fill)
    torch.cuda.synchronize()
    return t


def _server_proc(port: int, device: str, sizes: list[int]):
    import starway_amd as sw

    async def inner():
        server = sw.Server()
        server.listen("127.0.0.1", port)
        for size in sizes:
            warmup, iters = pick_iters(size)
            rbuf = _alloc(size, device)
            sbuf = _alloc(size, device, fill=1)
            ep = None
            for i in range(warmup + iters):
                await server.arecv(rbuf, 1, FULL)
                if ep is None:
                    ep = next(iter(server.list_clients()))
                await server.asend(ep, sbuf, 2)
        # Close without flush would drop the last in-flight reply (the
        # delivery contract the flush tests pin down).
        await server.aflush()
        await server.aclose()

    asyncio.run(inner())


async def run_sweep(args) -> list[dict]:
    import starway_amd as sw

    device = args.device
    results = []

    if args.cross_process:
        ctx = mp.get_context("spawn")
        port = 41000 + os.getpid() % 1000
        p = ctx.Process(target=_server_proc, args=(port, device, SIZES))
        p.start()
        client = sw.Client()
        await client.aconnect("127.0.0.1", port)

        async def one_rt(sbuf, rbuf):
            fut = client.arecv(rbuf, 2, FULL)
            await client.asend(sbuf, 1)
            await fut

        for size in SIZES:
            warmup, iters = pick_iters(size)
            sbuf = _alloc(size, device, fill=3)
            rbuf = _alloc(size, device)
            for _ in range(warmup):
                await one_rt(sbuf, rbuf)
            samples = []
            for _ in range(iters):
                t0 = time.perf_counter()
                await one_rt(sbuf, rbuf)
                samples.append(time.perf_counter() - t0)
            results.append(_mk_result(size, samples, "cross-process"))
            print(_fmt(results[-1]), flush=True)
        await client.aclose()
        p.join(timeout=60)
        if p.is_alive():
            p.kill()
            p.join()
    else:
        server = sw.Server()
        client = sw.Client()
        addr = server.listen_address()
        await client.aconnect_address(addr)
        for size in SIZES:
            warmup, iters = pick_iters(size)
            sbuf = _alloc(size, device, fill=3)
            rbuf = _alloc(size, device)
            pong = _alloc(size, device, fill=1)
            ep = next(iter(server.list_clients()))

            async def one_rt():
                sfut = server.arecv(rbuf, 1, FULL)
                cfut = client.arecv(pong, 2, FULL)
                await client.asend(sbuf, 1)
                await sfut
                await server.asend(ep, sbuf, 2)
                await cfut

            for _ in range(warmup):
                await one_rt()
            samples = []
            for _ in range(iters):
                t0 = time.perf_counter()
                await one_rt()
                samples.append(time.perf_counter() - t0)
            results.append(_mk_result(size, samples, "loopback"))
            print(_fmt(results[-1]), flush=True)
        await client.aclose()
        await server.aclose()
    return results


def _mk_result(size: int, samples: list[float], mode: str) -> dict:
    arr = np.array(samples)
    p50 = float(np.percentile(arr, 50))
    return {
        "size_bytes": size,
        "mode": mode,
        "iters": len(samples),
        "p50_rtt_us": p50 * 1e6,
        "p50_half_rtt_us": p50 * 1e6 / 2,
        "gbps_per_direction": size / (p50 / 2) / 1e9,
    }


def _fmt(r: dict) -> str:
    return (f"{r['size_bytes']:>12} B  p50 half-RTT {r['p50_half_rtt_us']:>10.1f} us  "
            f"{r['gbps_per_direction']:>8.2f} GB/s/dir  [{r['mode']}]")


def run_raw_latency(device: str) -> list[dict]:
    """Raw-callback pingpong (no asyncio): the transport's latency floor.
    Uses threading.Event for completion wakeups."""
    import threading

    import starway_amd as sw

    server, client = sw.Server(), sw.Client()
    addr = server.listen_address()
    ev = threading.Event()
    client._client.connect_address(addr, lambda s: ev.set())
    assert ev.wait(10)
    ep = None
    for _ in range(1000):
        eps = server.list_clients()
        if eps:
            ep = next(iter(eps))
            break
        time.sleep(0.001)

    results = []
    for size in [64, 4096, 65536]:
        ping_out = _alloc(size, device, fill=1)
        pong_out = _alloc(size, device, fill=2)
        srv_in = _alloc(size, device)
        cli_in = _alloc(size, device)
        warmup, iters = pick_iters(size)
        samples = []

        def one_rt():
            # Strict RTT: the server's pong is issued from its ping-recv
            # completion callback (runs on the server's progress thread).
            done = threading.Event()

            def on_ping(tag, length):
                server._server.send(ep, pong_out, 2,
                                    lambda: None, lambda e: None)

            server._server.recv(srv_in, 1, FULL, on_ping, lambda e: None)
            client._client.recv(cli_in, 2, FULL,
                                lambda t, l: done.set(),
                                lambda e: done.set())
            client._client.send(ping_out, 1, lambda: None, lambda e: None)
            assert done.wait(10)

        for _ in range(warmup):
            one_rt()
        for _ in range(iters):
            t0 = time.perf_counter()
            one_rt()
            samples.append(time.perf_counter() - t0)
        results.append(_mk_result(size, samples, "raw-callback"))
        print(_fmt(results[-1]), flush=True)

    ev2 = threading.Event()
    client._client.close(lambda: ev2.set())
    ev2.wait(10)
    ev3 = threading.Event()
    server._server.close(lambda: ev3.set())
    ev3.wait(10)
    return results


def run_raw_rate(device: str, msg_bytes: int = 1024, depth: int = 64,
                 total: int = 20000) -> list[dict]:
    """Raw-callback small-message rate (no asyncio): `depth` messages kept
    in flight; measures sustained msgs/s to locate the engine-side cap."""
    import threading

    import starway_amd as sw

    server, client = sw.Server(), sw.Client()
    addr = server.listen_address()
    ev = threading.Event()
    client._client.connect_address(addr, lambda s: ev.set())
    assert ev.wait(10)

    sbufs = [_alloc(msg_bytes, device, fill=i % 251) for i in range(depth)]
    rbufs = [_alloc(msg_bytes, device) for i in range(depth)]
    done = threading.Semaphore(0)
    remaining = [total]   # completions outstanding
    posted = [0]          # recvs posted so far
    lock = threading.Lock()

    def recv_cb(slot):
        def cb(tag, length):
            repost = False
            with lock:
                remaining[0] -= 1
                if remaining[0] <= 0:
                    done.release()
                    return
                if posted[0] < total:
                    posted[0] += 1
                    repost = True
            if repost:
                server._server.recv(rbufs[slot], 0, 0, recv_cb(slot),
                                    lambda e: done.release())
        return cb

    t0 = time.perf_counter()
    for i in range(depth):
        with lock:
            posted[0] += 1
        server._server.recv(rbufs[i], 0, 0, recv_cb(i),
                            lambda e: done.release())

    sent = [0]

    def send_more(slot):
        def cb():
            with lock:
                if sent[0] >= total:
                    return
                sent[0] += 1
            client._client.send(sbufs[slot], sent[0], cb,
                                lambda e: None)
        return cb

    for i in range(depth):
        with lock:
            sent[0] += 1
        client._client.send(sbufs[i], i, send_more(i), lambda e: None)

    assert done.acquire(timeout=120)
    dt = time.perf_counter() - t0
    rate = total / dt
    print(f"raw rate: {rate:,.0f} msgs/s ({msg_bytes} B x {total}, depth "
          f"{depth}, {dt:.2f}s)", flush=True)

    ev2 = threading.Event()
    client._client.close(lambda: ev2.set())
    ev2.wait(10)
    ev3 = threading.Event()
    server._server.close(lambda: ev3.set())
    ev3.wait(10)
    return [{"mode": "raw-rate", "msgs_per_s": rate, "msg_bytes": msg_bytes,
             "depth": depth}]


def main():
    if os.environ.get("SW_FH"):
        import faulthandler

        faulthandler.dump_traceback_later(25, exit=True)
    ap = argparse.ArgumentParser()
    ap.add_argument("--device", default=None)
    ap.add_argument("--cross-process", action="store_true")
    ap.add_argument("--raw", action="store_true",
                    help="raw-callback latency floor (no asyncio)")
    ap.add_argument("--rate", action="store_true",
                    help="raw-callback small-message rate probe")
    ap.add_argument("--out", default="gpurun_out/sweep.json")
    args = ap.parse_args()
    if args.device is None:
        try:
            import torch

            args.device = "cuda" if torch.cuda.is_available() else "cpu"
        except ImportError:
            args.device = "cpu"

    if args.rate:
        results = run_raw_rate(args.device)
    elif args.raw:
        results = run_raw_latency(args.device)
    else:
        results = asyncio.run(run_sweep(args))
    out = Path(args.out)
    out.parent.mkdir(parents=True, exist_ok=True)
    out.write_text(json.dumps(
        {"device": args.device,
         "cross_process": args.cross_process,
         "results": results}, indent=2))
    print(f"wrote {out}")


if __name__ == "__main__":
    main()
