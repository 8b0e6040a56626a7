"""Benchmark CLI — ``python -m starway_amd.bench``.

Role/orchestration parity with the reference CLI (reference
src/starway/bench.py: roles server|client|loopback, a JSON-over-tagged-
messages control protocol with READY/DONE handshakes, per-scenario
overrides, JSON reports). Differences: ``--device cpu|cuda`` selects host
numpy buffers vs HIP device tensors (the reference had no GPU path), and
``--tls`` is gone — the transport is always our native TCP+xGMI stack.
"""
from __future__ import annotations

import argparse
import gc
import asyncio
import json
import os
import sys
import time
from pathlib import Path
from typing import Any, Mapping, Sequence

import numpy as np

from . import Client, Server
from .benchmarks import get_scenario, list_scenarios
from .benchmarks.scenarios import (
    CONTROL_TAG,
    DONE_TAG,
    READY_TAG,
    SCENARIOS,
    TAG_MASK,
    ScenarioResult,
)


def parse_size(value: str) -> int:
    text = value.strip().lower().replace("_", "")
    suffixes = {
        "k": 1024, "kb": 1024, "ki": 1024, "kib": 1024,
        "m": 1024 ** 2, "mb": 1024 ** 2, "mi": 1024 ** 2, "mib": 1024 ** 2,
        "g": 1024 ** 3, "gb": 1024 ** 3, "gi": 1024 ** 3, "gib": 1024 ** 3,
    }
    for suffix, mult in suffixes.items():
        if text.endswith(suffix):
            return int(float(text[: -len(suffix)]) * mult)
    return int(float(text))


def parse_worker_address(value: str) -> bytes:
    cleaned = value.replace(":", "").replace(" ", "").strip()
    return bytes.fromhex(cleaned)


def build_parser() -> argparse.ArgumentParser:
    p = argparse.ArgumentParser(
        prog="starway_amd.bench", description="starway_amd benchmark suite"
    )
    p.add_argument("--role", choices=("server", "client", "loopback"),
                   required=True)
    p.add_argument("--addr", default="0.0.0.0",
                   help="Server listen address (socket mode).")
    p.add_argument("--port", type=int, default=17777)
    p.add_argument("--server-host", default="127.0.0.1")
    p.add_argument("--listen-mode", choices=("socket", "worker"),
                   default="socket")
    p.add_argument("--connect-mode", choices=("socket", "worker"),
                   default="socket")
    p.add_argument("--worker-address",
                   help="Hex worker address for connect-mode=worker.")
    p.add_argument("--device", default=None,
                   help="Buffer device for all scenarios: cpu (default) or "
                        "cuda / cuda:N.")
    p.add_argument("--tls",
                   help="Accepted for reference-CLI compatibility; the "
                        "native transport selects tcp/shm/cma/xgmi "
                        "automatically (use STARWAY_* env knobs to pin).")
    p.add_argument("--scenarios", nargs="*",
                   help="Scenarios (default all): " + ", ".join(list_scenarios()))
    p.add_argument("--large-bytes", type=parse_size)
    p.add_argument("--large-iterations", type=int)
    p.add_argument("--large-warmup", type=int)
    p.add_argument("--small-bytes", type=parse_size)
    p.add_argument("--small-iterations", type=int)
    p.add_argument("--small-warmup", type=int)
    p.add_argument("--small-concurrency", type=int)
    p.add_argument("--flag-iterations", type=int)
    p.add_argument("--flag-warmup", type=int)
    p.add_argument("--stream-bytes", type=parse_size)
    p.add_argument("--stream-iterations", type=int)
    p.add_argument("--stream-warmup", type=int)
    p.add_argument("--output", type=Path)
    p.add_argument("--store-trace", action="store_true")
    return p


def scenario_plan(args: argparse.Namespace) -> list[tuple[str, dict[str, Any]]]:
    if not args.scenarios or (
        len(args.scenarios) == 1 and args.scenarios[0].lower() == "all"
    ):
        requested: Sequence[str] = list_scenarios()
    else:
        requested = args.scenarios

    plan: list[tuple[str, dict[str, Any]]] = []
    for name in requested:
        if name not in SCENARIOS:
            raise ValueError(
                f"Unknown scenario '{name}'. Available: {', '.join(list_scenarios())}"
            )
        ov: dict[str, Any] = {}
        if name == "large-array":
            ov = {"message_bytes": args.large_bytes,
                  "iterations": args.large_iterations,
                  "warmup": args.large_warmup}
        elif name == "small-messages":
            ov = {"message_bytes": args.small_bytes,
                  "iterations": args.small_iterations,
                  "warmup_batches": args.small_warmup,
                  "concurrency": args.small_concurrency}
        elif name == "pingpong-flag":
            ov = {"iterations": args.flag_iterations,
                  "warmup": args.flag_warmup}
        elif name == "streaming-duplex":
            ov = {"message_bytes": args.stream_bytes,
                  "iterations": args.stream_iterations,
                  "warmup": args.stream_warmup}
        ov = {k: v for k, v in ov.items() if v is not None}
        if args.device:
            ov["device"] = args.device
        plan.append((name, ov))
    return plan


def encode_control(payload: Mapping[str, Any]) -> np.ndarray:
    data = json.dumps(payload, separators=(",", ":"), sort_keys=True).encode()
    return np.frombuffer(data, dtype=np.uint8).copy()


def decode_control(buffer: np.ndarray, length: int) -> Mapping[str, Any]:
    return json.loads(memoryview(buffer)[:length].tobytes().decode())


class ClientSession:
    def __init__(self, client: Client):
        self.client = client
        self.tag_mask = TAG_MASK
        self._ready = np.zeros(1, dtype=np.uint8)
        self._done = np.zeros(1, dtype=np.uint8)

    async def send_control(self, payload: Mapping[str, Any]) -> None:
        await self.client.asend(encode_control(payload), CONTROL_TAG)
        await self.flush()

    async def wait_ready(self) -> None:
        await self.client.arecv(self._ready, READY_TAG, self.tag_mask)

    async def wait_done(self) -> None:
        await self.client.arecv(self._done, DONE_TAG, self.tag_mask)

    async def flush(self) -> None:
        await self.client.aflush()


class ClientScenarioContext:
    def __init__(self, session: ClientSession):
        self._session = session
        self.client = session.client
        self.tag_mask = session.tag_mask

    async def flush(self) -> None:
        await self._session.flush()


class ServerSession:
    def __init__(self, server: Server, endpoint):
        self.server = server
        self.endpoint = endpoint
        self.tag_mask = TAG_MASK
        self._ready = np.array([1], dtype=np.uint8)
        self._done = np.array([1], dtype=np.uint8)

    async def recv_control(self, max_bytes: int = 4096) -> Mapping[str, Any]:
        buffer = np.empty(max_bytes, dtype=np.uint8)
        _, length = await self.server.arecv(buffer, CONTROL_TAG, self.tag_mask)
        return decode_control(buffer, length)

    async def send_ready(self) -> None:
        await self.server.asend(self.endpoint, self._ready, READY_TAG)

    async def send_done(self) -> None:
        await self.server.asend(self.endpoint, self._done, DONE_TAG)


class ServerScenarioContext:
    def __init__(self, session: ServerSession):
        self._session = session
        self.server = session.server
        self.endpoint = session.endpoint
        self.tag_mask = session.tag_mask

    async def signal_ready(self) -> None:
        await self._session.send_ready()

    async def flush_endpoint(self) -> None:
        await self.server.aflush_ep(self.endpoint)


async def run_client(args: argparse.Namespace) -> list[ScenarioResult]:
    client = Client()
    results: list[ScenarioResult] = []
    try:
        if args.connect_mode == "worker":
            if not args.worker_address:
                raise ValueError("--worker-address required for connect-mode=worker")
            addr = parse_worker_address(args.worker_address)
            await client.aconnect_address(addr)
            print(f"[client] Connected via worker address ({len(addr)} bytes).")
        else:
            await client.aconnect(args.server_host, args.port)
            print(f"[client] Connected to {args.server_host}:{args.port}.")

        session = ClientSession(client)
        context = ClientScenarioContext(session)

        for name, overrides in scenario_plan(args):
            scenario = get_scenario(name)
            print(f"[client] Starting '{name}' with {overrides or 'defaults'}.")
            await session.send_control({"scenario": name, "config": overrides})
            await session.wait_ready()
            # Bench hygiene: a gen-2 GC pause (~40 ms, triggered by the
            # future/closure churn of concurrent-op scenarios) would
            # dominate a batch sample; collect up front, disable during
            # the timed region.
            gc.collect()
            gc.disable()
            try:
                result = await scenario.client_runner(context, overrides)
            finally:
                gc.enable()
            results.append(result)
            await session.wait_done()
            print(f"[client] Completed '{name}'.")

        await session.send_control({"scenario": "__shutdown__"})
        await session.flush()
    finally:
        await client.aclose()
    return results


async def run_server(args: argparse.Namespace) -> None:
    server = Server()
    loop = asyncio.get_running_loop()
    accepted: asyncio.Queue = asyncio.Queue()
    server.set_accept_cb(
        lambda ep: loop.call_soon_threadsafe(accepted.put_nowait, ep)
    )

    if args.listen_mode == "worker":
        worker_address = server.listen_address()
        print(f"[server] Listening via worker address: {worker_address.hex()}")
    else:
        server.listen(args.addr, args.port)
        print(f"[server] Listening on {args.addr}:{args.port}")

    endpoint = await accepted.get()
    print("[server] Client accepted.")
    session = ServerSession(server, endpoint)
    try:
        while True:
            control = await session.recv_control()
            name = control.get("scenario")
            if name == "__shutdown__":
                print("[server] Shutdown request received.")
                break
            if name not in SCENARIOS:
                raise ValueError(f"Unknown scenario '{name}' from client.")
            overrides = control.get("config", {})
            scenario = get_scenario(name)
            print(f"[server] Running '{name}' with {overrides or 'defaults'}.")
            await scenario.server_runner(ServerScenarioContext(session), overrides)
            await session.send_done()
            print(f"[server] Scenario '{name}' completed.")
    finally:
        await server.aclose()
        print("[server] Closed.")


async def run_loopback(args: argparse.Namespace) -> list[ScenarioResult]:
    client_done: asyncio.Future = asyncio.get_running_loop().create_future()

    async def client_task() -> None:
        try:
            client_done.set_result(await run_client(args))
        except Exception as exc:
            if not client_done.done():
                client_done.set_exception(exc)
            raise

    server_task = asyncio.create_task(run_server(args))
    client_fut = asyncio.create_task(client_task())
    try:
        results = await client_done
    finally:
        await client_fut
        await server_task
    return results


def dump_results(results: Sequence[ScenarioResult], args: argparse.Namespace) -> None:
    if not results:
        print("No results collected.")
        return
    print("\n=== Benchmark Results ===")
    for result in results:
        scenario = get_scenario(result.name)
        print(f"\n[{result.name}] {scenario.description}")
        for key, value in result.metrics.items():
            print(f"  {key}: {value:.6f}" if isinstance(value, float)
                  else f"  {key}: {value}")
    if args.output:
        args.output.parent.mkdir(parents=True, exist_ok=True)
        report = {
            "timestamp": time.time(),
            "transport": "starway_amd-native (tcp + xgmi/hipipc)",
            "device": args.device or "cpu",
            "scenarios": [r.to_dict(include_samples=args.store_trace)
                          for r in results],
        }
        args.output.write_text(json.dumps(report, indent=2))
        print(f"\nJSON results written to {args.output}")


def main(argv: Sequence[str] | None = None) -> int:
    args = build_parser().parse_args(argv)
    if args.tls:
        print(f"[bench] --tls {args.tls} noted; transports are selected "
              "natively (tcp/shm_ring/cma/xgmi) — see STARWAY_* env knobs.")
    if args.role == "server":
        asyncio.run(run_server(args))
        return 0
    if args.role == "client":
        results = asyncio.run(run_client(args))
        dump_results(results, args)
        return 0
    if args.role == "loopback":
        results = asyncio.run(run_loopback(args))
        dump_results(results, args)
        return 0
    raise ValueError(f"Unknown role {args.role}")


if __name__ == "__main__":  # pragma: no cover
    sys.exit(main())
