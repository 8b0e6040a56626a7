"""Benchmark CLI — ``python -m starway_amd.bench``.

Role/orchestration parity with the reference CLI: roles
``server|client|loopback``, a JSON-over-tagged-messages control protocol
with READY/DONE handshakes, per-scenario overrides, JSON reports.
Differences from the reference: ``--device cpu|cuda`` selects host numpy
buffers vs HIP device tensors (the reference had no GPU path), and
``--tls`` is accepted but inert — the transport is always the native
TCP + shm-ring + CMA + xGMI stack.

Wire protocol (one control channel per connected pair):
  client --CONTROL_TAG--> {"scenario": name, "config": {...}}   per run
  server --READY_TAG----> 1 byte        (scenario server is listening)
  <both run the scenario's data traffic on its own tags>
  server --DONE_TAG-----> 1 byte        (server-side runner finished)
  client --CONTROL_TAG--> {"scenario": "__shutdown__"}          at the end
"""
from __future__ import annotations

import argparse
import asyncio
import gc
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

SHUTDOWN = "__shutdown__"


# ---------------------------------------------------------------------------
# CLI
# ---------------------------------------------------------------------------

def parse_size(value: str) -> int:
    text = value.strip().lower().replace("_", "")
    for suffix, mult in (("kib", 1024), ("kb", 1024), ("ki", 1024),
                         ("k", 1024), ("mib", 1024 ** 2), ("mb", 1024 ** 2),
                         ("mi", 1024 ** 2), ("m", 1024 ** 2),
                         ("gib", 1024 ** 3), ("gb", 1024 ** 3),
                         ("gi", 1024 ** 3), ("g", 1024 ** 3)):
        if text.endswith(suffix):
            return int(float(text[: -len(suffix)]) * mult)
    return int(float(text))


def parse_worker_address(value: str) -> bytes:
    return bytes.fromhex(value.replace(":", "").replace(" ", "").strip())


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
    p.add_argument("--stream-warmup", type=int)
    p.add_argument("--stream-iterations", type=int)
    p.add_argument("--output", type=Path)
    p.add_argument("--store-trace", action="store_true")
    return p


_OVERRIDE_KNOBS = {
    "large-array": {"message_bytes": "large_bytes",
                    "iterations": "large_iterations",
                    "warmup": "large_warmup"},
    "small-messages": {"message_bytes": "small_bytes",
                       "iterations": "small_iterations",
                       "warmup_batches": "small_warmup",
                       "concurrency": "small_concurrency"},
    "pingpong-flag": {"iterations": "flag_iterations",
                      "warmup": "flag_warmup"},
    "streaming-duplex": {"message_bytes": "stream_bytes",
                         "iterations": "stream_iterations",
                         "warmup": "stream_warmup"},
}


def scenario_plan(args: argparse.Namespace) -> list[tuple[str, dict[str, Any]]]:
    wanted: Sequence[str]
    if not args.scenarios or (
        len(args.scenarios) == 1 and args.scenarios[0].lower() == "all"
    ):
        wanted = list_scenarios()
    else:
        wanted = args.scenarios
    plan = []
    for name in wanted:
        if name not in SCENARIOS:
            raise ValueError(
                f"Unknown scenario '{name}'. Available: {', '.join(list_scenarios())}"
            )
        overrides = {
            knob: getattr(args, attr)
            for knob, attr in _OVERRIDE_KNOBS.get(name, {}).items()
            if getattr(args, attr) is not None
        }
        if args.device:
            overrides["device"] = args.device
        plan.append((name, overrides))
    return plan


# ---------------------------------------------------------------------------
# Control plane: JSON frames + READY/DONE flags over reserved tags
# ---------------------------------------------------------------------------

def _to_frame(payload: Mapping[str, Any]) -> np.ndarray:
    blob = json.dumps(payload, separators=(",", ":"), sort_keys=True).encode()
    return np.frombuffer(blob, dtype=np.uint8).copy()


class ClientPeer:
    """Client half of a bench session: drives scenarios and the control
    protocol. Doubles as the scenarios' ClientRuntime (attributes
    ``client`` / ``tag_mask`` + ``flush``)."""

    def __init__(self, client: Client) -> None:
        self.client = client
        self.tag_mask = TAG_MASK
        self._flag = np.zeros(1, dtype=np.uint8)

    async def flush(self) -> None:
        await self.client.aflush()

    async def request(self, payload: Mapping[str, Any]) -> None:
        await self.client.asend(_to_frame(payload), CONTROL_TAG)
        await self.client.aflush()

    async def await_flag(self, tag: int) -> None:
        await self.client.arecv(self._flag, tag, self.tag_mask)

    async def run_plan(self, plan) -> list[ScenarioResult]:
        results = []
        for name, overrides in plan:
            print(f"[client] Starting '{name}' with {overrides or 'defaults'}.")
            await self.request({"scenario": name, "config": overrides})
            await self.await_flag(READY_TAG)
            # Bench hygiene: a gen-2 GC pause (~40 ms from future/closure
            # churn) would dominate a batch sample; collect up front and
            # keep the collector off inside the timed region.
            gc.collect()
            gc.disable()
            try:
                results.append(
                    await get_scenario(name).client_runner(self, overrides))
            finally:
                gc.enable()
            await self.await_flag(DONE_TAG)
            print(f"[client] Completed '{name}'.")
        await self.request({"scenario": SHUTDOWN})
        return results


class ServerPeer:
    """Server half: executes scenario requests until shutdown. Doubles as
    the scenarios' ServerRuntime (``server`` / ``endpoint`` / ``tag_mask``
    + ``signal_ready`` / ``flush_endpoint``)."""

    def __init__(self, server: Server, endpoint) -> None:
        self.server = server
        self.endpoint = endpoint
        self.tag_mask = TAG_MASK
        self._flag = np.ones(1, dtype=np.uint8)

    async def signal_ready(self) -> None:
        await self.server.asend(self.endpoint, self._flag, READY_TAG)

    async def flush_endpoint(self) -> None:
        await self.server.aflush_ep(self.endpoint)

    async def next_request(self, max_bytes: int = 4096) -> Mapping[str, Any]:
        sink = np.empty(max_bytes, dtype=np.uint8)
        _, n = await self.server.arecv(sink, CONTROL_TAG, self.tag_mask)
        return json.loads(sink[:n].tobytes().decode())

    async def serve_until_shutdown(self) -> None:
        while True:
            request = await self.next_request()
            name = request.get("scenario")
            if name == SHUTDOWN:
                print("[server] Shutdown request received.")
                return
            if name not in SCENARIOS:
                raise ValueError(f"Unknown scenario '{name}' from client.")
            overrides = request.get("config", {})
            print(f"[server] Running '{name}' with {overrides or 'defaults'}.")
            await get_scenario(name).server_runner(self, overrides)
            await self.server.asend(self.endpoint, self._flag, DONE_TAG)
            print(f"[server] Scenario '{name}' completed.")


# ---------------------------------------------------------------------------
# Roles
# ---------------------------------------------------------------------------

async def run_client(args: argparse.Namespace) -> list[ScenarioResult]:
    client = Client()
    try:
        if args.connect_mode == "worker":
            if not args.worker_address:
                raise ValueError("--worker-address required for connect-mode=worker")
            blob = parse_worker_address(args.worker_address)
            await client.aconnect_address(blob)
            print(f"[client] Connected via worker address ({len(blob)} bytes).")
        else:
            await client.aconnect(args.server_host, args.port)
            print(f"[client] Connected to {args.server_host}:{args.port}.")
        peer = ClientPeer(client)
        results = await peer.run_plan(scenario_plan(args))
        await peer.flush()
    finally:
        await client.aclose()
    return results


async def run_server(args: argparse.Namespace) -> None:
    server = Server()
    loop = asyncio.get_running_loop()
    first_ep: asyncio.Future = loop.create_future()
    server.set_accept_cb(
        lambda ep: loop.call_soon_threadsafe(
            lambda: first_ep.done() or first_ep.set_result(ep)))

    if args.listen_mode == "worker":
        blob = server.listen_address()
        print(f"[server] Listening via worker address: {blob.hex()}")
    else:
        server.listen(args.addr, args.port)
        print(f"[server] Listening on {args.addr}:{args.port}")

    endpoint = await first_ep
    print("[server] Client accepted.")
    try:
        await ServerPeer(server, endpoint).serve_until_shutdown()
    finally:
        await server.aclose()
        print("[server] Closed.")


async def run_loopback(args: argparse.Namespace) -> list[ScenarioResult]:
    # Both halves share one event loop and talk over localhost.
    server_side = asyncio.create_task(run_server(args))
    try:
        return await run_client(args)
    finally:
        await server_side


# ---------------------------------------------------------------------------
# Reporting
# ---------------------------------------------------------------------------

def dump_results(results: Sequence[ScenarioResult],
                 args: argparse.Namespace) -> None:
    if not results:
        print("No results collected.")
        return
    print("\n=== Benchmark Results ===")
    for result in results:
        print(f"\n[{result.name}] {get_scenario(result.name).description}")
        for key, value in result.metrics.items():
            shown = f"{value:.6f}" if isinstance(value, float) else str(value)
            print(f"  {key}: {shown}")
    if args.output:
        args.output.parent.mkdir(parents=True, exist_ok=True)
        args.output.write_text(json.dumps({
            "timestamp": time.time(),
            "transport": "starway_amd-native (tcp + xgmi/hipipc)",
            "device": args.device or "cpu",
            "scenarios": [r.to_dict(include_samples=args.store_trace)
                          for r in results],
        }, indent=2))
        print(f"\nJSON results written to {args.output}")


def main(argv: Sequence[str] | None = None) -> int:
    args = build_parser().parse_args(argv)
    if args.tls:
        print(f"[bench] --tls {args.tls} noted; transports are selected "
              "natively (tcp/shm_ring/cma/xgmi) — see STARWAY_* env knobs.")
    if args.role == "server":
        asyncio.run(run_server(args))
        return 0
    runner = run_client if args.role == "client" else run_loopback
    results = asyncio.run(runner(args))
    dump_results(results, args)
    return 0


if __name__ == "__main__":  # pragma: no cover
    sys.exit(main())
