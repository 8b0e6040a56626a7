"""Benchmark scenarios — same four workloads and metric definitions as the
reference (benchmark.md / scenarios.py there), extended with an optional
``device`` knob: ``cpu`` uses numpy buffers (reference behavior), ``cuda``
uses torch HIP device tensors moved zero-copy over xGMI.
"""
from __future__ import annotations

import asyncio
import time
from dataclasses import dataclass, field
from typing import Any, Awaitable, Callable, Dict, List, Mapping, Protocol

import numpy as np

TAG_MASK: int = (1 << 64) - 1

CONTROL_TAG = 0x1AA0
READY_TAG = 0x1AA1
DONE_TAG = 0x1AA2

LARGE_DATA_TAG = 0x2B00
SMALL_DATA_TAG = 0x2B10
SMALL_ACK_TAG = 0x2B11
FLAG_PING_TAG = 0x2B20
FLAG_PONG_TAG = 0x2B21
STREAM_UP_TAG = 0x2B30
STREAM_DOWN_TAG = 0x2B31


class ClientRuntime(Protocol):
    client: Any
    tag_mask: int

    async def flush(self) -> None: ...


class ServerRuntime(Protocol):
    server: Any
    endpoint: Any
    tag_mask: int

    async def signal_ready(self) -> None: ...

    async def flush_endpoint(self) -> None: ...


@dataclass
class ScenarioResult:
    name: str
    metrics: Dict[str, float]
    samples: Dict[str, List[float]] = field(default_factory=dict)
    config: Dict[str, Any] = field(default_factory=dict)

    def to_dict(self, include_samples: bool = True) -> Dict[str, Any]:
        payload: Dict[str, Any] = {
            "name": self.name,
            "metrics": self.metrics,
            "config": self.config,
        }
        if include_samples:
            payload["samples"] = self.samples
        return payload


ClientRunner = Callable[[ClientRuntime, Mapping[str, Any]], Awaitable[ScenarioResult]]
ServerRunner = Callable[[ServerRuntime, Mapping[str, Any]], Awaitable[None]]


@dataclass
class ScenarioDefinition:
    name: str
    description: str
    defaults: Dict[str, Any]
    client_runner: ClientRunner
    server_runner: ServerRunner


def _merged(defaults: Mapping[str, Any], overrides: Mapping[str, Any]) -> Dict[str, Any]:
    out = dict(defaults)
    out.update({k: v for k, v in overrides.items() if v is not None})
    return out


def _alloc(nbytes: int, device: str, fill: int | None = None):
    """Message buffer on the requested device ('cpu' -> numpy uint8,
    'cuda' / 'cuda:N' -> torch uint8 HIP tensor)."""
    if device == "cpu":
        arr = np.empty(nbytes, dtype=np.uint8)
        if fill is not None:
            arr.fill(fill)
        return arr
    import torch

    t = torch.empty(nbytes, dtype=torch.uint8, device=device)
    if fill is not None:
        t.fill_(fill)
    torch.cuda.synchronize()
    return t


# ---------------------------------------------------------------------------
# large-array: one-way bandwidth of a single big message
# ---------------------------------------------------------------------------

async def _large_array_client(ctx: ClientRuntime, config: Mapping[str, Any]) -> ScenarioResult:
    cfg = _merged(LARGE_ARRAY.defaults, config)
    message_bytes = int(cfg["message_bytes"])
    warmup = int(cfg["warmup"])
    iterations = int(cfg["iterations"])
    device = str(cfg.get("device", "cpu"))

    payload = _alloc(message_bytes, device, fill=0x5A)
    durations: list[float] = []
    per_iter_gbps: list[float] = []

    for idx in range(warmup + iterations):
        start = time.perf_counter()
        await ctx.client.asend(payload, LARGE_DATA_TAG)
        await ctx.flush()
        elapsed = time.perf_counter() - start
        if idx >= warmup:
            durations.append(elapsed)
            if elapsed > 0:
                per_iter_gbps.append((message_bytes / elapsed) / 1e9)

    total_time = sum(durations)
    metrics = {
        "total_seconds": total_time,
        "avg_seconds_per_iter": total_time / iterations if iterations else 0.0,
        "avg_gbps": (message_bytes * iterations / total_time) / 1e9 if total_time else 0.0,
        "best_gbps": max(per_iter_gbps, default=0.0),
        "worst_gbps": min(per_iter_gbps, default=0.0),
    }
    return ScenarioResult(
        name="large-array",
        metrics=metrics,
        samples={"duration_seconds": durations, "per_iter_gbps": per_iter_gbps},
        config=dict(cfg),
    )


async def _large_array_server(ctx: ServerRuntime, config: Mapping[str, Any]) -> None:
    cfg = _merged(LARGE_ARRAY.defaults, config)
    message_bytes = int(cfg["message_bytes"])
    total = int(cfg["warmup"]) + int(cfg["iterations"])
    device = str(cfg.get("device", "cpu"))
    recv_buffer = _alloc(message_bytes, device)
    await ctx.signal_ready()
    for _ in range(total):
        await ctx.server.arecv(recv_buffer, LARGE_DATA_TAG, ctx.tag_mask)
    await ctx.flush_endpoint()


# ---------------------------------------------------------------------------
# small-messages: many concurrent small sends
# ---------------------------------------------------------------------------

async def _small_messages_client(ctx: ClientRuntime, config: Mapping[str, Any]) -> ScenarioResult:
    cfg = _merged(SMALL_MESSAGES.defaults, config)
    message_bytes = int(cfg["message_bytes"])
    warmup = int(cfg["warmup_batches"])
    iterations = int(cfg["iterations"])
    concurrency = int(cfg["concurrency"])
    device = str(cfg.get("device", "cpu"))

    payloads = [_alloc(message_bytes, device, fill=i % 251) for i in range(concurrency)]
    durations: list[float] = []
    per_message_latency: list[float] = []

    for batch in range(warmup + iterations):
        start = time.perf_counter()
        await asyncio.gather(
            *(ctx.client.asend(buf, SMALL_DATA_TAG) for buf in payloads)
        )
        await ctx.flush()
        elapsed = time.perf_counter() - start
        if batch >= warmup:
            durations.append(elapsed)
            if concurrency:
                per_message_latency.append(elapsed / concurrency)

    total_messages = iterations * concurrency
    total_time = sum(durations)
    lat_us = np.array(per_message_latency) * 1e6 if per_message_latency else np.zeros(1)
    metrics = {
        "total_seconds": total_time,
        "messages_per_second": total_messages / total_time if total_time else 0.0,
        "bandwidth_gbps": (message_bytes * total_messages / total_time) / 1e9
        if total_time
        else 0.0,
        "latency_p50_us": float(np.percentile(lat_us, 50)),
        "latency_p95_us": float(np.percentile(lat_us, 95)),
    }
    return ScenarioResult(
        name="small-messages",
        metrics=metrics,
        samples={
            "batch_duration_seconds": durations,
            "avg_latency_seconds": per_message_latency,
        },
        config=dict(cfg),
    )


async def _small_messages_server(ctx: ServerRuntime, config: Mapping[str, Any]) -> None:
    cfg = _merged(SMALL_MESSAGES.defaults, config)
    message_bytes = int(cfg["message_bytes"])
    total = int(cfg["warmup_batches"]) + int(cfg["iterations"])
    concurrency = int(cfg["concurrency"])
    device = str(cfg.get("device", "cpu"))
    buffers = [_alloc(message_bytes, device) for _ in range(concurrency)]
    await ctx.signal_ready()
    for _ in range(total):
        await asyncio.gather(
            *(ctx.server.arecv(buf, SMALL_DATA_TAG, ctx.tag_mask) for buf in buffers)
        )
    await ctx.flush_endpoint()


# ---------------------------------------------------------------------------
# pingpong-flag: 1-byte round-trip latency
# ---------------------------------------------------------------------------

async def _pingpong_client(ctx: ClientRuntime, config: Mapping[str, Any]) -> ScenarioResult:
    cfg = _merged(PINGPONG_FLAG.defaults, config)
    warmup = int(cfg["warmup"])
    iterations = int(cfg["iterations"])
    device = str(cfg.get("device", "cpu"))
    message_bytes = int(cfg.get("message_bytes", 1))

    send_buf = _alloc(message_bytes, device, fill=1)
    recv_buf = _alloc(message_bytes, device, fill=0)
    durations: list[float] = []

    for _ in range(warmup):
        fut = ctx.client.arecv(recv_buf, FLAG_PONG_TAG, ctx.tag_mask)
        await ctx.client.asend(send_buf, FLAG_PING_TAG)
        await fut

    for _ in range(iterations):
        fut = ctx.client.arecv(recv_buf, FLAG_PONG_TAG, ctx.tag_mask)
        start = time.perf_counter()
        await ctx.client.asend(send_buf, FLAG_PING_TAG)
        await fut
        durations.append(time.perf_counter() - start)

    await ctx.flush()
    lat = np.array(durations) * 1e6 if durations else np.zeros(1)
    metrics = {
        "avg_rtt_us": float(np.mean(lat)),
        "median_rtt_us": float(np.median(lat)),
        "min_rtt_us": float(np.min(lat)),
        "max_rtt_us": float(np.max(lat)),
        "avg_one_way_us": float(np.mean(lat)) / 2.0,
    }
    return ScenarioResult(
        name="pingpong-flag",
        metrics=metrics,
        samples={"rtt_seconds": durations},
        config=dict(cfg),
    )


async def _pingpong_server(ctx: ServerRuntime, config: Mapping[str, Any]) -> None:
    cfg = _merged(PINGPONG_FLAG.defaults, config)
    total = int(cfg["warmup"]) + int(cfg["iterations"])
    device = str(cfg.get("device", "cpu"))
    message_bytes = int(cfg.get("message_bytes", 1))
    recv_buf = _alloc(message_bytes, device, fill=0)
    ack_buf = _alloc(message_bytes, device, fill=1)
    await ctx.signal_ready()
    for _ in range(total):
        await ctx.server.arecv(recv_buf, FLAG_PING_TAG, ctx.tag_mask)
        await ctx.server.asend(ctx.endpoint, ack_buf, FLAG_PONG_TAG)
    await ctx.flush_endpoint()


# ---------------------------------------------------------------------------
# streaming-duplex: concurrent streams both directions
# ---------------------------------------------------------------------------

async def _streaming_duplex_client(ctx: ClientRuntime, config: Mapping[str, Any]) -> ScenarioResult:
    cfg = _merged(STREAMING_DUPLEX.defaults, config)
    message_bytes = int(cfg["message_bytes"])
    warmup = int(cfg["warmup"])
    iterations = int(cfg["iterations"])
    device = str(cfg.get("device", "cpu"))

    send_buf = _alloc(message_bytes, device, fill=0x7B)
    recv_buf = _alloc(message_bytes, device)
    durations: list[float] = []

    for idx in range(warmup + iterations):
        recv_future = ctx.client.arecv(recv_buf, STREAM_DOWN_TAG, ctx.tag_mask)
        start = time.perf_counter()
        send_future = ctx.client.asend(send_buf, STREAM_UP_TAG)
        await asyncio.gather(send_future, recv_future)
        elapsed = time.perf_counter() - start
        if idx >= warmup:
            durations.append(elapsed)

    await ctx.flush()
    total_time = sum(durations)
    per_dir = message_bytes * iterations
    metrics = {
        "total_seconds": total_time,
        "avg_seconds_per_iter": total_time / iterations if iterations else 0.0,
        "client_to_server_gbps": per_dir / total_time / 1e9 if total_time else 0.0,
        "server_to_client_gbps": per_dir / total_time / 1e9 if total_time else 0.0,
        "aggregate_gbps": 2 * per_dir / total_time / 1e9 if total_time else 0.0,
    }
    return ScenarioResult(
        name="streaming-duplex",
        metrics=metrics,
        samples={"iteration_seconds": durations},
        config=dict(cfg),
    )


async def _streaming_duplex_server(ctx: ServerRuntime, config: Mapping[str, Any]) -> None:
    cfg = _merged(STREAMING_DUPLEX.defaults, config)
    message_bytes = int(cfg["message_bytes"])
    total = int(cfg["warmup"]) + int(cfg["iterations"])
    device = str(cfg.get("device", "cpu"))
    send_buf = _alloc(message_bytes, device, fill=0x3C)
    recv_buf = _alloc(message_bytes, device)
    await ctx.signal_ready()
    for _ in range(total):
        recv_future = ctx.server.arecv(recv_buf, STREAM_UP_TAG, ctx.tag_mask)
        send_future = ctx.server.asend(ctx.endpoint, send_buf, STREAM_DOWN_TAG)
        await asyncio.gather(recv_future, send_future)
    await ctx.flush_endpoint()


# ---------------------------------------------------------------------------

LARGE_ARRAY = ScenarioDefinition(
    name="large-array",
    description="Measure one-way bandwidth by transferring a single large buffer.",
    defaults={"message_bytes": 1 << 30, "warmup": 1, "iterations": 3, "device": "cpu"},
    client_runner=_large_array_client,
    server_runner=_large_array_server,
)

SMALL_MESSAGES = ScenarioDefinition(
    name="small-messages",
    description="Stress many small messages with configurable concurrency.",
    defaults={
        "message_bytes": 1024,
        "warmup_batches": 2,
        "iterations": 10,
        "concurrency": 64,
        "device": "cpu",
    },
    client_runner=_small_messages_client,
    server_runner=_small_messages_server,
)

PINGPONG_FLAG = ScenarioDefinition(
    name="pingpong-flag",
    description="Round-trip a single-byte control flag to capture latency.",
    defaults={"warmup": 100, "iterations": 1000, "message_bytes": 1, "device": "cpu"},
    client_runner=_pingpong_client,
    server_runner=_pingpong_server,
)

STREAMING_DUPLEX = ScenarioDefinition(
    name="streaming-duplex",
    description="Bidirectional medium-sized streaming in both directions.",
    defaults={"message_bytes": 4 * 1024 * 1024, "warmup": 8, "iterations": 64, "device": "cpu"},
    client_runner=_streaming_duplex_client,
    server_runner=_streaming_duplex_server,
)

SCENARIOS: Dict[str, ScenarioDefinition] = {
    s.name: s for s in (LARGE_ARRAY, SMALL_MESSAGES, PINGPONG_FLAG, STREAMING_DUPLEX)
}

__all__ = [
    "SCENARIOS",
    "ScenarioDefinition",
    "ScenarioResult",
    "ClientRuntime",
    "ServerRuntime",
    "CONTROL_TAG",
    "READY_TAG",
    "DONE_TAG",
    "TAG_MASK",
    "LARGE_DATA_TAG",
    "SMALL_DATA_TAG",
    "FLAG_PING_TAG",
    "FLAG_PONG_TAG",
    "STREAM_UP_TAG",
    "STREAM_DOWN_TAG",
]
