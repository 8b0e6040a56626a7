"""Benchmark scenarios for the MI355X-native transport.

The four workloads, their default knobs, and their metric names are the
measurement contract shared with the reference (its benchmark.md defines
them); everything below — the class-based scenario objects, the sampling
clock, the buffer factory — is this repo's own implementation, extended
with a ``device`` knob: ``cpu`` runs on numpy buffers, ``cuda``/``cuda:N``
on torch HIP tensors moved zero-copy over xGMI.
"""
from __future__ import annotations

import asyncio
import time
from dataclasses import dataclass, field
from typing import Any, Awaitable, Callable, Dict, List, Mapping, Protocol

import numpy as np

TAG_MASK: int = (1 << 64) - 1

# Control-plane tags (bench.py session protocol).
CONTROL_TAG = 0x1AA0
READY_TAG = 0x1AA1
DONE_TAG = 0x1AA2

# Data-plane tags, one block per scenario.
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


# ---------------------------------------------------------------------------
# shared machinery
# ---------------------------------------------------------------------------

def buffer_on(device: str, nbytes: int, fill: int | None = None):
    """Allocate a message buffer: numpy uint8 on 'cpu', torch uint8 HIP
    tensor on 'cuda'/'cuda:N' (synchronized so timing never includes the
    fill kernel)."""
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


class SampleClock:
    """Wall-clock sampler: lap() times one unit of work, keeping samples
    only past the warmup count; derived stats come out of laps()."""

    def __init__(self, warmup: int) -> None:
        self._warmup = warmup
        self._seen = 0
        self._t0 = 0.0
        self.samples: List[float] = []

    def __enter__(self) -> "SampleClock":
        self._t0 = time.perf_counter()
        return self

    def __exit__(self, *exc) -> None:
        dt = time.perf_counter() - self._t0
        self._seen += 1
        if self._seen > self._warmup:
            self.samples.append(dt)

    @property
    def total(self) -> float:
        return sum(self.samples)

    def us(self) -> np.ndarray:
        return (np.asarray(self.samples) if self.samples else np.zeros(1)) * 1e6


def _rate_gbps(nbytes: float, seconds: float) -> float:
    return (nbytes / seconds) / 1e9 if seconds > 0 else 0.0


class Scenario:
    """One benchmark workload: subclasses set the class attributes and
    implement drive() (client side, returns the result) and serve()
    (server side). resolve() folds CLI overrides onto the defaults."""

    name: str = ""
    describe: str = ""
    knobs: Dict[str, Any] = {}

    @classmethod
    def resolve(cls, overrides: Mapping[str, Any]) -> Dict[str, Any]:
        cfg = dict(cls.knobs)
        for key, val in overrides.items():
            if val is not None:
                cfg[key] = val
        return cfg

    @classmethod
    def repetitions(cls, cfg: Mapping[str, Any]) -> tuple[int, int]:
        warm = int(cfg.get("warmup", cfg.get("warmup_batches", 0)))
        return warm, int(cfg["iterations"])

    async def drive(self, ctx: ClientRuntime, cfg: Mapping[str, Any]) -> ScenarioResult:
        raise NotImplementedError

    async def serve(self, ctx: ServerRuntime, cfg: Mapping[str, Any]) -> None:
        raise NotImplementedError


# ---------------------------------------------------------------------------
# large-array: one-way bandwidth of a single big message (flush-inclusive)
# ---------------------------------------------------------------------------

class LargeArrayScenario(Scenario):
    name = "large-array"
    describe = "Measure one-way bandwidth by transferring a single large buffer."
    knobs = {"message_bytes": 1 << 30, "warmup": 1, "iterations": 3,
             "device": "cpu"}

    async def drive(self, ctx, overrides):
        cfg = self.resolve(overrides)
        nbytes = int(cfg["message_bytes"])
        warm, iters = self.repetitions(cfg)
        payload = buffer_on(str(cfg["device"]), nbytes, fill=0x5A)

        clock = SampleClock(warm)
        for _ in range(warm + iters):
            with clock:
                await ctx.client.asend(payload, LARGE_DATA_TAG)
                await ctx.flush()

        rates = [_rate_gbps(nbytes, dt) for dt in clock.samples]
        return ScenarioResult(
            name=self.name,
            metrics={
                "total_seconds": clock.total,
                "avg_seconds_per_iter": clock.total / iters if iters else 0.0,
                "avg_gbps": _rate_gbps(nbytes * iters, clock.total),
                "best_gbps": max(rates, default=0.0),
                "worst_gbps": min(rates, default=0.0),
            },
            samples={"duration_seconds": clock.samples,
                     "per_iter_gbps": rates},
            config=cfg,
        )

    async def serve(self, ctx, overrides):
        cfg = self.resolve(overrides)
        warm, iters = self.repetitions(cfg)
        sink = buffer_on(str(cfg["device"]), int(cfg["message_bytes"]))
        await ctx.signal_ready()
        for _ in range(warm + iters):
            await ctx.server.arecv(sink, LARGE_DATA_TAG, ctx.tag_mask)
        await ctx.flush_endpoint()


# ---------------------------------------------------------------------------
# small-messages: batches of concurrent small sends, rate + latency
# ---------------------------------------------------------------------------

class SmallMessagesScenario(Scenario):
    name = "small-messages"
    describe = "Stress many small messages with configurable concurrency."
    knobs = {"message_bytes": 1024, "warmup_batches": 2, "iterations": 10,
             "concurrency": 64, "device": "cpu"}

    async def drive(self, ctx, overrides):
        cfg = self.resolve(overrides)
        nbytes = int(cfg["message_bytes"])
        fanout = int(cfg["concurrency"])
        warm, iters = self.repetitions(cfg)
        wave = [buffer_on(str(cfg["device"]), nbytes, fill=i % 251)
                for i in range(fanout)]

        clock = SampleClock(warm)
        for _ in range(warm + iters):
            with clock:
                await asyncio.gather(
                    *(ctx.client.asend(buf, SMALL_DATA_TAG) for buf in wave))
                await ctx.flush()

        # Batch-time / concurrency approximation of per-message latency
        # (the reference's small-messages metric definition).
        per_msg = [dt / fanout for dt in clock.samples] if fanout else []
        lat_us = (np.asarray(per_msg) if per_msg else np.zeros(1)) * 1e6
        sent = iters * fanout
        return ScenarioResult(
            name=self.name,
            metrics={
                "total_seconds": clock.total,
                "messages_per_second": sent / clock.total if clock.total else 0.0,
                "bandwidth_gbps": _rate_gbps(nbytes * sent, clock.total),
                "latency_p50_us": float(np.percentile(lat_us, 50)),
                "latency_p95_us": float(np.percentile(lat_us, 95)),
            },
            samples={"batch_duration_seconds": clock.samples,
                     "avg_latency_seconds": per_msg},
            config=cfg,
        )

    async def serve(self, ctx, overrides):
        cfg = self.resolve(overrides)
        fanout = int(cfg["concurrency"])
        warm, iters = self.repetitions(cfg)
        sinks = [buffer_on(str(cfg["device"]), int(cfg["message_bytes"]))
                 for _ in range(fanout)]
        await ctx.signal_ready()
        for _ in range(warm + iters):
            await asyncio.gather(
                *(ctx.server.arecv(b, SMALL_DATA_TAG, ctx.tag_mask)
                  for b in sinks))
        await ctx.flush_endpoint()


# ---------------------------------------------------------------------------
# pingpong-flag: tiny-message round-trip latency (pre-posted recv)
# ---------------------------------------------------------------------------

class PingpongFlagScenario(Scenario):
    name = "pingpong-flag"
    describe = "Round-trip a single-byte control flag to capture latency."
    knobs = {"warmup": 100, "iterations": 1000, "message_bytes": 1,
             "device": "cpu"}

    async def drive(self, ctx, overrides):
        cfg = self.resolve(overrides)
        nbytes = int(cfg["message_bytes"])
        warm, iters = self.repetitions(cfg)
        ping = buffer_on(str(cfg["device"]), nbytes, fill=1)
        pong = buffer_on(str(cfg["device"]), nbytes, fill=0)

        clock = SampleClock(warm)
        for _ in range(warm + iters):
            # Post the reply recv before sending so the pong never lands
            # unexpected — this is the latency-path contract.
            reply = ctx.client.arecv(pong, FLAG_PONG_TAG, ctx.tag_mask)
            with clock:
                await ctx.client.asend(ping, FLAG_PING_TAG)
                await reply
        await ctx.flush()

        rtt = clock.us()
        return ScenarioResult(
            name=self.name,
            metrics={
                "avg_rtt_us": float(rtt.mean()),
                "median_rtt_us": float(np.median(rtt)),
                "min_rtt_us": float(rtt.min()),
                "max_rtt_us": float(rtt.max()),
                "avg_one_way_us": float(rtt.mean()) / 2.0,
            },
            samples={"rtt_seconds": clock.samples},
            config=cfg,
        )

    async def serve(self, ctx, overrides):
        cfg = self.resolve(overrides)
        nbytes = int(cfg["message_bytes"])
        warm, iters = self.repetitions(cfg)
        flag = buffer_on(str(cfg["device"]), nbytes, fill=0)
        ack = buffer_on(str(cfg["device"]), nbytes, fill=1)
        await ctx.signal_ready()
        for _ in range(warm + iters):
            await ctx.server.arecv(flag, FLAG_PING_TAG, ctx.tag_mask)
            await ctx.server.asend(ctx.endpoint, ack, FLAG_PONG_TAG)
        await ctx.flush_endpoint()


# ---------------------------------------------------------------------------
# streaming-duplex: both directions stream concurrently
# ---------------------------------------------------------------------------

class StreamingDuplexScenario(Scenario):
    name = "streaming-duplex"
    describe = "Bidirectional medium-sized streaming in both directions."
    knobs = {"message_bytes": 4 * 1024 * 1024, "warmup": 8, "iterations": 64,
             "device": "cpu"}

    async def drive(self, ctx, overrides):
        cfg = self.resolve(overrides)
        nbytes = int(cfg["message_bytes"])
        warm, iters = self.repetitions(cfg)
        up = buffer_on(str(cfg["device"]), nbytes, fill=0x7B)
        down = buffer_on(str(cfg["device"]), nbytes)

        clock = SampleClock(warm)
        for _ in range(warm + iters):
            inbound = ctx.client.arecv(down, STREAM_DOWN_TAG, ctx.tag_mask)
            with clock:
                await asyncio.gather(
                    ctx.client.asend(up, STREAM_UP_TAG), inbound)
        await ctx.flush()

        one_way = nbytes * iters
        return ScenarioResult(
            name=self.name,
            metrics={
                "total_seconds": clock.total,
                "avg_seconds_per_iter": clock.total / iters if iters else 0.0,
                "client_to_server_gbps": _rate_gbps(one_way, clock.total),
                "server_to_client_gbps": _rate_gbps(one_way, clock.total),
                "aggregate_gbps": _rate_gbps(2 * one_way, clock.total),
            },
            samples={"iteration_seconds": clock.samples},
            config=cfg,
        )

    async def serve(self, ctx, overrides):
        cfg = self.resolve(overrides)
        nbytes = int(cfg["message_bytes"])
        warm, iters = self.repetitions(cfg)
        down = buffer_on(str(cfg["device"]), nbytes, fill=0x3C)
        up = buffer_on(str(cfg["device"]), nbytes)
        await ctx.signal_ready()
        for _ in range(warm + iters):
            await asyncio.gather(
                ctx.server.arecv(up, STREAM_UP_TAG, ctx.tag_mask),
                ctx.server.asend(ctx.endpoint, down, STREAM_DOWN_TAG))
        await ctx.flush_endpoint()


# ---------------------------------------------------------------------------
# registry
# ---------------------------------------------------------------------------

def _definition(cls: type[Scenario]) -> ScenarioDefinition:
    inst = cls()
    return ScenarioDefinition(
        name=cls.name,
        description=cls.describe,
        defaults=dict(cls.knobs),
        client_runner=inst.drive,
        server_runner=inst.serve,
    )


SCENARIOS: Dict[str, ScenarioDefinition] = {
    cls.name: _definition(cls)
    for cls in (LargeArrayScenario, SmallMessagesScenario,
                PingpongFlagScenario, StreamingDuplexScenario)
}

__all__ = [
    "SCENARIOS",
    "Scenario",
    "ScenarioDefinition",
    "ScenarioResult",
    "ClientRuntime",
    "ServerRuntime",
    "SampleClock",
    "buffer_on",
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
