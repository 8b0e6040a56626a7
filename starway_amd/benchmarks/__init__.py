"""Benchmark scenario registry (parity: reference src/starway/benchmarks)."""
from __future__ import annotations

from .scenarios import SCENARIOS, ScenarioDefinition, ScenarioResult


def list_scenarios() -> list[str]:
    return list(SCENARIOS.keys())


def get_scenario(name: str) -> ScenarioDefinition:
    try:
        return SCENARIOS[name]
    except KeyError:
        raise KeyError(
            f"unknown scenario '{name}'; available: {', '.join(SCENARIOS)}"
        ) from None


__all__ = [
    "SCENARIOS",
    "ScenarioDefinition",
    "ScenarioResult",
    "list_scenarios",
    "get_scenario",
]
