"""Benchmark scenario registry (parity surface: reference
src/starway/benchmarks — same names, lookup API, and error shape)."""
from __future__ import annotations

from .scenarios import SCENARIOS, ScenarioDefinition, ScenarioResult


def list_scenarios() -> list[str]:
    return list(SCENARIOS)


def get_scenario(name: str) -> ScenarioDefinition:
    if name not in SCENARIOS:
        known = ", ".join(SCENARIOS)
        raise KeyError(f"unknown scenario '{name}'; available: {known}")
    return SCENARIOS[name]


__all__ = ["SCENARIOS", "ScenarioDefinition", "ScenarioResult",
           "list_scenarios", "get_scenario"]
