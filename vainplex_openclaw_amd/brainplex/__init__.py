"""Brainplex: installer CLI that wires the suite into openclaw.json
(rebuild of reference `packages/brainplex` — SURVEY.md §2.6)."""

from .cli import main, parse_args, plan_installation, run_init
from .configurator import (
    build_trust_defaults,
    compute_trust_score,
    generate_configs,
)
from .scanner import extract_agents, find_config, parse_config, scan
from .writer import update_openclaw_config, write_configs

__all__ = [
    "main",
    "parse_args",
    "plan_installation",
    "run_init",
    "build_trust_defaults",
    "compute_trust_score",
    "generate_configs",
    "extract_agents",
    "find_config",
    "parse_config",
    "scan",
    "update_openclaw_config",
    "write_configs",
]
