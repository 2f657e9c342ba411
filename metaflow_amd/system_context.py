"""Execution-phase tracking for the engine's own telemetry.

Parity target: /root/reference/metaflow/system_context.py (ExecutionPhase
:5, _phase_from_cli_args :53): every process in a run is in exactly one
phase — the SCHEDULER process driving the poll loop, a TASK subprocess
executing one step, or a CLIENT reading results — and system-level
telemetry (event logger / monitor records) tags its records with it so
multi-process traces are attributable without parsing argv.

The phase is derived once from the CLI subcommand (cli.py sets it before
dispatch) and inherited by anything the process imports afterwards;
``MFX_PHASE`` overrides for embedded uses (e.g. the Runner tagging its
driver process).
"""

import os

LAUNCH = "launch"          # CLI parsed, no subcommand dispatched yet
SCHEDULER = "scheduler"    # run/resume: the poll-loop process
TASK = "task"              # step subcommand: one task subprocess
CLIENT = "client"          # client/CLI read paths (dump, logs, card)

_PHASES = (LAUNCH, SCHEDULER, TASK, CLIENT)
_current = os.environ.get("MFX_PHASE", LAUNCH)


def set_phase(phase):
    global _current
    if phase not in _PHASES:
        raise ValueError("unknown phase %r (one of %s)"
                         % (phase, ", ".join(_PHASES)))
    _current = phase


def current_phase():
    return _current


def phase_from_subcommand(subcommand):
    """Map a CLI subcommand to its phase (cli.py calls this at
    dispatch; reference _phase_from_cli_args)."""
    return {
        "run": SCHEDULER,
        "resume": SCHEDULER,
        "spin": SCHEDULER,
        "step": TASK,
    }.get(subcommand, CLIENT)
