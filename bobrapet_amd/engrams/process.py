"""External-process engrams: the bubu-sdk contract as REAL env vars.

In the reference an engram is a container whose entrypoint reads the
`BUBU_*` environment (SDK env contract, SURVEY.md §2.6 /
steprun_controller.go:1692-1732) and reports status by exit code + output.
Here an EngramTemplate with `command: [argv...]` runs that contract as a
local subprocess: same env surface, same exit-code classes (0 success,
1 retry, 2 terminal, 3 rateLimited, other unknown), output = the last
stdout line parsed as JSON, earlier stdout lines become step logs.
"""
from __future__ import annotations

import json
import os
import subprocess
import typing as _t

from .base import Engram, EngramContext, EngramFailure, EngramResult


def _jsonable(value):
    """Best-effort JSON for env payloads (tensors → shape markers)."""
    try:
        return json.dumps(value)
    except TypeError:
        return json.dumps(value, default=lambda o: f"<{type(o).__name__}>")


class ProcessEngram(Engram):
    name = "process"
    wants_gpu = False

    def __init__(self, command: _t.List[str]):
        self.command = [str(c) for c in command]

    def run(self, ctx: EngramContext) -> EngramResult:
        env = dict(os.environ)
        env.update(
            {
                "BUBU_STORY_NAME": ctx.story_name or "",
                "BUBU_STORYRUN_ID": ctx.story_run or "",
                "BUBU_STEP_NAME": ctx.step_name or "",
                "BUBU_STEPRUN_NAME": ctx.step_run or "",
                "BUBU_STEPRUN_NAMESPACE": ctx.namespace or "default",
                "BUBU_TRIGGER_DATA": _jsonable(ctx.input),
                "BUBU_STEP_CONFIG": _jsonable(ctx.config),
                "BUBU_EXECUTION_MODE": str(ctx.execution_mode or "job"),
                "BUBU_MAX_INLINE_SIZE": str(ctx.max_inline_size or 0),
                "BUBU_MAX_RECURSION_DEPTH": str(ctx.max_recursion_depth or 0),
                "BUBU_DEBUG": "0",
            }
        )
        if ctx.timeout_seconds:
            env["BUBU_STEP_TIMEOUT"] = str(ctx.timeout_seconds)
        if ctx.device is not None:
            env["BUBU_DEVICE"] = str(ctx.device)
            env["HIP_VISIBLE_DEVICES"] = str(ctx.device)
        try:
            proc = subprocess.run(
                self.command,
                env=env,
                capture_output=True,
                text=True,
                timeout=ctx.timeout_seconds or 600,
            )
        except subprocess.TimeoutExpired:
            raise EngramFailure(f"process {self.command[0]} timed out", exit_code=1)
        except OSError as exc:  # missing binary etc. — terminal
            raise EngramFailure(f"process spawn failed: {exc}", exit_code=2)
        lines = [ln for ln in proc.stdout.splitlines() if ln.strip()]
        for ln in lines[:-1]:
            ctx.log(ln)
        if proc.returncode != 0:
            tail = (proc.stderr or proc.stdout or "").strip()[-500:]
            raise EngramFailure(
                f"process exited {proc.returncode}: {tail}", exit_code=proc.returncode
            )
        output: _t.Any = {}
        if lines:
            try:
                output = json.loads(lines[-1])
            except ValueError:
                ctx.log(lines[-1])
                output = {"stdout": lines[-1]}
        return EngramResult(output=output)
