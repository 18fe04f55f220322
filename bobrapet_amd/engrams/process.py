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

    def __init__(self, command: _t.List[str], secret_defs=None, secrets=None):
        self.command = [str(c) for c in command]
        # secret artifact delivery (reference: pkg/podspec/secrets.go +
        # SecretDefinition catalog/shared_types.go:296-322 — env/file mounts)
        self.secret_defs = list(secret_defs or [])
        self.secrets = dict(secrets or {})

    def _secret_env(self, tmpdir: _t.Optional[str]) -> _t.Dict[str, str]:
        env: _t.Dict[str, str] = {}
        for sd in self.secret_defs:
            name = getattr(sd, "name", None) or (sd.get("name") if isinstance(sd, dict) else None)
            if not name:
                continue
            mount = getattr(sd, "mount_type", None) or (
                sd.get("mountType") if isinstance(sd, dict) else None
            ) or "env"
            required = getattr(sd, "required", None)
            if required is None and isinstance(sd, dict):
                required = sd.get("required")
            value = self.secrets.get(name)
            if value is None:
                if required:
                    raise EngramFailure(
                        f"missing required secret {name!r}", exit_code=2
                    )
                continue
            key = name.upper().replace("-", "_").replace(".", "_")
            if mount in ("env", "both"):
                env[f"BUBU_SECRET_{key}"] = str(value)
            if mount in ("file", "both") and tmpdir is not None:
                path = os.path.join(tmpdir, key)
                with open(path, "w", encoding="utf-8") as fh:
                    fh.write(str(value))
                os.chmod(path, 0o600)
                env[f"BUBU_SECRET_FILE_{key}"] = path
        return env

    def run(self, ctx: EngramContext) -> EngramResult:
        import tempfile

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
        needs_files = any(
            (getattr(sd, "mount_type", None) or (sd.get("mountType") if isinstance(sd, dict) else None))
            in ("file", "both")
            for sd in self.secret_defs
        )
        tmp_ctx = tempfile.TemporaryDirectory(prefix="bubu-secrets-") if needs_files else None
        try:
            env.update(self._secret_env(tmp_ctx.name if tmp_ctx else None))
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
        finally:
            if tmp_ctx is not None:
                tmp_ctx.cleanup()
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
