"""Thin Python client for the REST control plane (`bobrapet serve`).

The in-process `RunEngine` API is the primary surface; this client covers
the out-of-process case (an operator box running `bobrapet serve
--checkpoint state.json` with remote submitters):

    from bobrapet_amd.client import Client

    c = Client("http://127.0.0.1:8080")
    c.apply(open("story.yaml").read())
    run = c.run_story("default/my-story", {"x": 1})       # waits
    print(run["phase"], run["output"])
    c.approve_gate("default", run["name"], "approval-step")
"""
from __future__ import annotations

import typing as _t


class ClientError(RuntimeError):
    def __init__(self, status: int, body: str):
        super().__init__(f"HTTP {status}: {body[:300]}")
        self.status = status


class Client:
    def __init__(self, base_url: str = "http://127.0.0.1:8080", session=None, timeout: float = 300.0):
        self.base = base_url.rstrip("/")
        self.timeout = timeout
        if session is None:
            import requests

            session = requests.Session()
        self._s = session

    # -- plumbing -------------------------------------------------------

    def _req(self, method: str, path: str, json_body=None, params=None) -> dict:
        kwargs = {"json": json_body, "params": params}
        # test sessions (starlette TestClient) reject per-request timeouts
        if type(self._s).__module__.startswith("requests"):
            kwargs["timeout"] = self.timeout
        r = self._s.request(method, f"{self.base}{path}", **kwargs)
        if r.status_code >= 400:
            raise ClientError(r.status_code, r.text)
        return r.json()

    # -- resources ------------------------------------------------------

    def apply(self, yaml_text: str) -> int:
        """Apply CRD-style YAML documents; returns how many were applied."""
        return self._req("POST", "/resources", {"yaml": yaml_text})["applied"]

    def stories(self) -> _t.List[dict]:
        return self._req("GET", "/stories")["stories"]

    # -- runs -----------------------------------------------------------

    def submit(self, story_key: str, inputs=None, run_name: _t.Optional[str] = None) -> dict:
        ns, _, name = story_key.rpartition("/")
        return self._req(
            "POST",
            f"/stories/{ns or 'default'}/{name}/runs",
            {"inputs": inputs or {}, "runName": run_name},
        )

    def run_story(self, story_key: str, inputs=None, timeout: float = 300.0) -> dict:
        """Submit and wait for the terminal record."""
        ns, _, name = story_key.rpartition("/")
        return self._req(
            "POST",
            f"/stories/{ns or 'default'}/{name}/runs",
            {"inputs": inputs or {}, "wait": True, "timeout": timeout},
        )

    def run(self, ns: str, name: str) -> dict:
        return self._req("GET", f"/runs/{ns}/{name}")

    def runs(self, phase: _t.Optional[str] = None) -> _t.List[dict]:
        return self._req("GET", "/runs", params={"phase": phase} if phase else None)["runs"]

    def cancel(self, ns: str, name: str, graceful: bool = True) -> dict:
        return self._req("POST", f"/runs/{ns}/{name}/cancel", {"graceful": graceful})

    def redrive(self, ns: str, name: str, from_step: _t.Optional[str] = None) -> dict:
        return self._req("POST", f"/runs/{ns}/{name}/redrive", {"fromStep": from_step})

    def approve_gate(self, ns: str, run_name: str, step: str, decided_by: str = "client") -> dict:
        return self._req(
            "POST", f"/runs/{ns}/{run_name}/gates/{step}", {"approve": True, "decidedBy": decided_by}
        )

    def reject_gate(self, ns: str, run_name: str, step: str, decided_by: str = "client") -> dict:
        return self._req(
            "POST", f"/runs/{ns}/{run_name}/gates/{step}", {"approve": False, "decidedBy": decided_by}
        )

    def trace(self, ns: str, name: str) -> dict:
        return self._req("GET", f"/runs/{ns}/{name}/trace")

    def trigger_impulse(self, ns: str, name: str, payload: dict) -> dict:
        return self._req("POST", f"/impulses/{ns}/{name}", payload)

    def metrics_text(self) -> str:
        r = self._s.get(f"{self.base}/metrics")
        if r.status_code >= 400:
            raise ClientError(r.status_code, r.text)
        return r.text
