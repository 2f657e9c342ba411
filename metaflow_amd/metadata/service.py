"""HTTP metadata service + client provider.

Parity target: /root/reference/metaflow/plugins/metadata_providers/
service.py (HTTP provider with version handshake + retries) and the
metadata-service it talks to. Here both halves ship in-repo:

* ``python -m metaflow_amd.metadata.service --root <ds_root>`` runs the
  service (FastAPI/uvicorn) over the SAME JSON layout the local provider
  uses, so local and service runs are interchangeable;
* ``ServiceMetadataProvider`` is the client, selected with
  ``--metadata service`` (+ MFX_SERVICE_URL).
"""

import json
import os
import time
import urllib.error
import urllib.request

from ..exceptions import MetadataException

API_VERSION = 1


class ServiceMetadataProvider(object):
    """HTTP client with retry/backoff; same interface as the local
    provider (metadata/local.py)."""

    TYPE = "service"

    def __init__(self, flow_name, storage=None, url=None, retries=4):
        self.flow_name = flow_name
        self.url = (url or os.environ.get("MFX_SERVICE_URL",
                                          "http://127.0.0.1:8787")
                    ).rstrip("/")
        self._retries = retries
        self._handshake_done = False

    def _request(self, method, path, payload=None):
        body = json.dumps(payload).encode() if payload is not None else None
        last = None
        for attempt in range(self._retries):
            try:
                req = urllib.request.Request(
                    self.url + path, data=body, method=method,
                    headers={"Content-Type": "application/json",
                             "X-MFX-API-Version": str(API_VERSION)})
                with urllib.request.urlopen(req, timeout=10) as resp:
                    data = resp.read()
                    return json.loads(data) if data else None
            except urllib.error.HTTPError as e:
                if e.code == 404:
                    return None
                last = e
            except Exception as e:  # noqa: BLE001
                last = e
            time.sleep(0.2 * (2 ** attempt))
        raise MetadataException("metadata service unreachable at %s: %s"
                                % (self.url, last))

    def _handshake(self):
        if not self._handshake_done:
            info = self._request("GET", "/version")
            if info.get("api_version") != API_VERSION:
                raise MetadataException(
                    "metadata service API version mismatch: %s" % info)
            self._handshake_done = True

    # -- provider interface -------------------------------------------------
    def new_run_id(self, tags=None):
        self._handshake()
        out = self._request("POST", "/flows/%s/runs" % self.flow_name,
                            {"tags": sorted(tags or [])})
        return out["run_id"]

    def register_run(self, run_id, tags=None, origin_run_id=None):
        self._handshake()
        self._request("POST", "/flows/%s/runs" % self.flow_name,
                      {"run_id": str(run_id), "tags": sorted(tags or []),
                       "origin_run_id": origin_run_id})

    def register_run_done(self, run_id, success):
        self._request("PATCH", "/flows/%s/runs/%s" % (self.flow_name,
                                                      run_id),
                      {"status": "successful" if success else "failed"})

    def update_run_info(self, run_id, extra):
        self._request("PATCH", "/flows/%s/runs/%s" % (self.flow_name,
                                                      run_id), extra)

    def register_task(self, run_id, step_name, task_id, attempt=0,
                      metadata=None):
        self._request(
            "POST", "/flows/%s/runs/%s/tasks" % (self.flow_name, run_id),
            {"step_name": step_name, "task_id": str(task_id),
             "attempt": attempt, "metadata": metadata or {}})

    def register_metadata(self, run_id, step_name, task_id, attempt,
                          metadata):
        self.register_task(run_id, step_name, task_id, attempt, metadata)

    def heartbeat(self, run_id):
        self._request("POST", "/flows/%s/runs/%s/heartbeat"
                      % (self.flow_name, run_id), {})

    def list_runs(self):
        return self._request("GET", "/flows/%s/runs" % self.flow_name) or []

    def get_run(self, run_id):
        return self._request("GET", "/flows/%s/runs/%s"
                             % (self.flow_name, run_id))

    def get_task(self, run_id, step_name, task_id):
        return self._request(
            "GET", "/flows/%s/runs/%s/tasks/%s.%s"
            % (self.flow_name, run_id, step_name, task_id))

    def list_tasks(self, run_id, step_name=None):
        out = self._request("GET", "/flows/%s/runs/%s/tasks"
                            % (self.flow_name, run_id)) or []
        if step_name is not None:
            out = [t for t in out if t.get("step_name") == step_name]
        return out

    def add_run_tags(self, run_id, tags):
        self._request("PATCH", "/flows/%s/runs/%s/tags"
                      % (self.flow_name, run_id),
                      {"add": list(tags)})

    def remove_run_tags(self, run_id, tags):
        self._request("PATCH", "/flows/%s/runs/%s/tags"
                      % (self.flow_name, run_id),
                      {"remove": list(tags)})

    def replace_run_tags(self, run_id, removals, additions):
        self._request("PATCH", "/flows/%s/runs/%s/tags"
                      % (self.flow_name, run_id),
                      {"remove": list(removals), "add": list(additions)})


# ---------------------------------------------------------------- service
def build_app(root):
    """FastAPI app exposing the local-JSON metadata layout over HTTP."""
    from fastapi import FastAPI, Request

    from ..datastore.storage import LocalStorage
    from .local import LocalMetadataProvider

    app = FastAPI(title="mfx metadata service")
    storage = LocalStorage(root)

    def provider(flow):
        return LocalMetadataProvider(flow, storage)

    @app.get("/version")
    def version():
        return {"api_version": API_VERSION, "service": "mfx-metadata"}

    @app.post("/flows/{flow}/runs")
    async def create_run(flow: str, request: Request):
        body = await request.json()
        p = provider(flow)
        run_id = body.get("run_id")
        if run_id:
            p.register_run(run_id, body.get("tags"),
                           body.get("origin_run_id"))
        else:
            run_id = p.new_run_id(body.get("tags"))
        return {"run_id": run_id}

    @app.get("/flows/{flow}/runs")
    def list_runs(flow: str):
        return provider(flow).list_runs()

    @app.get("/flows/{flow}/runs/{run_id}")
    def get_run(flow: str, run_id: str):
        from fastapi.responses import JSONResponse

        info = provider(flow).get_run(run_id)
        if info is None:
            return JSONResponse({"error": "not found"}, status_code=404)
        return info

    @app.patch("/flows/{flow}/runs/{run_id}")
    async def patch_run(flow: str, run_id: str, request: Request):
        body = await request.json()
        p = provider(flow)
        status = body.pop("status", None)
        if status is not None:
            p.register_run_done(run_id, status == "successful")
        if body:
            p.update_run_info(run_id, body)
        return {"ok": True}

    @app.post("/flows/{flow}/runs/{run_id}/heartbeat")
    def heartbeat(flow: str, run_id: str):
        provider(flow).heartbeat(run_id)
        return {"ok": True}

    @app.post("/flows/{flow}/runs/{run_id}/tasks")
    async def register_task(flow: str, run_id: str, request: Request):
        body = await request.json()
        provider(flow).register_task(
            run_id, body["step_name"], body["task_id"],
            body.get("attempt", 0), body.get("metadata"))
        return {"ok": True}

    @app.get("/flows/{flow}/runs/{run_id}/tasks")
    def list_tasks(flow: str, run_id: str):
        return provider(flow).list_tasks(run_id)

    @app.get("/flows/{flow}/runs/{run_id}/tasks/{task_ref}")
    def get_task(flow: str, run_id: str, task_ref: str):
        from fastapi.responses import JSONResponse

        step_name, _, task_id = task_ref.partition(".")
        info = provider(flow).get_task(run_id, step_name, task_id)
        if info is None:
            return JSONResponse({"error": "not found"}, status_code=404)
        return info

    @app.patch("/flows/{flow}/runs/{run_id}/tags")
    async def patch_tags(flow: str, run_id: str, request: Request):
        body = await request.json()
        p = provider(flow)
        if body.get("add") and body.get("remove"):
            p.replace_run_tags(run_id, body["remove"], body["add"])
        elif body.get("add"):
            p.add_run_tags(run_id, body["add"])
        elif body.get("remove"):
            p.remove_run_tags(run_id, body["remove"])
        return {"ok": True}

    return app


def main():
    import argparse

    import uvicorn

    ap = argparse.ArgumentParser()
    ap.add_argument("--root", required=True)
    ap.add_argument("--host", default="127.0.0.1")
    ap.add_argument("--port", type=int, default=8787)
    args = ap.parse_args()
    uvicorn.run(build_app(args.root), host=args.host, port=args.port,
                log_level="warning")


if __name__ == "__main__":
    main()
