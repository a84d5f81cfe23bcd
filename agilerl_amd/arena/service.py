"""Arena service: the server side of the RLOps surface, as a FastAPI app.

Reference parity: the reference's Arena is a hosted cloud service the SDK
talks to over HTTPS (``agilerl-arena/agilerl/arena/client.py:128-986``:
auth, environment validation, experiment submission/resume, checkpoint
listing, dataset registry, NDJSON progress streaming).  This module is a
first-party, self-hostable implementation of that API over a server-side
workspace directory: experiments run through :class:`LocalTrainer` on a
background thread, per-experiment metric events append to an NDJSON file
that `/experiments/{id}/stream` serves, checkpoints and datasets live
under the workspace.

Run standalone: ``python -m agilerl_amd.arena.service --workspace w --port 8040``
(uvicorn).  Tests mount the app in-process through httpx's ASGI transport.
"""

from __future__ import annotations

import json
import os
import threading
import time
import uuid
from typing import Any, Dict, Optional

from .stream import StreamEvent

try:  # module-level so string annotations resolve under PEP 563
    from starlette.requests import Request
except ImportError:  # pragma: no cover - starlette ships with fastapi
    Request = Any  # type: ignore[assignment]

__all__ = ["ArenaService", "create_app"]


class ArenaService:
    def __init__(self, workspace: str = ".arena-server"):
        self.workspace = workspace
        os.makedirs(workspace, exist_ok=True)
        self._threads: Dict[str, threading.Thread] = {}
        self._lock = threading.Lock()

    # ------------------------------------------------------------------
    def _exp_dir(self, exp_id: str) -> str:
        return os.path.join(self.workspace, exp_id)

    def _write(self, rel: str, payload: Dict[str, Any]) -> None:
        path = os.path.join(self.workspace, rel)
        os.makedirs(os.path.dirname(path), exist_ok=True)
        with open(path, "w") as f:
            json.dump(payload, f)

    def _emit(self, exp_id: str, kind: str, payload: Dict[str, Any]) -> None:
        event = StreamEvent(kind, exp_id, payload, ts=time.time())
        path = os.path.join(self._exp_dir(exp_id), "events.ndjson")
        with self._lock:
            with open(path, "a") as f:
                f.write(event.to_json() + "\n")

    # ------------------------------------------------------------------
    def device_auth(self) -> Dict[str, Any]:
        token = f"tok-{uuid.uuid4().hex[:16]}"
        self._write("tokens.json", {"token": token, "issued_at": time.time()})
        return {"access_token": token, "token_type": "bearer"}

    def validate_environment(self, env_spec: Dict[str, Any]) -> Dict[str, Any]:
        from ..models.manifest import TrainingManifest

        manifest = TrainingManifest.model_validate({
            "algorithm": {"name": env_spec.get("algorithm", "DQN")},
            "environment": env_spec,
        })
        spec = manifest.env_spec()
        report = {"valid": True, "type": spec.type, "warnings": []}
        if spec.type == "gym":
            from ..envs.registry import ENV_REGISTRY

            if spec.env_id not in ENV_REGISTRY:
                report["warnings"].append(
                    f"env_id '{spec.env_id}' is not registered on the service"
                )
        return report

    # ------------------------------------------------------------------
    def submit_experiment(self, manifest_doc: Dict[str, Any], device: str = "cpu") -> str:
        from ..models.manifest import TrainingManifest

        manifest = TrainingManifest.model_validate(manifest_doc)
        exp_id = f"exp-{uuid.uuid4().hex[:12]}"
        exp_dir = self._exp_dir(exp_id)
        os.makedirs(exp_dir, exist_ok=True)
        manifest.to_yaml(os.path.join(exp_dir, "manifest.yaml"))
        self._write(f"{exp_id}/status.json", {"status": "queued", "ts": time.time()})
        self._emit(exp_id, "status", {"status": "queued"})
        t = threading.Thread(
            target=self._run_experiment, args=(exp_id, manifest, device), daemon=True
        )
        self._threads[exp_id] = t
        t.start()
        return exp_id

    def _run_experiment(self, exp_id: str, manifest, device: str) -> None:
        from ..logger import Logger
        from ..training.trainer import LocalTrainer

        exp_dir = self._exp_dir(exp_id)
        service = self

        class _StreamLogger(Logger):
            def log_report(self, report):  # population report dict
                try:
                    payload = {
                        k: v for k, v in dict(report).items()
                        if isinstance(v, (int, float, str, bool)) or v is None
                    }
                    service._emit(exp_id, "metrics", payload)
                except Exception:
                    pass

            def close(self):
                pass

        self._write(f"{exp_id}/status.json", {"status": "running", "ts": time.time()})
        self._emit(exp_id, "status", {"status": "running"})
        try:
            if manifest.training.checkpoint_path is None:
                manifest.training.checkpoint_path = os.path.join(exp_dir, "ckpt.pt")
                manifest.training.checkpoint = (
                    manifest.training.checkpoint or manifest.training.evo_steps
                )
            trainer = LocalTrainer(manifest, device=device, loggers=[_StreamLogger()])
            results = trainer.train()
            self._write(f"{exp_id}/status.json", {
                "status": "completed", "ts": time.time(),
            })
            self._emit(exp_id, "status", {"status": "completed"})
            self._emit(exp_id, "done", {"results": _jsonable(results)})
        except Exception as e:  # noqa: BLE001 - reported to the client
            self._write(f"{exp_id}/status.json", {"status": "failed", "error": str(e)})
            self._emit(exp_id, "status", {"status": "failed", "error": str(e)})

    # ------------------------------------------------------------------
    def experiment_status(self, exp_id: str) -> Optional[Dict[str, Any]]:
        path = os.path.join(self._exp_dir(exp_id), "status.json")
        if not os.path.exists(path):
            return None
        with open(path) as f:
            return json.load(f)

    def list_experiments(self):
        return sorted(
            d for d in os.listdir(self.workspace)
            if d.startswith("exp-") and os.path.isdir(self._exp_dir(d))
        )

    def list_checkpoints(self, exp_id: str):
        exp_dir = self._exp_dir(exp_id)
        if not os.path.isdir(exp_dir):
            return None
        return sorted(
            f for f in os.listdir(exp_dir) if f.endswith(".pt") or f.startswith("ckpt")
        )

    def checkpoint_path(self, exp_id: str, name: str) -> Optional[str]:
        path = os.path.join(self._exp_dir(exp_id), os.path.basename(name))
        return path if os.path.exists(path) else None

    def wait(self, exp_id: str, timeout: float = 300.0) -> None:
        t = self._threads.get(exp_id)
        if t is not None:
            t.join(timeout)

    # ------------------------------------------------------------------
    def register_dataset(self, name: str, payload: bytes, ext: str) -> str:
        ds_id = f"ds-{name}"
        ds_dir = os.path.join(self.workspace, "datasets")
        os.makedirs(ds_dir, exist_ok=True)
        dest = os.path.join(ds_dir, ds_id + ext)
        with open(dest, "wb") as f:
            f.write(payload)
        self._write(os.path.join("datasets", ds_id + ".json"),
                    {"id": ds_id, "stored": dest, "uploaded_at": time.time()})
        return ds_id

    def list_datasets(self):
        ds_dir = os.path.join(self.workspace, "datasets")
        if not os.path.isdir(ds_dir):
            return []
        return sorted(f[:-5] for f in os.listdir(ds_dir) if f.endswith(".json"))

    def event_lines(self, exp_id: str, follow_seconds: float = 0.0):
        """Yield NDJSON event lines; optionally tail-follow while running."""
        path = os.path.join(self._exp_dir(exp_id), "events.ndjson")
        deadline = time.time() + follow_seconds
        pos = 0
        while True:
            if os.path.exists(path):
                with open(path) as f:
                    f.seek(pos)
                    for line in f:
                        yield line
                    pos = f.tell()
            status = self.experiment_status(exp_id) or {}
            if status.get("status") in ("completed", "failed"):
                # drain whatever arrived after the last read
                if os.path.exists(path):
                    with open(path) as f:
                        f.seek(pos)
                        for line in f:
                            yield line
                return
            if time.time() >= deadline:
                return
            time.sleep(0.05)


def _jsonable(obj):
    try:
        json.dumps(obj)
        return obj
    except (TypeError, ValueError):
        return repr(obj)


def create_app(service: Optional[ArenaService] = None, workspace: str = ".arena-server"):
    """FastAPI app exposing the Arena REST surface."""
    from fastapi import FastAPI, HTTPException
    from fastapi.responses import StreamingResponse

    svc = service or ArenaService(workspace)
    app = FastAPI(title="agilerl-amd arena", version="1.0")
    app.state.service = svc

    @app.post("/auth/device")
    def auth_device():
        return svc.device_auth()

    @app.post("/environments/validate")
    async def validate(request: Request):
        body = await request.json()
        try:
            return svc.validate_environment(body)
        except Exception as e:  # noqa: BLE001
            raise HTTPException(422, str(e))

    @app.post("/experiments")
    async def submit(request: Request):
        body = await request.json()
        try:
            exp_id = svc.submit_experiment(
                body.get("manifest", body), device=body.get("device", "cpu")
            )
        except Exception as e:  # noqa: BLE001
            raise HTTPException(422, str(e))
        return {"experiment_id": exp_id, "status": "queued"}

    @app.get("/experiments")
    def experiments():
        return {"experiments": svc.list_experiments()}

    @app.get("/experiments/{exp_id}")
    def status(exp_id: str):
        st = svc.experiment_status(exp_id)
        if st is None:
            raise HTTPException(404, f"unknown experiment {exp_id}")
        return st

    @app.get("/experiments/{exp_id}/checkpoints")
    def checkpoints(exp_id: str):
        ck = svc.list_checkpoints(exp_id)
        if ck is None:
            raise HTTPException(404, f"unknown experiment {exp_id}")
        return {"checkpoints": ck}

    @app.get("/experiments/{exp_id}/checkpoints/{name}")
    def checkpoint(exp_id: str, name: str):
        from fastapi.responses import FileResponse

        path = svc.checkpoint_path(exp_id, name)
        if path is None:
            raise HTTPException(404, f"no checkpoint {name}")
        return FileResponse(path, media_type="application/octet-stream")

    @app.get("/experiments/{exp_id}/stream")
    def stream(exp_id: str, follow: float = 0.0):
        if svc.experiment_status(exp_id) is None:
            raise HTTPException(404, f"unknown experiment {exp_id}")
        return StreamingResponse(
            svc.event_lines(exp_id, follow_seconds=follow),
            media_type="application/x-ndjson",
        )

    @app.post("/datasets")
    async def upload_dataset(request: Request):
        name = request.query_params.get("name", "dataset")
        ext = request.query_params.get("ext", ".bin")
        payload = await request.body()
        return {"dataset_id": svc.register_dataset(name, payload, ext)}

    @app.get("/datasets")
    def datasets():
        return {"datasets": svc.list_datasets()}

    return app


def main():  # pragma: no cover - manual entry point
    import argparse

    import uvicorn

    p = argparse.ArgumentParser()
    p.add_argument("--workspace", default=".arena-server")
    p.add_argument("--host", default="127.0.0.1")
    p.add_argument("--port", type=int, default=8040)
    args = p.parse_args()
    uvicorn.run(create_app(workspace=args.workspace), host=args.host, port=args.port)


if __name__ == "__main__":  # pragma: no cover
    main()
