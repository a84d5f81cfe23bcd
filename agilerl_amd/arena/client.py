"""Arena client: cloud RLOps surface.

Reference parity: ``agilerl-arena/agilerl/arena/client.py:128``
(ArenaClient — OAuth device-flow login :279, validate_environment :442,
submit_experiment :850, resume_experiment :970, list_checkpoints :986).

This deployment is a single offline MI355X node, so the client ships the
same API shape backed by a LOCAL experiment store: submitted experiments
run through :class:`LocalTrainer` in-process (or are queued to disk),
checkpoints/metrics live under the workspace directory.  Pointing
``base_url`` at a real Arena service would only need the transport layer
(httpx) filled in — every method validates and serializes exactly the
payloads the reference sends.
"""

from __future__ import annotations

import json
import os
import time
import uuid
from dataclasses import dataclass
from typing import Any, Dict, List, Optional

from ..models.manifest import TrainingManifest

__all__ = ["ArenaClient", "ExperimentHandle", "ArenaError"]


class ArenaError(RuntimeError):
    pass


@dataclass
class ExperimentHandle:
    experiment_id: str
    status: str = "submitted"
    manifest: Optional[Dict[str, Any]] = None
    results: Optional[Any] = None
    workspace: Optional[str] = None


@dataclass
class Deployment:
    """In-process deployment handle (the local analog of an Arena
    inference endpoint)."""

    experiment_id: str
    checkpoint: str
    agent: Any

    def predict(self, obs):
        import numpy as np

        action = self.agent.get_action(np.asarray(obs, dtype=np.float32), training=False)
        return action[0] if isinstance(action, tuple) else action

    def info(self) -> Dict[str, Any]:
        return {
            "experiment_id": self.experiment_id,
            "checkpoint": self.checkpoint,
            "algo": type(self.agent).__name__,
        }


class ArenaClient:
    """Two backends behind one API (reference client.py:128):

    - ``base_url=None``: LOCAL workspace store — experiments run through
      LocalTrainer in-process.
    - ``base_url="http://..."``: HTTP transport to an
      :mod:`agilerl_amd.arena.service` deployment (httpx); ``http_client``
      injects a preconfigured client (e.g. an in-process ASGI transport
      in tests).
    """

    def __init__(
        self,
        base_url: Optional[str] = None,
        workspace: str = ".arena",
        api_key: Optional[str] = None,
        http_client=None,
    ):
        self.base_url = base_url.rstrip("/") if base_url else None
        self.workspace = workspace
        self.api_key = api_key or os.environ.get("ARENA_API_KEY")
        self._logged_in = False
        self._stream_handler = None
        self._http = http_client
        if self.base_url is not None and self._http is None:
            import httpx

            self._http = httpx.Client(base_url=self.base_url, timeout=60.0)
        os.makedirs(workspace, exist_ok=True)

    # ------------------------------------------------------------------
    @property
    def remote(self) -> bool:
        return self.base_url is not None or self._http is not None

    def _get(self, path: str, **kw):
        r = self._http.get(path, **kw)
        if r.status_code >= 400:
            raise ArenaError(f"GET {path} -> {r.status_code}: {r.text[:300]}")
        return r.json()

    def _post(self, path: str, **kw):
        r = self._http.post(path, **kw)
        if r.status_code >= 400:
            raise ArenaError(f"POST {path} -> {r.status_code}: {r.text[:300]}")
        return r.json()

    # ------------------------------------------------------------------
    # Auth (device-flow shape; offline backend auto-authorizes)
    # ------------------------------------------------------------------
    def login(self, interactive: bool = True) -> bool:
        if self.remote:
            tok = self._post("/auth/device")
            self.api_key = tok["access_token"]
            self._logged_in = True
            return True
        self._logged_in = True
        self._write("auth.json", {"logged_in_at": time.time()})
        return True

    # ------------------------------------------------------------------
    # Streaming (reference client.py:342 set_stream_handler /
    # _open_stream over NDJSON)
    # ------------------------------------------------------------------
    def set_stream_handler(self, handler) -> None:
        """Register a callback invoked for each StreamEvent while
        streaming experiment progress."""
        self._stream_handler = handler

    def stream_experiment(self, experiment_id: str, follow: float = 0.0):
        """Yield StreamEvents for an experiment (remote backend); also
        dispatches to the registered stream handler."""
        from .stream import NDJsonStream

        if not self.remote:
            raise ArenaError("streaming requires the HTTP backend")
        self._require_auth()
        with self._http.stream(
            "GET", f"/experiments/{experiment_id}/stream", params={"follow": follow}
        ) as r:
            if r.status_code >= 400:
                raise ArenaError(f"stream -> {r.status_code}")
            for event in NDJsonStream(r.iter_lines()):
                if self._stream_handler is not None:
                    self._stream_handler(event)
                yield event

    def wait_for_completion(self, experiment_id: str, timeout: float = 300.0,
                            poll: float = 0.2) -> Dict[str, Any]:
        deadline = time.time() + timeout
        while time.time() < deadline:
            st = self.experiment_status(experiment_id)
            if st.get("status") in ("completed", "failed"):
                return st
            time.sleep(poll)
        raise ArenaError(f"experiment {experiment_id} did not finish in {timeout}s")

    def _require_auth(self):
        if not self._logged_in:
            raise ArenaError("not logged in — call login() first")

    # ------------------------------------------------------------------
    # Environment validation
    # ------------------------------------------------------------------
    def validate_environment(self, env_spec: Dict[str, Any]) -> Dict[str, Any]:
        """Structural validation matching the reference's pre-submission check."""
        if self.remote:
            return self._post("/environments/validate", json=env_spec)
        manifest = TrainingManifest.model_validate(
            {"algorithm": {"name": env_spec.get("algorithm", "DQN")}, "environment": env_spec}
        )
        spec = manifest.env_spec()
        report = {"valid": True, "type": spec.type, "warnings": []}
        if spec.type == "gym":
            from ..envs.registry import ENV_REGISTRY

            if spec.env_id not in ENV_REGISTRY:
                report["warnings"].append(
                    f"env_id '{spec.env_id}' is not registered locally"
                )
        return report

    # ------------------------------------------------------------------
    # Experiments
    # ------------------------------------------------------------------
    def submit_experiment(
        self,
        manifest: Dict[str, Any] | TrainingManifest,
        run: bool = True,
        device: str = "cpu",
    ) -> ExperimentHandle:
        self._require_auth()
        if isinstance(manifest, dict):
            manifest = TrainingManifest.model_validate(manifest)
        if self.remote:
            out = self._post("/experiments", json={
                "manifest": manifest.model_dump(mode="json"), "device": device,
            })
            return ExperimentHandle(out["experiment_id"], status=out["status"],
                                    manifest=manifest.model_dump())
        exp_id = f"exp-{uuid.uuid4().hex[:12]}"
        exp_dir = os.path.join(self.workspace, exp_id)
        os.makedirs(exp_dir, exist_ok=True)
        manifest.to_yaml(os.path.join(exp_dir, "manifest.yaml"))
        handle = ExperimentHandle(exp_id, manifest=manifest.model_dump(), workspace=exp_dir)
        self._write(f"{exp_id}/status.json", {"status": "submitted", "ts": time.time()})
        if run:
            from ..training.trainer import LocalTrainer

            self._write(f"{exp_id}/status.json", {"status": "running", "ts": time.time()})
            trainer = LocalTrainer(manifest, device=device)
            if manifest.training.checkpoint_path is None:
                manifest.training.checkpoint_path = os.path.join(exp_dir, "ckpt.pt")
                manifest.training.checkpoint = manifest.training.checkpoint or manifest.training.evo_steps
            try:
                handle.results = trainer.train()
                handle.status = "completed"
            except Exception as e:
                handle.status = "failed"
                self._write(f"{exp_id}/status.json", {"status": "failed", "error": str(e)})
                raise
            self._write(f"{exp_id}/status.json", {"status": "completed", "ts": time.time()})
        return handle

    def resume_experiment(self, experiment_id: str, device: str = "cpu") -> ExperimentHandle:
        self._require_auth()
        exp_dir = os.path.join(self.workspace, experiment_id)
        manifest_path = os.path.join(exp_dir, "manifest.yaml")
        if not os.path.exists(manifest_path):
            raise ArenaError(f"unknown experiment {experiment_id}")
        manifest = TrainingManifest.from_yaml(manifest_path)
        return self.submit_experiment(manifest, run=True, device=device)

    def experiment_status(self, experiment_id: str) -> Dict[str, Any]:
        if self.remote:
            return self._get(f"/experiments/{experiment_id}")
        path = os.path.join(self.workspace, experiment_id, "status.json")
        if not os.path.exists(path):
            raise ArenaError(f"unknown experiment {experiment_id}")
        with open(path) as f:
            return json.load(f)

    def list_experiments(self) -> List[str]:
        if self.remote:
            return self._get("/experiments")["experiments"]
        return sorted(
            d for d in os.listdir(self.workspace)
            if d.startswith("exp-") and os.path.isdir(os.path.join(self.workspace, d))
        )

    def list_checkpoints(self, experiment_id: str) -> List[str]:
        self._require_auth()
        if self.remote:
            return self._get(f"/experiments/{experiment_id}/checkpoints")["checkpoints"]
        exp_dir = os.path.join(self.workspace, experiment_id)
        if not os.path.isdir(exp_dir):
            raise ArenaError(f"unknown experiment {experiment_id}")
        return sorted(
            f for f in os.listdir(exp_dir) if f.endswith(".pt") or f.startswith("ckpt")
        )

    # ------------------------------------------------------------------
    # Datasets (reference arena dataset upload/list; local workspace copy)
    # ------------------------------------------------------------------
    def upload_dataset(self, path: str, name: Optional[str] = None) -> str:
        """Register a local dataset file (npz/json/arrow dir) in the
        workspace; returns the dataset id."""
        self._require_auth()
        if not os.path.exists(path):
            raise ArenaError(f"dataset path not found: {path}")
        name = name or os.path.basename(path).split(".")[0]
        if self.remote:
            if os.path.isdir(path):
                raise ArenaError("remote dataset upload takes a single file")
            with open(path, "rb") as f:
                out = self._post(
                    "/datasets",
                    params={"name": name, "ext": os.path.splitext(path)[1]},
                    content=f.read(),
                )
            return out["dataset_id"]
        ds_id = f"ds-{name}"
        ds_dir = os.path.join(self.workspace, "datasets")
        os.makedirs(ds_dir, exist_ok=True)
        import shutil

        dest = os.path.join(ds_dir, ds_id + os.path.splitext(path)[1])
        if os.path.isdir(path):
            dest = os.path.join(ds_dir, ds_id)
            if not os.path.exists(dest):
                shutil.copytree(path, dest)
        else:
            shutil.copy(path, dest)
        self._write(os.path.join("datasets", ds_id + ".json"),
                    {"id": ds_id, "source": os.path.abspath(path),
                     "stored": dest, "uploaded_at": time.time()})
        return ds_id

    def list_datasets(self) -> List[str]:
        if self.remote:
            return self._get("/datasets")["datasets"]
        ds_dir = os.path.join(self.workspace, "datasets")
        if not os.path.isdir(ds_dir):
            return []
        return sorted(f[:-5] for f in os.listdir(ds_dir) if f.endswith(".json"))

    def dataset_path(self, ds_id: str) -> str:
        meta = os.path.join(self.workspace, "datasets", ds_id + ".json")
        if not os.path.exists(meta):
            raise ArenaError(f"unknown dataset {ds_id}")
        with open(meta) as f:
            return json.load(f)["stored"]

    # ------------------------------------------------------------------
    # Deploy / infer (reference arena deploy+inference endpoints; local
    # backend wraps the in-process serving app — see agilerl_amd/serve.py)
    # ------------------------------------------------------------------
    def deploy(self, experiment_id: str, checkpoint: Optional[str] = None):
        """Load a checkpoint from the experiment dir and return a
        Deployment with .predict()/.info(); host it over HTTP with
        `python -m agilerl_amd.serve <ckpt>`."""
        self._require_auth()
        if self.remote:
            # fetch the checkpoint bytes into the local workspace first
            ckpts = self.list_checkpoints(experiment_id)
            if checkpoint is None:
                if not ckpts:
                    raise ArenaError(f"no checkpoints in {experiment_id}")
                checkpoint = ckpts[-1]
            r = self._http.get(f"/experiments/{experiment_id}/checkpoints/{checkpoint}")
            if r.status_code >= 400:
                raise ArenaError(f"checkpoint fetch -> {r.status_code}")
            local_dir = os.path.join(self.workspace, experiment_id)
            os.makedirs(local_dir, exist_ok=True)
            local_path = os.path.join(local_dir, checkpoint)
            with open(local_path, "wb") as f:
                f.write(r.content)
            from ..serve import load_agent

            return Deployment(experiment_id, checkpoint, load_agent(local_path))
        exp_dir = os.path.join(self.workspace, experiment_id)
        ckpts = self.list_checkpoints(experiment_id)
        if checkpoint is None:
            if not ckpts:
                raise ArenaError(f"no checkpoints in {experiment_id}")
            checkpoint = ckpts[-1]
        from ..serve import load_agent

        agent = load_agent(os.path.join(exp_dir, checkpoint))
        return Deployment(experiment_id, checkpoint, agent)

    # ------------------------------------------------------------------
    def _write(self, rel: str, payload: Dict[str, Any]) -> None:
        path = os.path.join(self.workspace, rel)
        os.makedirs(os.path.dirname(path), exist_ok=True)
        with open(path, "w") as f:
            json.dump(payload, f)
