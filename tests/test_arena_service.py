"""Arena HTTP service + client transport (reference agilerl-arena SDK:
auth, validation, experiments, checkpoints, datasets, NDJSON streaming —
client.py:128-986).  The FastAPI app mounts in-process via httpx's ASGI
transport, so the full request path (serialization, status codes,
streaming) runs without sockets."""

import time

import httpx
import pytest

from agilerl_amd.arena.client import ArenaClient, ArenaError
from agilerl_amd.arena.service import ArenaService, create_app
from agilerl_amd.arena.stream import NDJsonStream, StreamEvent


@pytest.fixture()
def arena(tmp_path):
    from starlette.testclient import TestClient

    service = ArenaService(str(tmp_path / "server"))
    app = create_app(service)
    # TestClient IS an httpx.Client over an in-process ASGI portal
    http = TestClient(app)
    client = ArenaClient(base_url="http://testserver", http_client=http,
                         workspace=str(tmp_path / "client"))
    return service, client


TINY_MANIFEST = {
    "algorithm": {"name": "DQN", "batch_size": 16, "lr": 1e-3},
    "environment": {"name": "CartPole-v1", "num_envs": 2},
    "training": {"max_steps": 300, "pop_size": 2, "evo_steps": 100},
    "selection_strategy": {"tournament_size": 2, "elitism": True},
}


class TestArenaHttp:
    def test_login_issues_token(self, arena):
        _, client = arena
        assert client.login()
        assert client.api_key.startswith("tok-")

    def test_validate_environment(self, arena):
        _, client = arena
        report = client.validate_environment({"name": "CartPole-v1", "num_envs": 2})
        assert report["valid"] is True
        report2 = client.validate_environment({"name": "NotAnEnv-v9"})
        assert report2["warnings"]

    def test_submit_run_stream_checkpoints_deploy(self, arena):
        service, client = arena
        client.login()
        handle = client.submit_experiment(TINY_MANIFEST)
        assert handle.experiment_id.startswith("exp-")
        status = client.wait_for_completion(handle.experiment_id, timeout=240)
        assert status["status"] == "completed", status
        assert handle.experiment_id in client.list_experiments()

        # NDJSON stream: status transitions + at least one metrics event
        seen = []
        client.set_stream_handler(seen.append)
        events = list(client.stream_experiment(handle.experiment_id))
        kinds = [e.kind for e in events]
        assert "status" in kinds and "done" in kinds
        assert any(e.payload.get("status") == "completed" for e in events
                   if e.kind == "status")
        assert seen == events  # handler dispatched for every event

        # checkpoints listed and deployable through the HTTP fetch path
        ckpts = client.list_checkpoints(handle.experiment_id)
        assert ckpts, "training must have produced a checkpoint"
        dep = client.deploy(handle.experiment_id)
        import numpy as np

        action = dep.predict(np.zeros(4, dtype=np.float32))
        assert int(action) in (0, 1)

    def test_dataset_registry_round_trip(self, arena, tmp_path):
        _, client = arena
        client.login()
        p = tmp_path / "data.json"
        p.write_text('{"rows": [1, 2, 3]}')
        ds_id = client.upload_dataset(str(p), name="rows")
        assert ds_id == "ds-rows"
        assert "ds-rows" in client.list_datasets()

    def test_unknown_experiment_404(self, arena):
        _, client = arena
        client.login()
        with pytest.raises(ArenaError, match="404"):
            client.experiment_status("exp-nope")
        with pytest.raises(ArenaError, match="404"):
            client.list_checkpoints("exp-nope")

    def test_invalid_manifest_422(self, arena):
        _, client = arena
        client.login()
        # client-side validation catches it before the wire...
        with pytest.raises(Exception, match="Unknown algorithm"):
            client.submit_experiment({
                "algorithm": {"name": "NotARealAlgo"},
                "environment": {"name": "CartPole-v1"},
            })
        # ...and the SERVER independently rejects a raw bad payload with 422
        with pytest.raises(ArenaError, match="422"):
            client._post("/experiments", json={
                "manifest": {"algorithm": {"name": "NotARealAlgo"},
                             "environment": {"name": "CartPole-v1"}},
            })


class TestNDJsonStream:
    def test_parses_and_skips_garbage(self):
        lines = [
            StreamEvent("status", "e1", {"status": "running"}, ts=1.0).to_json(),
            "",
            "not json at all {",
            StreamEvent("metrics", "e1", {"best_fitness": 3.2}).to_json(),
        ]
        events = list(NDJsonStream(lines))
        assert [e.kind for e in events] == ["status", "metrics"]
        assert events[1].payload["best_fitness"] == 3.2

    def test_bytes_lines(self):
        raw = [StreamEvent("done", "e2", {}).to_json().encode()]
        events = list(NDJsonStream(raw))
        assert events[0].kind == "done" and events[0].experiment_id == "e2"
