"""Arena CLI: ``python -m agilerl_amd.arena.cli <command>``.

Reference parity: ``agilerl-arena/agilerl/arena/cli.py`` (the ``arena``
command): login / validate / submit / resume / status / list.
"""

from __future__ import annotations

import argparse
import json

from .client import ArenaClient


def main(argv=None):
    p = argparse.ArgumentParser(prog="arena")
    p.add_argument("--workspace", default=".arena")
    sub = p.add_subparsers(dest="cmd", required=True)
    sub.add_parser("login")
    v = sub.add_parser("validate")
    v.add_argument("manifest")
    s = sub.add_parser("submit")
    s.add_argument("manifest")
    s.add_argument("--device", default="cpu")
    s.add_argument("--no-run", action="store_true")
    r = sub.add_parser("resume")
    r.add_argument("experiment_id")
    st = sub.add_parser("status")
    st.add_argument("experiment_id")
    sub.add_parser("list")
    du = sub.add_parser("dataset-upload")
    du.add_argument("path")
    du.add_argument("--name", default=None)
    sub.add_parser("dataset-list")
    dp = sub.add_parser("deploy")
    dp.add_argument("experiment_id")
    dp.add_argument("--checkpoint", default=None)
    dp.add_argument("--port", type=int, default=8000)
    dp.add_argument("--host", default="127.0.0.1")
    args = p.parse_args(argv)

    client = ArenaClient(workspace=args.workspace)
    client.login()
    if args.cmd == "login":
        print("logged in (local workspace backend)")
    elif args.cmd == "validate":
        from ..models.manifest import TrainingManifest

        m = TrainingManifest.from_yaml(args.manifest)
        print(json.dumps(client.validate_environment(
            {**m.environment, "algorithm": m.algorithm.name}), indent=2))
    elif args.cmd == "submit":
        from ..models.manifest import TrainingManifest

        handle = client.submit_experiment(
            TrainingManifest.from_yaml(args.manifest), run=not args.no_run,
            device=args.device,
        )
        print(handle.experiment_id, handle.status)
    elif args.cmd == "resume":
        handle = client.resume_experiment(args.experiment_id)
        print(handle.experiment_id, handle.status)
    elif args.cmd == "status":
        print(json.dumps(client.experiment_status(args.experiment_id)))
    elif args.cmd == "list":
        print("\n".join(client.list_experiments()))
    elif args.cmd == "dataset-upload":
        print(client.upload_dataset(args.path, name=args.name))
    elif args.cmd == "dataset-list":
        print("\n".join(client.list_datasets()))
    elif args.cmd == "deploy":
        dep = client.deploy(args.experiment_id, checkpoint=args.checkpoint)
        print(json.dumps(dep.info()))
        import uvicorn

        from ..serve import create_app

        uvicorn.run(create_app(dep.agent), host=args.host, port=args.port)


if __name__ == "__main__":
    main()
