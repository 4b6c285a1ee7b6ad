"""Deployment-parity validation: no container
runtime is reachable in the build environment, so the compose topology is
validated structurally against the reference's 5-worker edge simulation
(``/root/reference/docker-compose.yml:1-67``): 5 workers, RANK 0-4,
WORLD_SIZE=5, rendezvous worker1:29500, 0.4 CPU / 1 GB caps, shared ./data
volume, and a Dockerfile whose CMD is the env-var entry ``train.py``."""
import os

import yaml

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def _load():
    with open(os.path.join(REPO, "docker-compose.yml")) as f:
        return yaml.safe_load(f)


def test_compose_topology_parity():
    doc = _load()
    services = doc["services"]
    assert sorted(services) == [f"worker{i}" for i in range(1, 6)], \
        "reference topology is exactly 5 workers"
    ranks = set()
    for name, svc in services.items():
        env = svc["environment"]
        assert str(env["WORLD_SIZE"]) == "5"
        assert env["MASTER_ADDR"] == "worker1"
        assert str(env["MASTER_PORT"]) == "29500"
        ranks.add(int(env["RANK"]))
        # resource caps: 0.4 CPU / 1 GB per worker (edge-node simulation)
        limits = svc["deploy"]["resources"]["limits"]
        assert float(limits["cpus"]) == 0.4
        assert str(limits["memory"]).lower() in ("1g", "1gb", "1073741824")
        # shared dataset volume
        assert any(v.startswith("./data:") for v in svc["volumes"])
        assert svc["build"] == "."
    assert ranks == {0, 1, 2, 3, 4}, "RANK 0-4, one per worker"
    # master is rank 0's own container (worker1)
    assert int(services["worker1"]["environment"]["RANK"]) == 0


def test_dockerfile_runs_env_entry():
    with open(os.path.join(REPO, "Dockerfile")) as f:
        content = f.read()
    assert "train.py" in content, "CMD must launch the env-var entry"
