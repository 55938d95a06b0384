"""No-import-change CLI + connect server tests (reference
tests_no_import_change/ + jvm plugin suite analogs)."""

import json
import os
import subprocess
import sys

import numpy as np
import pytest


def test_install_aliases():
    from spark_rapids_ml_amd.install import install_aliases

    install_aliases()
    import spark_rapids_ml  # noqa: F401 - the alias

    from spark_rapids_ml.clustering import KMeans
    from spark_rapids_ml.classification import LogisticRegression
    from spark_rapids_ml.tuning import CrossValidator, ParamGridBuilder

    import spark_rapids_ml_amd

    assert KMeans is spark_rapids_ml_amd.KMeans
    assert LogisticRegression is spark_rapids_ml_amd.LogisticRegression


def test_runner_executes_reference_style_script(tmp_path):
    script = tmp_path / "app.py"
    script.write_text(
        """
import numpy as np
from spark_rapids_ml.clustering import KMeans
from spark_rapids_ml_amd.data import DataFrame

X = np.random.default_rng(0).normal(size=(200, 8)).astype("float32")
model = KMeans(k=3, maxIter=5, seed=1).fit(DataFrame.from_numpy(X))
print("CENTERS", model.cluster_centers_.shape)
"""
    )
    repo = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    out = subprocess.run(
        [sys.executable, "-m", "spark_rapids_ml_amd", str(script)],
        capture_output=True,
        text=True,
        cwd=repo,
        timeout=300,
    )
    assert out.returncode == 0, out.stderr
    assert "CENTERS (3, 8)" in out.stdout


@pytest.fixture
def client(tmp_path, monkeypatch):
    from starlette.testclient import TestClient

    from spark_rapids_ml_amd.connect_server import create_app

    # the server confines data/output paths to SRML_SERVER_DATA_ROOT
    # (defaulting to its CWD); point it at the test's tmp dir
    monkeypatch.setenv("SRML_SERVER_DATA_ROOT", str(tmp_path))
    return TestClient(create_app())


def test_server_fit_transform_roundtrip(client, tmp_path):
    from spark_rapids_ml_amd.data import DataFrame

    rng = np.random.default_rng(0)
    X = rng.normal(size=(300, 8)).astype(np.float32)
    data_path = str(tmp_path / "data")
    DataFrame.from_numpy(X).write_parquet(data_path)

    r = client.post(
        "/fit",
        json={"estimator": "KMeans", "params": {"k": 4, "maxIter": 5}, "data_path": data_path},
    )
    assert r.status_code == 200, r.text
    body = r.json()
    mid = body["model_id"]
    assert np.asarray(body["attributes"]["cluster_centers_"]).shape == (4, 8)

    out_path = str(tmp_path / "out")
    r2 = client.post(
        "/transform",
        json={"model_id": mid, "data_path": data_path, "output_path": out_path},
    )
    assert r2.status_code == 200, r2.text
    assert "prediction" in r2.json()["columns"]
    back = DataFrame.read_parquet(out_path)
    assert back.num_rows == 300

    r3 = client.get("/models")
    assert mid in r3.json()

    save_path = str(tmp_path / "saved")
    r4 = client.post(f"/models/{mid}/save", json={"path": save_path})
    assert r4.status_code == 200
    from spark_rapids_ml_amd import KMeansModel

    loaded = KMeansModel.load(save_path)
    assert loaded.cluster_centers_.shape == (4, 8)


def test_server_bad_estimator(client):
    r = client.post("/fit", json={"estimator": "Nope", "params": {}, "data_path": "/none"})
    assert r.status_code == 400


def test_launch_single_rank(tmp_path):
    script = tmp_path / "app.py"
    script.write_text(
        "import spark_rapids_ml\n"
        "from spark_rapids_ml_amd.parallel.context import get_comm\n"
        "print('LAUNCH_OK', get_comm().world_size)\n"
    )
    r = subprocess.run(
        [sys.executable, "-m", "spark_rapids_ml_amd.launch", "--gpus", "0", str(script)],
        capture_output=True, text=True, timeout=120,
    )
    assert r.returncode == 0, r.stderr
    assert "LAUNCH_OK 1" in r.stdout


def test_launch_two_ranks_gloo(tmp_path):
    script = tmp_path / "app.py"
    script.write_text(
        "from spark_rapids_ml_amd.parallel.context import get_comm\n"
        "c = get_comm()\n"
        "print(f'LAUNCH_RANK {c.rank}/{c.world_size}')\n"
    )
    import random

    env = dict(os.environ, SRML_BACKEND="gloo")
    r = None
    for _ in range(2):  # retry once on a port collision
        port = str(random.randint(21000, 32000))  # below the ephemeral range
        r = subprocess.run(
            [sys.executable, "-m", "spark_rapids_ml_amd.launch", "--gpus", "2",
             "--master-port", port, str(script)],
            env=env, capture_output=True, text=True, timeout=240,
        )
        if r.returncode == 0:
            break
    assert r.returncode == 0, r.stderr
    assert "LAUNCH_RANK 0/2" in r.stdout
    assert "LAUNCH_RANK 1/2" in r.stdout


def test_server_rejects_out_of_root_paths(client):
    # default confinement (advisor finding): without an explicit wider root,
    # arbitrary filesystem paths must be refused
    r = client.post(
        "/fit",
        json={"estimator": "KMeans", "params": {"k": 2}, "data_path": "/etc/passwd"},
    )
    assert r.status_code == 400
