"""Serving endpoint: score via HTTP (TestClient) against a fresh export."""
import numpy as np
import pytest
import torch

fastapi = pytest.importorskip("fastapi")
from fastapi.testclient import TestClient

from shifu_amd.models.wide_deep import WideDeep
from shifu_amd.train.export import export_model
from shifu_amd.server import create_app


@pytest.fixture()
def exported(tmp_path):
    model = WideDeep(4, [10, 20], 4, [8], ["relu"], seed=1)
    export_model(model, str(tmp_path / "final"))
    return model, str(tmp_path / "final")


def test_health_and_score(exported):
    model, path = exported
    app = create_app(path)
    client = TestClient(app)

    h = client.get("/health").json()
    assert h["status"] == "ok" and h["num_dense"] == 4 and h["num_cat"] == 2

    row = [0.1, -0.2, 0.3, 0.4, 3, 7]      # 4 dense + 2 cat ids
    r = client.post("/score", json={"rows": [row, row]}).json()
    assert len(r["scores"]) == 2
    assert 0.0 <= r["scores"][0] <= 1.0
    # parity with direct model predict
    p = model.predict(torch.tensor([row[:4]]), torch.tensor([[3, 7]]))
    assert abs(r["scores"][0] - float(p[0])) < 1e-5

    r2 = client.post("/score_named", json={
        "dense": [row[:4]], "cats": [[3, 7]]}).json()
    assert abs(r2["scores"][0] - r["scores"][0]) < 1e-6


def test_bad_input_rejected(exported):
    _, path = exported
    client = TestClient(create_app(path))
    resp = client.post("/score", json={"rows": [[1.0]]})  # too few features
    assert resp.status_code == 400


def test_reference_eval_fixture_equivalent(tmp_path):
    """Mirror of the reference's ONLY real test (TensorflowModelTest.java:35-60):
    score a 1522-feature random row through an exported model and assert the
    sigmoid output lies in [0,1]."""
    import numpy as np
    from shifu_amd.models.mlp import ShifuMLP
    from shifu_amd.serve import ShifuScorer
    model = ShifuMLP(1522, [50, 20], ["tanh", "tanh"], seed=9)
    export_model(model, str(tmp_path / "final"))
    sc = ShifuScorer()
    sc.init(str(tmp_path / "final" / "GenericModelConfig.json"))
    rng = np.random.default_rng(0)
    for _ in range(5):
        p = sc.compute(list(rng.standard_normal(1522)))
        assert 0.0 <= p <= 1.0
