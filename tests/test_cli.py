"""End-to-end CLI test: ColumnConfig-driven run through shifu_amd.run.main
(the `shifu train` successor path)."""
import json
import os

import pytest

from shifu_amd.data.synthetic import generate_synthetic_csv


def test_run_main_end_to_end(tmp_path):
    from shifu_amd.run import main
    data_dir = str(tmp_path / "data")
    generate_synthetic_csv(data_dir, n_rows=600, n_dense=4, vocab_sizes=[20, 30],
                           n_files=2, seed=9)

    # ColumnConfig describing the generated layout: target=0, weight=1,
    # dense 2..5, categorical 6..7
    cc = [{"columnNum": 0, "columnName": "target", "columnFlag": "Target"},
          {"columnNum": 1, "columnName": "w", "columnFlag": "Weight"}]
    cc += [{"columnNum": i, "columnName": f"d{i}", "finalSelect": True,
            "columnType": "N"} for i in range(2, 6)]
    cc += [{"columnNum": 6, "columnName": "c0", "finalSelect": True,
            "columnType": "C", "vocabSize": 20},
           {"columnNum": 7, "columnName": "c1", "finalSelect": True,
            "columnType": "C", "vocabSize": 30}]
    cc_path = str(tmp_path / "ColumnConfig.json")
    with open(cc_path, "w") as f:
        json.dump(cc, f)

    mc = {"train": {"numTrainEpochs": 2, "validSetRate": 0.2,
                    "params": {"NumHiddenLayers": 2, "NumHiddenNodes": [16, 8],
                               "ActivationFunc": ["relu", "relu"],
                               "LearningRate": 0.02, "Optimizer": "adam",
                               "Loss": "sigmoid_ce", "MiniBatchSize": 64,
                               "L2Reg": 0.0}}}
    mc_path = str(tmp_path / "ModelConfig.json")
    with open(mc_path, "w") as f:
        json.dump(mc, f)

    run_cfg = {
        "num_gpus": 2, "backend": "gloo", "master_port": 29731,
        "training_data_path": [data_dir],
        "tmp_model_path": str(tmp_path / "ckpt"),
        "final_model_path": str(tmp_path / "final"),
        "log_dir": str(tmp_path / "logs"),
        "model_type": "wide_deep", "embed_dim": 4,
        "device": "cpu",
    }
    rc_path = str(tmp_path / "run.json")
    with open(rc_path, "w") as f:
        json.dump(run_cfg, f)

    rcode = main(["--run-config", rc_path, "--model-config", mc_path,
                  "--column-config", cc_path])
    assert rcode == 0
    # exported artifacts
    final = tmp_path / "final"
    assert (final / "GenericModelConfig.json").exists()
    gmc = json.loads((final / "GenericModelConfig.json").read_text())
    assert gmc["outputnames"] == ["shifu_output_0"]
    graph = json.loads((final / "graph.json").read_text())
    assert graph["family"] == "wide_deep"
    assert graph["vocab_sizes"] == [20, 30]
    # progress board written
    board = (tmp_path / "logs" / "progress.board").read_text()
    assert "epoch 0:" in board and "epoch 1:" in board
    # the export is self-contained: at world=2 the run used EP-sharded
    # arenas, which must have been consolidated to full arenas for export
    import torch
    from shifu_amd.train.export import load_exported
    m = load_exported(str(final))
    p = m.predict(torch.zeros(1, 4), torch.tensor([[3, 7]]))
    assert 0.0 <= float(p[0]) <= 1.0


def test_score_cli_end_to_end(tmp_path, capfd):
    """Batch-scoring CLI over an exported model: scores every row, valid
    probabilities, AUC reported against the target column."""
    import torch
    from shifu_amd.models.wide_deep import WideDeep
    from shifu_amd.train.export import export_model
    from shifu_amd.score import main as score_main

    data_dir = str(tmp_path / "data")
    generate_synthetic_csv(data_dir, n_rows=400, n_dense=4, vocab_sizes=[20, 30],
                           n_files=2, seed=11)
    cc = [{"columnNum": 0, "columnName": "target", "columnFlag": "Target"},
          {"columnNum": 1, "columnName": "w", "columnFlag": "Weight"}]
    cc += [{"columnNum": i, "columnName": f"d{i}", "finalSelect": True,
            "columnType": "N"} for i in range(2, 6)]
    cc += [{"columnNum": 6, "columnName": "c0", "finalSelect": True,
            "columnType": "C", "vocabSize": 20},
           {"columnNum": 7, "columnName": "c1", "finalSelect": True,
            "columnType": "C", "vocabSize": 30}]
    cc_path = str(tmp_path / "ColumnConfig.json")
    with open(cc_path, "w") as f:
        json.dump(cc, f)

    model = WideDeep(4, [20, 30], 4, [8], ["relu"], seed=2)
    export_model(model, str(tmp_path / "final"))

    out_path = str(tmp_path / "scores.csv")
    rc = score_main(["--model", str(tmp_path / "final"), "--data", data_dir,
                     "--column-config", cc_path, "--output", out_path, "--auc"])
    assert rc == 0
    lines = open(out_path).read().strip().splitlines()
    assert len(lines) == 400
    vals = [float(l) for l in lines]
    assert all(0.0 <= v <= 1.0 for v in vals)
    err = capfd.readouterr().err
    summary = json.loads(err.strip().splitlines()[-1])
    assert summary["rows"] == 400 and "auc" in summary


def test_score_cli_distributed_2rank(tmp_path):
    """Distributed eval: the scoring CLI under torchrun (2 ranks, gloo) —
    file shards scored per rank, per-rank part files, whole-set AUC from
    rank 0 (successor of the reference's parallel Hadoop eval job)."""
    import subprocess
    import sys
    from shifu_amd.models.wide_deep import WideDeep
    from shifu_amd.train.export import export_model

    data_dir = str(tmp_path / "data")
    generate_synthetic_csv(data_dir, n_rows=400, n_dense=4, vocab_sizes=[20, 30],
                           n_files=4, seed=13)
    cc = [{"columnNum": 0, "columnName": "target", "columnFlag": "Target"},
          {"columnNum": 1, "columnName": "w", "columnFlag": "Weight"}]
    cc += [{"columnNum": i, "columnName": f"d{i}", "finalSelect": True,
            "columnType": "N"} for i in range(2, 6)]
    cc += [{"columnNum": 6, "columnName": "c0", "finalSelect": True,
            "columnType": "C", "vocabSize": 20},
           {"columnNum": 7, "columnName": "c1", "finalSelect": True,
            "columnType": "C", "vocabSize": 30}]
    cc_path = str(tmp_path / "ColumnConfig.json")
    with open(cc_path, "w") as f:
        json.dump(cc, f)

    model = WideDeep(4, [20, 30], 4, [8], ["relu"], seed=2)
    export_model(model, str(tmp_path / "final"))

    out_path = str(tmp_path / "scores.csv")
    r = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
         "--master-port", "29761", "-m", "shifu_amd.score",
         "--model", str(tmp_path / "final"), "--data", data_dir,
         "--column-config", cc_path, "--output", out_path, "--auc"],
        capture_output=True, text=True, timeout=600)
    assert r.returncode == 0, r.stderr[-2000:]
    parts = [open(f"{out_path}.part{i}").read().strip().splitlines()
             for i in range(2)]
    assert sum(len(p) for p in parts) == 400
    assert all(0.0 <= float(v) <= 1.0 for p in parts for v in p)
    summary = json.loads([l for l in r.stderr.strip().splitlines()
                          if l.startswith("{")][-1])
    assert summary["rows"] == 400 and summary["ranks"] == 2 and "auc" in summary

    # single-process result over the same data must agree per file shard
    from shifu_amd.score import main as score_main
    sp = str(tmp_path / "sp.csv")
    assert score_main(["--model", str(tmp_path / "final"), "--data", data_dir,
                       "--column-config", cc_path, "--output", sp]) == 0
    n_single = len(open(sp).read().strip().splitlines())
    assert n_single == 400


def test_module_alias_help():
    """`python -m shifu_amd` is the documented CLI alias for shifu_amd.run."""
    import subprocess
    import sys
    r = subprocess.run([sys.executable, "-m", "shifu_amd", "--help"],
                       capture_output=True, text=True, timeout=120)
    assert r.returncode == 0
    assert "--run-config" in r.stdout


def test_quickstart_example(tmp_path):
    """examples/quickstart must run end to end exactly as its README says
    (data -> train -> score with AUC)."""
    import shutil
    import subprocess
    import sys
    ex = os.path.join(os.path.dirname(os.path.dirname(os.path.abspath(__file__))),
                      "examples", "quickstart")
    work = tmp_path / "qs"
    shutil.copytree(ex, work)
    env = dict(os.environ,
               PYTHONPATH=os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
    # fewer epochs for test speed
    mc = json.loads((work / "ModelConfig.json").read_text())
    mc["train"]["numTrainEpochs"] = 2
    (work / "ModelConfig.json").write_text(json.dumps(mc))
    r = subprocess.run([sys.executable, "make_data.py", "./data"], cwd=work,
                       env=env, capture_output=True, text=True, timeout=300)
    assert r.returncode == 0, r.stderr[-1000:]
    r = subprocess.run([sys.executable, "-m", "shifu_amd.run",
                        "--run-config", "run.json",
                        "--model-config", "ModelConfig.json",
                        "--column-config", "ColumnConfig.json"],
                       cwd=work, env=env, capture_output=True, text=True,
                       timeout=900)
    assert r.returncode == 0, r.stderr[-2000:]
    assert (work / "final_model" / "GenericModelConfig.json").exists()
    r = subprocess.run([sys.executable, "-m", "shifu_amd.score",
                        "--model", "final_model", "--data", "./data",
                        "--column-config", "ColumnConfig.json",
                        "--output", "scores.csv", "--auc"],
                       cwd=work, env=env, capture_output=True, text=True,
                       timeout=600)
    assert r.returncode == 0, r.stderr[-2000:]
    assert len((work / "scores.csv").read_text().strip().splitlines()) == 20000
    summary = json.loads([l for l in r.stderr.strip().splitlines()
                          if l.startswith("{")][-1])
    assert summary["auc"] > 0.6, f"quickstart model failed to learn: {summary}"
