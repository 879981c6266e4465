"""ModelConfig / ColumnConfig / RunConfig parsing tests (SURVEY.md §2.5 contract)."""
import json

import pytest

from shifu_amd.config.model_config import ColumnConfig, ModelConfig
from shifu_amd.config.run_config import RunConfig


MODEL_CONFIG = {
    "basic": {"name": "demo"},
    "dataSet": {"dataDelimiter": "|"},
    "train": {
        "numTrainEpochs": 5,
        "validSetRate": 0.3,
        "algorithm": "NN",
        "params": {
            "NumHiddenLayers": 3,
            "NumHiddenNodes": [30, 20, 10],
            "ActivationFunc": ["tanh", "relu", "sigmoid"],
            "LearningRate": 0.5,
        },
    },
}


def test_model_config_from_dict():
    mc = ModelConfig.from_dict(MODEL_CONFIG)
    assert mc.num_train_epochs == 5
    assert mc.valid_set_rate == 0.3
    assert mc.params.num_hidden_nodes == [30, 20, 10]
    assert mc.params.activation_funcs == ["tanh", "relu", "sigmoid"]
    assert mc.params.learning_rate == 0.5
    assert mc.params.optimizer == "adadelta"  # reference default
    assert mc.params.l2_reg == 0.1


def test_model_config_roundtrip(tmp_path):
    mc = ModelConfig.from_dict(MODEL_CONFIG)
    p = str(tmp_path / "ModelConfig.json")
    mc.save(p)
    mc2 = ModelConfig.load(p)
    assert mc2.params.num_hidden_nodes == mc.params.num_hidden_nodes
    assert mc2.num_train_epochs == mc.num_train_epochs


def test_single_activation_broadcast():
    d = json.loads(json.dumps(MODEL_CONFIG))
    d["train"]["params"]["ActivationFunc"] = ["tanh"]
    mc = ModelConfig.from_dict(d)
    assert mc.params.activation_funcs == ["tanh"] * 3


def test_bad_activation_rejected():
    d = json.loads(json.dumps(MODEL_CONFIG))
    d["train"]["params"]["ActivationFunc"] = ["selu", "selu", "selu"]
    with pytest.raises(ValueError):
        ModelConfig.from_dict(d)


def test_column_config_views():
    cc = ColumnConfig.from_list([
        {"columnNum": 0, "columnName": "target", "columnFlag": "Target", "columnType": "N"},
        {"columnNum": 1, "columnName": "wgt", "columnFlag": "Weight", "columnType": "N"},
        {"columnNum": 2, "columnName": "a", "finalSelect": True, "columnType": "N"},
        {"columnNum": 3, "columnName": "b", "finalSelect": True, "columnType": "C",
         "vocabSize": 100},
        {"columnNum": 4, "columnName": "c", "finalSelect": False, "columnType": "N"},
    ])
    assert cc.target_column == 0
    assert cc.weight_column == 1
    assert cc.selected_numeric_columns == [2]
    assert cc.selected_categorical_columns == [3]
    assert cc.vocab_sizes() == {3: 100}


def test_run_config_roundtrip(tmp_path):
    rc = RunConfig(num_gpus=8, training_data_path=["/tmp/x.csv"], bucket_mb=256)
    p = str(tmp_path / "run.json")
    rc.save(p)
    rc2 = RunConfig.load(p)
    assert rc2.num_gpus == 8
    assert rc2.bucket_mb == 256
    assert rc2.training_data_path == ["/tmp/x.csv"]


def test_run_config_apply_column_config():
    cc = ColumnConfig.from_list([
        {"columnNum": 0, "columnFlag": "Target"},
        {"columnNum": 2, "finalSelect": True, "columnType": "N"},
    ])
    rc = RunConfig()
    rc.apply_column_config(cc)
    assert rc.target_column == 0
    assert rc.selected_numeric_columns == [2]
    assert rc.weight_column == -1
