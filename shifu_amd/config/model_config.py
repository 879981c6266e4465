"""Shifu ModelConfig.json / ColumnConfig.json parsing.

These two JSON files are the public API surface shared with the Shifu
pipeline.  The reference reads ModelConfig.json directly from the container
working directory (reference: shifu-tensorflow-on-yarn/src/main/resources/
ssgd_monitor.py:178-183) and consumes:

    train.numTrainEpochs
    train.validSetRate
    train.params.NumHiddenLayers
    train.params.NumHiddenNodes   (list[int])
    train.params.ActivationFunc   (list[str], per hidden layer)
    train.params.LearningRate

ColumnConfig.json is uploaded but parsed only by Shifu core in the reference
(column indices arrive pre-digested via env vars — SURVEY.md §2.5).  Here we
parse it natively so the framework is self-contained: selected columns,
target/weight columns, and numeric-vs-categorical typing for the
Wide&Deep/DeepFM embedding path.
"""
from __future__ import annotations

import json
import os
from dataclasses import dataclass, field
from typing import Any, Dict, List, Optional

# Activations supported by the reference (ssgd_monitor.py:74-88).
SUPPORTED_ACTIVATIONS = ("sigmoid", "tanh", "relu", "leakyrelu")

# Optimizers the reference instantiates.
SUPPORTED_OPTIMIZERS = ("adadelta", "adam", "sgd", "adagrad")


@dataclass
class TrainParams:
    num_hidden_layers: int = 2
    num_hidden_nodes: List[int] = field(default_factory=lambda: [50, 50])
    activation_funcs: List[str] = field(default_factory=lambda: ["tanh", "tanh"])
    learning_rate: float = 1.0
    optimizer: str = "adadelta"          # reference default: AdadeltaOptimizer (ssgd_monitor.py:138)
    l2_reg: float = 0.1                  # l2_regularizer(0.1) on every dense layer (ssgd_monitor.py:58-68)
    loss: str = "weighted_mse"           # tf.losses.mean_squared_error w/ sample weights (ssgd_monitor.py:129)
    batch_size: int = 100                # BATCH_SIZE=100 (ssgd_monitor.py:33)
    # Local-SGD / gradient-accumulation window mode (SAGN.py:111-167):
    # all-reduce every `update_window` local steps instead of every step.
    update_window: int = 1

    def __post_init__(self):
        self.activation_funcs = [a.lower() for a in self.activation_funcs]
        for a in self.activation_funcs:
            if a not in SUPPORTED_ACTIVATIONS:
                raise ValueError(f"unsupported activation {a!r}; supported: {SUPPORTED_ACTIVATIONS}")
        if self.optimizer.lower() not in SUPPORTED_OPTIMIZERS:
            raise ValueError(f"unsupported optimizer {self.optimizer!r}; supported: {SUPPORTED_OPTIMIZERS}")
        self.optimizer = self.optimizer.lower()
        if len(self.num_hidden_nodes) != self.num_hidden_layers:
            raise ValueError(
                f"NumHiddenNodes has {len(self.num_hidden_nodes)} entries but "
                f"NumHiddenLayers={self.num_hidden_layers}")
        if len(self.activation_funcs) == 1 and self.num_hidden_layers > 1:
            self.activation_funcs = self.activation_funcs * self.num_hidden_layers
        if len(self.activation_funcs) != self.num_hidden_layers:
            raise ValueError(
                f"ActivationFunc has {len(self.activation_funcs)} entries but "
                f"NumHiddenLayers={self.num_hidden_layers}")


@dataclass
class ModelConfig:
    """Parsed ModelConfig.json (the subset of Shifu's schema this framework uses)."""
    num_train_epochs: int = 10
    valid_set_rate: float = 0.2
    params: TrainParams = field(default_factory=TrainParams)
    algorithm: str = "NN"
    model_name: str = "model"
    data_delimiter: str = "|"
    raw: Dict[str, Any] = field(default_factory=dict)

    @classmethod
    def from_dict(cls, d: Dict[str, Any]) -> "ModelConfig":
        train = d.get("train", {}) or {}
        p = train.get("params", {}) or {}
        nhl = int(p.get("NumHiddenLayers", 2))
        nodes = p.get("NumHiddenNodes", [50] * nhl)
        acts = p.get("ActivationFunc", ["tanh"] * nhl)
        params = TrainParams(
            num_hidden_layers=nhl,
            num_hidden_nodes=[int(x) for x in nodes],
            activation_funcs=list(acts),
            learning_rate=float(p.get("LearningRate", 1.0)),
            optimizer=str(p.get("Optimizer", p.get("optimizer", "adadelta"))),
            l2_reg=float(p.get("L2Reg", p.get("RegularizedConstant", 0.1))),
            loss=str(p.get("Loss", "weighted_mse")).lower(),
            batch_size=int(p.get("MiniBatchSize", p.get("BatchSize", 100))),
            update_window=int(p.get("UpdateWindow", 1)),
        )
        ds = d.get("dataSet", {}) or {}
        basic = d.get("basic", {}) or {}
        return cls(
            num_train_epochs=int(train.get("numTrainEpochs", 10)),
            valid_set_rate=float(train.get("validSetRate", 0.2)),
            params=params,
            algorithm=str(train.get("algorithm", "NN")),
            model_name=str(basic.get("name", "model")),
            data_delimiter=str(ds.get("dataDelimiter", "|")),
            raw=d,
        )

    @classmethod
    def load(cls, path: str) -> "ModelConfig":
        with open(path, "r") as f:
            return cls.from_dict(json.load(f))

    def to_dict(self) -> Dict[str, Any]:
        return {
            "basic": {"name": self.model_name},
            "dataSet": {"dataDelimiter": self.data_delimiter},
            "train": {
                "numTrainEpochs": self.num_train_epochs,
                "validSetRate": self.valid_set_rate,
                "algorithm": self.algorithm,
                "params": {
                    "NumHiddenLayers": self.params.num_hidden_layers,
                    "NumHiddenNodes": self.params.num_hidden_nodes,
                    "ActivationFunc": self.params.activation_funcs,
                    "LearningRate": self.params.learning_rate,
                    "Optimizer": self.params.optimizer,
                    "L2Reg": self.params.l2_reg,
                    "Loss": self.params.loss,
                    "MiniBatchSize": self.params.batch_size,
                    "UpdateWindow": self.params.update_window,
                },
            },
        }

    def save(self, path: str) -> None:
        os.makedirs(os.path.dirname(os.path.abspath(path)), exist_ok=True)
        with open(path, "w") as f:
            json.dump(self.to_dict(), f, indent=2)


@dataclass
class Column:
    column_num: int
    column_name: str
    column_type: str = "N"     # "N" numeric | "C" categorical
    final_select: bool = False
    column_flag: Optional[str] = None  # "Target" | "Weight" | "ForceSelect" | "Meta" | None
    vocab_size: int = 0        # categorical: number of distinct bins/values

    @property
    def is_target(self) -> bool:
        return (self.column_flag or "").lower() == "target" or self.column_type == "T"

    @property
    def is_weight(self) -> bool:
        return (self.column_flag or "").lower() == "weight"

    @property
    def is_categorical(self) -> bool:
        return self.column_type.upper().startswith("C")


@dataclass
class ColumnConfig:
    """Parsed ColumnConfig.json: a list of column descriptors."""
    columns: List[Column] = field(default_factory=list)

    @classmethod
    def from_list(cls, items: List[Dict[str, Any]]) -> "ColumnConfig":
        cols = []
        for it in items:
            stats = it.get("columnStats", {}) or {}
            binning = it.get("columnBinning", {}) or {}
            bins = binning.get("binCategory") or []
            cols.append(Column(
                column_num=int(it.get("columnNum", len(cols))),
                column_name=str(it.get("columnName", f"col_{len(cols)}")),
                column_type=str(it.get("columnType", "N") or "N"),
                final_select=bool(it.get("finalSelect", False)),
                column_flag=it.get("columnFlag"),
                vocab_size=int(it.get("vocabSize", stats.get("distinctCount", len(bins)) or 0)),
            ))
        return cls(cols)

    @classmethod
    def load(cls, path: str) -> "ColumnConfig":
        with open(path, "r") as f:
            return cls.from_list(json.load(f))

    def save(self, path: str) -> None:
        items = []
        for c in self.columns:
            items.append({
                "columnNum": c.column_num,
                "columnName": c.column_name,
                "columnType": c.column_type,
                "finalSelect": c.final_select,
                "columnFlag": c.column_flag,
                "vocabSize": c.vocab_size,
            })
        os.makedirs(os.path.dirname(os.path.abspath(path)), exist_ok=True)
        with open(path, "w") as f:
            json.dump(items, f, indent=2)

    # -- derived views the trainer consumes (successor of the env-var contract
    #    SELECTED_COLUMN_NUMS / TARGET_COLUMN_NUM / WEIGHT_COLUMN_NUM,
    #    reference: TensorflowTaskExecutor.java:200-238) --
    @property
    def target_column(self) -> int:
        for c in self.columns:
            if c.is_target:
                return c.column_num
        return 0  # reference default: TARGET_COLUMN_NUM=0

    @property
    def weight_column(self) -> int:
        for c in self.columns:
            if c.is_weight:
                return c.column_num
        return -1  # reference default: WEIGHT_COLUMN_NUM=-1 (all-ones weights)

    @property
    def selected_numeric_columns(self) -> List[int]:
        return [c.column_num for c in self.columns
                if c.final_select and not c.is_categorical and not c.is_target and not c.is_weight]

    @property
    def selected_categorical_columns(self) -> List[int]:
        return [c.column_num for c in self.columns
                if c.final_select and c.is_categorical and not c.is_target and not c.is_weight]

    @property
    def selected_columns(self) -> List[int]:
        return sorted(self.selected_numeric_columns + self.selected_categorical_columns)

    def vocab_sizes(self) -> Dict[int, int]:
        return {c.column_num: max(c.vocab_size, 1)
                for c in self.columns if c.final_select and c.is_categorical}
