"""Single-row / batch scorer over an exported model.

Python equivalent of the reference eval module's Computable implementation
(reference: shifu-tensorflow-eval/.../TensorflowModel.java:32,52-94,111-172):
`init(GenericModelConfig.json path)` loads the exported bundle;
`compute(row)` feeds one feature vector and returns the scalar sigmoid score
(the [0][0] output the Java side returns).
"""
from __future__ import annotations

import json
import os
from typing import Optional, Sequence

import numpy as np
import torch


class _SavedModelGraph:
    """Minimal predict() wrapper over a frozen saved_model.pb (numpy
    execution — train/tf_saved_model.py run_saved_model)."""

    def __init__(self, nodes):
        self.nodes = nodes

    def predict(self, dense: "torch.Tensor"):
        from shifu_amd.train.tf_saved_model import run_saved_model
        x = dense.detach().cpu().float().numpy()
        p = run_saved_model(self.nodes, {"shifu_input_0": x})
        return torch.from_numpy(np.ascontiguousarray(p))


class ShifuScorer:
    def __init__(self):
        self.model: Optional[torch.nn.Module] = None
        self.num_dense = 0
        self.num_cat = 0
        self.device = "cpu"

    def init(self, generic_model_config_path: str, device: str = "cpu") -> None:
        with open(generic_model_config_path) as f:
            gmc = json.load(f)
        if gmc.get("outputnames", ["shifu_output_0"])[0] != "shifu_output_0":
            raise ValueError("unsupported outputnames (expect shifu_output_0)")
        model_path = gmc.get("modelpath") or os.path.dirname(
            os.path.abspath(generic_model_config_path))
        if not os.path.isdir(model_path):
            model_path = os.path.dirname(os.path.abspath(generic_model_config_path))
        gj = os.path.join(model_path, "graph.json")
        if os.path.exists(gj):
            from shifu_amd.train.export import load_exported
            self.model = load_exported(model_path, device=device)
            self.device = device
            with open(gj) as f:
                spec = json.load(f)
            self.num_dense = int(spec["num_dense"])
            self.num_cat = len(spec.get("vocab_sizes", []))
            return
        # no portable layout: score straight from the TF SavedModel artifact
        # (the same file the Java SavedModelBundle path loads)
        pb = os.path.join(model_path, "saved_model.pb")
        if not os.path.exists(pb):
            raise FileNotFoundError(
                f"{model_path}: neither graph.json nor saved_model.pb")
        from shifu_amd.train.tf_saved_model import load_saved_model
        self._pb_nodes, _ = load_saved_model(pb)
        op, _, attrs = self._pb_nodes["shifu_input_0"]
        assert op == "Placeholder"
        # input width from the placeholder's declared shape
        from shifu_amd.train.tf_saved_model import _parse_shape
        dims = _parse_shape(attrs["shape"][7][0][1])
        self.num_dense = int(dims[-1])
        self.num_cat = 0
        self.model = _SavedModelGraph(self._pb_nodes)
        self.device = "cpu"

    def compute(self, row: Sequence[float]) -> float:
        """Score one row: first num_dense values are normalized floats, the
        rest (if any) categorical ids.  Returns p in [0,1]."""
        if self.model is None:
            raise RuntimeError("scorer not initialized")
        arr = np.asarray(row, dtype=np.float64)
        dense = torch.tensor(arr[:self.num_dense], dtype=torch.float32,
                             device=self.device).reshape(1, -1)
        if self.num_cat:
            cats = torch.tensor(arr[self.num_dense:self.num_dense + self.num_cat],
                                dtype=torch.int64, device=self.device).reshape(1, -1)
            p = self.model.predict(dense, cats)
        else:
            p = self.model.predict(dense)
        return float(p.reshape(-1)[0])

    def compute_batch(self, dense: np.ndarray, cats: Optional[np.ndarray] = None) -> np.ndarray:
        if self.model is None:
            raise RuntimeError("scorer not initialized")
        d = torch.tensor(np.asarray(dense, dtype=np.float32), device=self.device)
        if cats is not None and self.num_cat:
            c = torch.tensor(np.asarray(cats, dtype=np.int64), device=self.device)
            p = self.model.predict(d, c)
        else:
            p = self.model.predict(d)
        return p.cpu().numpy()
