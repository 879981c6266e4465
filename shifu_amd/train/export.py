"""Final model export + single-row scorer.

Successor of the reference chief's export step (rebuild inference graph,
restore newest checkpoint, simple_save SavedModel + GenericModelConfig.json —
reference: ssgd_monitor.py:304-339,457-490).  The export contract kept:

* a `GenericModelConfig.json` with the exact fields the eval side reads
  (reference: shifu-tensorflow-eval/.../TensorflowModel.java:111-172):
  inputnames=["shifu_input_0"], outputnames=["shifu_output_0"],
  normtype="ZSCALE", algorithm, tags=["serve"], modelpath;
* the model itself as `model.safetensors` (tensor weights) +
  `graph.json` (architecture: layer shapes, activations, column layout) —
  a self-contained layout any runtime can reconstruct without TensorFlow.
  (The reference's binary TF SavedModel .pb is TF-runtime-specific; the
  eval-side loader for THIS layout is shifu_amd/serve.py, which implements
  the same init(config)/compute(row) interface as the Java Computable.)

The exported score is sigmoid(logits) — identical to the reference's
`shifu_output_0` sigmoid unit.
"""
from __future__ import annotations

import json
import os
from typing import Dict, List, Optional, Sequence

import torch


def _graph_spec(model: torch.nn.Module) -> Dict:
    """Introspect the supported model families into a portable JSON spec."""
    from shifu_amd.models.mlp import ShifuMLP
    from shifu_amd.models.wide_deep import WideDeep
    from shifu_amd.models.deepfm import DeepFM
    from shifu_amd.ops.linear import FusedLinear

    def layer_spec(l: FusedLinear) -> Dict:
        return {"in": l.in_features, "out": l.out_features, "act": l.activation}

    if isinstance(model, ShifuMLP):
        return {
            "family": "mlp",
            "num_dense": model.num_features,
            "hidden": [layer_spec(l) for l in model.hidden],
            "head": layer_spec(model.shifu_output_0),
        }
    if isinstance(model, (WideDeep, DeepFM)):
        fam = "wide_deep" if isinstance(model, WideDeep) else "deepfm"
        return {
            "family": fam,
            "num_dense": model.num_dense,
            "embed_dim": model.embed_dim,
            "vocab_sizes": model.vocab_sizes,
            "tower": [layer_spec(l) for l in model.tower],
            "head": layer_spec(model.shifu_output_0),
        }
    raise ValueError(f"unsupported model family for export: {type(model).__name__}")


def _split_unified_arenas(model: torch.nn.Module, state: Dict) -> Dict:
    """Unified [R, D+2] arenas export as the LOGICAL split layout (deep
    [R, D] under the original name + wide [R, 1] under the split model's
    wide-arena name), so serving, score.py and old bundles are identical for
    both training layouts (ROADMAP item 3)."""
    wide_name = {"WideDeep": "wide_cat.arena", "DeepFM": "fm_first.arena"} \
        .get(type(model).__name__)
    for name, p in list(model.named_parameters()):
        D = getattr(p, "_unified_split", None)
        if D is None or name not in state:
            continue
        full = state.pop(name)
        state[name] = full[:, :D].contiguous()
        if wide_name:
            state[wide_name] = full[:, D:D + 1].contiguous()
    return state


def export_model(model: torch.nn.Module, final_model_path: str,
                 model_name: str = "model", algorithm: str = "NN",
                 selected_columns: Optional[Sequence[int]] = None) -> str:
    """Write the export directory; returns its path."""
    os.makedirs(final_model_path, exist_ok=True)
    state = {k: v.detach().float().cpu().contiguous() for k, v in model.state_dict().items()}
    state = _split_unified_arenas(model, state)
    try:
        from safetensors.torch import save_file
        weights_file = "model.safetensors"
        save_file(state, os.path.join(final_model_path, weights_file))
    except ImportError:  # pragma: no cover
        weights_file = "model.pt"
        torch.save(state, os.path.join(final_model_path, weights_file))

    with open(os.path.join(final_model_path, "graph.json"), "w") as f:
        json.dump(_graph_spec(model), f, indent=2)

    # TF-1.x SavedModel for the Java eval drop-in (SavedModelBundle.load,
    # TensorflowModel.java:169): frozen-Const graph, dense-input families
    # only (the reference's eval path feeds a single float vector)
    from shifu_amd.models.mlp import ShifuMLP
    if isinstance(model, ShifuMLP):
        from shifu_amd.train.tf_saved_model import (emit_saved_model,
                                                    layers_from_mlp)
        emit_saved_model(final_model_path, layers_from_mlp(model),
                         model.num_features)

    # GenericModelConfig.json — field-compatible with the reference's
    # export_generic_config (ssgd_monitor.py:476-490)
    gmc = {
        "inputnames": ["shifu_input_0"],
        "properties": {"algorithm": algorithm, "tags": ["serve"],
                       "normtype": "ZSCALE", "weightsfile": weights_file},
        "outputnames": ["shifu_output_0"],
        "modelpath": os.path.abspath(final_model_path),
    }
    if selected_columns is not None:
        gmc["properties"]["selectedcolumns"] = list(selected_columns)
    with open(os.path.join(final_model_path, "GenericModelConfig.json"), "w") as f:
        json.dump(gmc, f, indent=2)
    return final_model_path


def load_exported(final_model_path: str, device: str = "cpu") -> torch.nn.Module:
    """Reconstruct an exported model from graph.json + weights (the scorer path)."""
    from shifu_amd.models.mlp import ShifuMLP
    from shifu_amd.models.wide_deep import WideDeep
    from shifu_amd.models.deepfm import DeepFM

    with open(os.path.join(final_model_path, "graph.json")) as f:
        spec = json.load(f)
    fam = spec["family"]
    if fam == "mlp":
        model = ShifuMLP(spec["num_dense"],
                         [l["out"] for l in spec["hidden"]],
                         [l["act"] for l in spec["hidden"]])
    else:
        cls = WideDeep if fam == "wide_deep" else DeepFM
        model = cls(spec["num_dense"], spec["vocab_sizes"], spec["embed_dim"],
                    [l["out"] for l in spec["tower"]],
                    [l["act"] for l in spec["tower"]])

    weights = os.path.join(final_model_path, "model.safetensors")
    if os.path.exists(weights):
        from safetensors.torch import load_file
        state = load_file(weights)
    else:
        state = torch.load(os.path.join(final_model_path, "model.pt"),
                           map_location="cpu", weights_only=True)
    model.load_state_dict(state)
    return model.to(device).eval()
