"""TF-1.x SavedModel emitter — the Java eval drop-in artifact.

The reference's scoring side loads the chief's export with the TensorFlow
Java API: `SavedModelBundle.load(modelPath, tags)` then feeds
`shifu_input_0` [1, N] float and fetches `shifu_output_0`
(reference: shifu-tensorflow-eval/.../TensorflowModel.java:111-172, load at
:169; export contract from ssgd_monitor.py:457-473).  This module writes a
`saved_model.pb` that loader reads UNCHANGED, with no TensorFlow
dependency here: the protobuf wire format is emitted directly.

Design choice: the graph is FROZEN — weights are Const nodes, so the
SavedModel has no variables directory.  TF's loader (loader.cc RunRestore)
explicitly skips restore when `variables/variables.index` is absent ("The
specified SavedModel has no variables; no checkpoints were restored."),
which holds for every TF 1.x line including the reference's 1.4.0 Java
binding.  This sidesteps the TensorBundle (SSTable) checkpoint format
entirely and makes the artifact self-contained.

Graph: Placeholder `shifu_input_0` [?, N] float32 -> per layer
MatMul(Const W [in,out]) -> Add(Const b) -> activation -> 1-unit head ->
Sigmoid node NAMED `shifu_output_0`.  Ops used (Placeholder, Const, MatMul,
Add, Mul, Maximum, Sigmoid, Tanh, Relu) all exist in TF 1.4; leakyrelu is
composed as Maximum(x, alpha*x) because the fused LeakyRelu op postdates
1.4.

Verified by tests/test_saved_model.py: an independent pure-python wire
parser decodes the emitted file, re-executes the graph with numpy, and the
scores must match the training model's forward exactly.
"""
from __future__ import annotations

import os
import struct
from typing import Dict, List, Sequence

import numpy as np

# --------------------------------------------------------------- wire format
DT_FLOAT = 1


def _varint(n: int) -> bytes:
    out = bytearray()
    while True:
        b = n & 0x7F
        n >>= 7
        if n:
            out.append(b | 0x80)
        else:
            out.append(b)
            return bytes(out)


def _key(field: int, wire: int) -> bytes:
    return _varint((field << 3) | wire)


def _len_delim(field: int, payload: bytes) -> bytes:
    return _key(field, 2) + _varint(len(payload)) + payload


def _vint_field(field: int, value: int) -> bytes:
    return _key(field, 0) + _varint(value)


def _float_field(field: int, value: float) -> bytes:
    return _key(field, 5) + struct.pack("<f", value)


def _string(field: int, s: str) -> bytes:
    return _len_delim(field, s.encode("utf-8"))


# ------------------------------------------------------------- proto pieces
def _tensor_shape(dims: Sequence[int]) -> bytes:
    # TensorShapeProto { repeated Dim dim = 2; }  Dim { int64 size = 1; }
    out = b""
    for d in dims:
        size = d & ((1 << 64) - 1) if d < 0 else d  # -1 (unknown) as uint64
        out += _len_delim(2, _key(1, 0) + _varint(size))
    return out


def _tensor_proto(arr: np.ndarray) -> bytes:
    # TensorProto { dtype=1; tensor_shape=2; tensor_content=4 }
    a = np.ascontiguousarray(arr, dtype=np.float32)
    out = _vint_field(1, DT_FLOAT)
    out += _len_delim(2, _tensor_shape(a.shape))
    out += _len_delim(4, a.tobytes())
    return out


def _attr_type(dt: int) -> bytes:
    return _vint_field(6, dt)          # AttrValue.type = 6


def _attr_shape(dims: Sequence[int]) -> bytes:
    return _len_delim(7, _tensor_shape(dims))   # AttrValue.shape = 7


def _attr_tensor(arr: np.ndarray) -> bytes:
    return _len_delim(8, _tensor_proto(arr))    # AttrValue.tensor = 8


def _attr_bool(v: bool) -> bytes:
    return _key(5, 0) + _varint(1 if v else 0)  # AttrValue.b = 5


def _node(name: str, op: str, inputs: Sequence[str] = (),
          attrs: Dict[str, bytes] = {}) -> bytes:
    # NodeDef { name=1; op=2; input=3; attr=5 map<string, AttrValue> }
    out = _string(1, name) + _string(2, op)
    for i in inputs:
        out += _string(3, i)
    for k, v in attrs.items():
        entry = _string(1, k) + _len_delim(2, v)
        out += _len_delim(5, entry)
    return out


def _tensor_info(name: str, dims: Sequence[int]) -> bytes:
    # TensorInfo { name=1; dtype=2; tensor_shape=3 }
    return (_string(1, name) + _vint_field(2, DT_FLOAT)
            + _len_delim(3, _tensor_shape(dims)))


# ------------------------------------------------------------- graph builder
_ACT_SIMPLE = {"sigmoid": "Sigmoid", "tanh": "Tanh", "relu": "Relu"}


def _emit_layers(nodes: List[bytes], layers, x: str, num_dense: int) -> str:
    """MatMul/Add/act chain; returns the final (pre-named) logits node."""
    tf = {"T": _attr_type(DT_FLOAT)}
    for li, (w, b, act) in enumerate(layers):
        is_head = li == len(layers) - 1
        base = f"layer{li}"
        wname, bname = f"{base}/W", f"{base}/b"
        nodes.append(_node(wname, "Const", (),
                           {"dtype": _attr_type(DT_FLOAT),
                            "value": _attr_tensor(w)}))
        nodes.append(_node(bname, "Const", (),
                           {"dtype": _attr_type(DT_FLOAT),
                            "value": _attr_tensor(b)}))
        mm = f"{base}/MatMul"
        nodes.append(_node(mm, "MatMul", (x, wname),
                           {"T": _attr_type(DT_FLOAT),
                            "transpose_a": _attr_bool(False),
                            "transpose_b": _attr_bool(False)}))
        add = f"{base}/Add"
        nodes.append(_node(add, "Add", (mm, bname), dict(tf)))
        a = act.lower()
        if is_head:
            x = add
        elif a in _ACT_SIMPLE:
            nm = f"{base}/{_ACT_SIMPLE[a]}"
            nodes.append(_node(nm, _ACT_SIMPLE[a], (add,), dict(tf)))
            x = nm
        elif a in ("leakyrelu", "leaky_relu"):
            alpha = f"{base}/alpha"
            nodes.append(_node(alpha, "Const", (),
                               {"dtype": _attr_type(DT_FLOAT),
                                "value": _attr_tensor(np.float32(0.01))}))
            mul = f"{base}/mul"
            nodes.append(_node(mul, "Mul", (add, alpha), dict(tf)))
            nm = f"{base}/Maximum"
            nodes.append(_node(nm, "Maximum", (add, mul), dict(tf)))
            x = nm
        else:   # "none"/linear
            x = add
    return x


def emit_saved_model(final_model_path: str, layers, num_dense: int,
                     input_name: str = "shifu_input_0",
                     output_name: str = "shifu_output_0",
                     tags: Sequence[str] = ("serve",)) -> str:
    """Write `<final_model_path>/saved_model.pb`.

    layers: [(W [in,out] f32, b [out] f32, activation-name), ...]; the last
    entry is the 1-unit head (its activation arg is ignored — the exported
    head is always the named Sigmoid, like the reference's
    `shifu_output_0` sigmoid unit, ssgd_monitor.py:121)."""
    nodes: List[bytes] = []
    nodes.append(_node(input_name, "Placeholder", (),
                       {"dtype": _attr_type(DT_FLOAT),
                        "shape": _attr_shape((-1, num_dense))}))
    logits = _emit_layers(nodes, layers, input_name, num_dense)
    nodes.append(_node(output_name, "Sigmoid", (logits,),
                       {"T": _attr_type(DT_FLOAT)}))

    graph_def = b"".join(_len_delim(1, n) for n in nodes)
    # VersionDef { producer=1 }: 24 = TF 1.4's GraphDef version
    graph_def += _len_delim(4, _vint_field(1, 24))

    # MetaInfoDef { tags=4; tensorflow_version=5 }
    meta_info = b"".join(_string(4, t) for t in tags)
    meta_info += _string(5, "1.4.0")

    # SignatureDef serving_default (the Java path feeds tensor names from
    # GenericModelConfig directly, but tools that read signatures get one)
    sig = (_len_delim(1, _string(1, "inputs")
                      + _len_delim(2, _tensor_info(f"{input_name}:0",
                                                   (-1, num_dense))))
           + _len_delim(2, _string(1, "outputs")
                        + _len_delim(2, _tensor_info(f"{output_name}:0",
                                                     (-1, 1))))
           + _string(3, "tensorflow/serving/predict"))
    sig_entry = _string(1, "serving_default") + _len_delim(2, sig)

    meta_graph = (_len_delim(1, meta_info)
                  + _len_delim(2, graph_def)
                  + _len_delim(5, sig_entry))
    saved_model = _vint_field(1, 1) + _len_delim(2, meta_graph)

    os.makedirs(final_model_path, exist_ok=True)
    out = os.path.join(final_model_path, "saved_model.pb")
    with open(out, "wb") as f:
        f.write(saved_model)
    return out


# --------------------------------------------------------------- reader side
# Independent decode + numpy re-execution of an emitted saved_model.pb.
# Used by tests (emitter verification) and by serve.ShifuScorer, which can
# score straight from the TF artifact — the same file the Java
# SavedModelBundle path loads.

def _read_varint(buf: bytes, i: int):
    shift, val = 0, 0
    while True:
        b = buf[i]
        i += 1
        val |= (b & 0x7F) << shift
        if not (b & 0x80):
            return val, i
        shift += 7


def parse_message(buf: bytes) -> Dict:
    """Generic wire decode -> {field: [(wire_type, value), ...]}."""
    out: Dict = {}
    i = 0
    while i < len(buf):
        tag, i = _read_varint(buf, i)
        field, wire = tag >> 3, tag & 7
        if wire == 0:
            v, i = _read_varint(buf, i)
        elif wire == 2:
            ln, i = _read_varint(buf, i)
            v = buf[i:i + ln]
            i += ln
        elif wire == 5:
            v = struct.unpack("<I", buf[i:i + 4])[0]
            i += 4
        elif wire == 1:
            v = struct.unpack("<Q", buf[i:i + 8])[0]
            i += 8
        else:
            raise ValueError(f"wire type {wire}")
        out.setdefault(field, []).append((wire, v))
    return out


def _parse_shape(buf: bytes) -> List[int]:
    dims = []
    for _, d in parse_message(buf).get(2, []):
        size = parse_message(d)[1][0][1]
        if size >= 1 << 63:
            size -= 1 << 64
        dims.append(size)
    return dims


def _parse_tensor(buf: bytes) -> np.ndarray:
    m = parse_message(buf)
    if m[1][0][1] != DT_FLOAT:
        raise ValueError("only DT_FLOAT tensors supported")
    dims = _parse_shape(m[2][0][1]) if 2 in m else []
    return np.frombuffer(m[4][0][1], dtype=np.float32).reshape(dims)


def load_saved_model(path: str):
    """saved_model.pb (file or directory) -> (nodes, tags).

    nodes: {name: (op, [inputs], {attr: parsed AttrValue})}."""
    if os.path.isdir(path):
        path = os.path.join(path, "saved_model.pb")
    with open(path, "rb") as f:
        blob = f.read()
    sm = parse_message(blob)
    if sm[1][0][1] != 1:
        raise ValueError("unsupported saved_model_schema_version")
    mg = parse_message(sm[2][0][1])
    meta_info = parse_message(mg[1][0][1])
    tags = [v.decode() for _, v in meta_info.get(4, [])]
    graph = parse_message(mg[2][0][1])
    nodes = {}
    for _, nb in graph[1]:
        n = parse_message(nb)
        name = n[1][0][1].decode()
        op = n[2][0][1].decode()
        inputs = [v.decode() for _, v in n.get(3, [])]
        attrs = {}
        for _, ab in n.get(5, []):
            e = parse_message(ab)
            attrs[e[1][0][1].decode()] = parse_message(e[2][0][1])
        nodes[name] = (op, inputs, attrs)
    return nodes, tags


def run_saved_model(nodes: Dict, feeds: Dict[str, np.ndarray],
                    fetch: str = "shifu_output_0") -> np.ndarray:
    """Execute the frozen graph with numpy (the op set the emitter uses)."""
    vals = dict(feeds)

    def ev(name):
        if name in vals:
            return vals[name]
        op, inputs, attrs = nodes[name]
        if op == "Const":
            v = _parse_tensor(attrs["value"][8][0][1])
        elif op == "MatMul":
            v = ev(inputs[0]) @ ev(inputs[1])
        elif op == "Add":
            v = ev(inputs[0]) + ev(inputs[1])
        elif op == "Mul":
            v = ev(inputs[0]) * ev(inputs[1])
        elif op == "Maximum":
            v = np.maximum(ev(inputs[0]), ev(inputs[1]))
        elif op == "Sigmoid":
            v = 1.0 / (1.0 + np.exp(-ev(inputs[0])))
        elif op == "Tanh":
            v = np.tanh(ev(inputs[0]))
        elif op == "Relu":
            v = np.maximum(ev(inputs[0]), 0.0)
        else:
            raise ValueError(f"unsupported op {op}")
        vals[name] = v
        return v

    return ev(fetch)


def layers_from_mlp(model) -> List:
    """(W [in,out], b, act) list from a ShifuMLP (weights stored [out,in])."""
    layers = []
    for lin in list(model.hidden) + [model.shifu_output_0]:
        w = lin.weight.detach().float().cpu().numpy().T.copy()
        b = lin.bias.detach().float().cpu().numpy()
        layers.append((w, b, lin.activation))
    return layers
