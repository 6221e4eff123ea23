"""SavedModel loader — `saved_model.pb` + `variables/` TensorBundle,
no TF dependency (reference behavior: `tf.saved_model.loader.load`).

Proto subset (field numbers are the public TF ABI):
  SavedModel    {1: saved_model_schema_version, 2: repeated MetaGraphDef}
  MetaGraphDef  {1: MetaInfoDef, 2: GraphDef graph_def,
                 5: map<string, SignatureDef> signature_def}
  SignatureDef  {1: map inputs TensorInfo, 2: map outputs, 3: method_name}
  TensorInfo    {1: name, 2: dtype, 3: TensorShapeProto}

Variable resolution: VariableV2 / VarHandleOp nodes are materialised as
Const tensors from the bundle (key = node name); ReadVariableOp and
Assign/Identity plumbing collapses onto them.
"""
from __future__ import annotations

import os
from typing import Dict, List, Optional, Tuple

from . import graphdef as gd
from . import wire
from .tensor_bundle import BundleReader


class SignatureDef:
    def __init__(self):
        self.inputs: Dict[str, str] = {}    # arg name -> tensor name
        self.outputs: Dict[str, str] = {}
        self.method_name = ""


def _parse_tensorinfo(buf: bytes) -> str:
    name = ""
    for f, wt, v in wire.iter_fields(buf):
        if f == 1 and wt == 2:
            name = v.decode("utf-8")
    return name


def _parse_signature(buf: bytes) -> SignatureDef:
    sig = SignatureDef()
    for f, wt, v in wire.iter_fields(buf):
        if f in (1, 2) and wt == 2:
            key, ti = None, ""
            for f2, wt2, v2 in wire.iter_fields(v):
                if f2 == 1 and wt2 == 2:
                    key = v2.decode("utf-8")
                elif f2 == 2 and wt2 == 2:
                    ti = _parse_tensorinfo(v2)
            if key:
                (sig.inputs if f == 1 else sig.outputs)[key] = ti
        elif f == 3 and wt == 2:
            sig.method_name = v.decode("utf-8")
    return sig


def parse_savedmodel_pb(data: bytes):
    """-> (GraphDef, {sig_name: SignatureDef}) of the first meta graph."""
    graph: Optional[gd.GraphDef] = None
    sigs: Dict[str, SignatureDef] = {}
    for f, wt, v in wire.iter_fields(data):
        if f == 2 and wt == 2 and graph is None:  # first MetaGraphDef
            for f2, wt2, v2 in wire.iter_fields(v):
                if f2 == 2 and wt2 == 2:
                    graph = gd.parse_graphdef(v2)
                elif f2 == 5 and wt2 == 2:
                    key, sig = None, None
                    for f3, wt3, v3 in wire.iter_fields(v2):
                        if f3 == 1 and wt3 == 2:
                            key = v3.decode("utf-8")
                        elif f3 == 2 and wt3 == 2:
                            sig = _parse_signature(v3)
                    if key and sig:
                        sigs[key] = sig
    if graph is None:
        raise ValueError("saved_model.pb has no MetaGraphDef/graph_def")
    return graph, sigs


def _materialize_variables(g: gd.GraphDef, bundle: BundleReader) -> None:
    """Rewrite VariableV2/VarHandleOp nodes into Const nodes with the
    bundle values; collapse ReadVariableOp onto them.

    Key resolution, in order: (1) the node name itself (TF1 name-based
    checkpoints), (2) name + '/.ATTRIBUTES/VARIABLE_VALUE' (the common
    TF2 layout when object paths mirror variable names), (3) the
    checkpoint's TrackableObjectGraph (`ckpt/object_graph.py`) — TF2
    object-based checkpoints record full_name -> checkpoint_key there,
    which covers renamed/nested objects."""
    from . import object_graph as og
    var_nodes = {}
    for n in g.node:
        if n.op in ("VariableV2", "Variable", "VarHandleOp"):
            var_nodes[n.name] = n
    og_index = None
    og_loaded = False
    for name, n in var_nodes.items():
        key = name
        if key not in bundle.entries:
            alt = name + "/.ATTRIBUTES/VARIABLE_VALUE"
            if alt in bundle.entries:
                key = alt
            else:
                if not og_loaded:
                    og_index = og.load_index(bundle)
                    og_loaded = True
                resolved = (og_index.resolve(name)
                            if og_index is not None else None)
                if resolved is not None and resolved in bundle.entries:
                    key = resolved
                else:
                    raise KeyError(
                        f"variable {name} not in bundle "
                        f"(keys: {list(bundle.keys())[:5]}...)")
        arr = bundle.get(key)
        n.op = "Const"
        n.input = []
        n.attr = {"dtype": gd.AttrValue(type=gd._NUMPY_TO_DT[arr.dtype]),
                  "value": gd.AttrValue(tensor=arr)}
    for n in g.node:
        if n.op == "ReadVariableOp":
            n.op = "Identity"


def _encode_tensorinfo(name: str) -> bytes:
    return wire.field_string(1, name)


def _encode_signature(inputs: Dict[str, str], outputs: Dict[str, str]) -> bytes:
    out = b""
    for key, tname in inputs.items():
        entry = (wire.field_string(1, key)
                 + wire.field_bytes(2, _encode_tensorinfo(tname)))
        out += wire.field_bytes(1, entry)
    for key, tname in outputs.items():
        entry = (wire.field_string(1, key)
                 + wire.field_bytes(2, _encode_tensorinfo(tname)))
        out += wire.field_bytes(2, entry)
    out += wire.field_string(3, "tensorflow/serving/predict")
    return out


def save_savedmodel(g: gd.GraphDef, path: str, inputs: List[str],
                    outputs: List[str], as_variables: bool = True,
                    min_var_elems: int = 64) -> None:
    """Export a SavedModel directory (saved_model.pb + variables bundle).

    With as_variables=True, large Const nodes are rewritten to VariableV2
    + a TensorBundle — producing the *checkpointed* SavedModel layout the
    reference loads, so the reader path is tested against real structure.
    """
    import copy
    import numpy as np
    g = copy.deepcopy(g)
    tensors: Dict[str, "np.ndarray"] = {}
    if as_variables:
        for n in g.node:
            if n.op == "Const":
                arr = n.attr["value"].tensor
                if arr is not None and arr.size >= min_var_elems \
                        and arr.dtype != object:
                    tensors[n.name] = arr
                    n.op = "VariableV2"
                    n.attr = {
                        "dtype": n.attr["dtype"],
                        "shape": gd.AttrValue(shape=list(arr.shape)),
                    }
    meta = b""
    meta += wire.field_bytes(2, gd.encode_graphdef(g))
    sig = _encode_signature({"input": inputs[0] + ":0"},
                            {f"output_{i}": o + ":0"
                             for i, o in enumerate(outputs)})
    entry = (wire.field_string(1, "serving_default")
             + wire.field_bytes(2, sig))
    meta += wire.field_bytes(5, entry)
    sm = wire.field_varint(1, 1) + wire.field_bytes(2, meta)
    os.makedirs(path, exist_ok=True)
    with open(os.path.join(path, "saved_model.pb"), "wb") as fh:
        fh.write(sm)
    if tensors:
        from .tensor_bundle import write_bundle
        write_bundle(os.path.join(path, "variables", "variables"), tensors)


def load_savedmodel(path: str,
                    signature: str = "serving_default"):
    """-> (GraphDef with variables materialised, SignatureDef or None)"""
    pb = os.path.join(path, "saved_model.pb")
    with open(pb, "rb") as fh:
        g, sigs = parse_savedmodel_pb(fh.read())
    prefix = os.path.join(path, "variables", "variables")
    if os.path.exists(prefix + ".index"):
        bundle = BundleReader(prefix)
        _materialize_variables(g, bundle)
    sig = sigs.get(signature) or (next(iter(sigs.values())) if sigs else None)
    return g, sig
