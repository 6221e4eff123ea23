"""TF2 TrackableObjectGraph: checkpoint-key resolution for object-based
SavedModels.

TF2 checkpoints key variables by OBJECT PATH
(`model/dense/kernel/.ATTRIBUTES/VARIABLE_VALUE`), not by the graph
node name (`dense/kernel`). The mapping lives in the bundle itself
under the reserved key `_CHECKPOINTABLE_OBJECT_GRAPH`: a serialized
TrackableObjectGraph proto (tensorflow/core/protobuf/
trackable_object_graph.proto):

  TrackableObjectGraph { repeated TrackableObject nodes = 1; }
  TrackableObject {
    repeated ObjectReference children = 1;     # node_id=1, local_name=2
    repeated SerializedTensor attributes = 2;  # name=1, full_name=2,
                                               # checkpoint_key=3
    ...
  }

This module parses that proto (hand-written wire decode, ckpt/wire.py)
and builds two lookups used by savedmodel._materialize_variables:
  * full_name  -> checkpoint_key   (full_name is the v1-style variable
    name TF records alongside, e.g. "dense/kernel")
  * object path -> checkpoint_key  (path = '/'-joined local_names from
    the root node)
"""
from __future__ import annotations

from typing import Dict, List, Optional, Tuple

from . import wire

OBJECT_GRAPH_KEY = "_CHECKPOINTABLE_OBJECT_GRAPH"


class SerializedTensor:
    def __init__(self):
        self.name = ""
        self.full_name = ""
        self.checkpoint_key = ""


class TrackableObject:
    def __init__(self):
        self.children: List[Tuple[int, str]] = []  # (node_id, local_name)
        self.attributes: List[SerializedTensor] = []


def parse_object_graph(buf: bytes) -> List[TrackableObject]:
    nodes: List[TrackableObject] = []
    for f, wt, v in wire.iter_fields(buf):
        if f == 1 and wt == 2:
            node = TrackableObject()
            for nf, nwt, nv in wire.iter_fields(v):
                if nf == 1 and nwt == 2:  # ObjectReference
                    nid, lname = 0, ""
                    for cf, cwt, cv in wire.iter_fields(nv):
                        if cf == 1 and cwt == 0:
                            nid = cv
                        elif cf == 2 and cwt == 2:
                            lname = cv.decode("utf-8")
                    node.children.append((nid, lname))
                elif nf == 2 and nwt == 2:  # SerializedTensor
                    st = SerializedTensor()
                    for af, awt, av in wire.iter_fields(nv):
                        if af == 1 and awt == 2:
                            st.name = av.decode("utf-8")
                        elif af == 2 and awt == 2:
                            st.full_name = av.decode("utf-8")
                        elif af == 3 and awt == 2:
                            st.checkpoint_key = av.decode("utf-8")
                    node.attributes.append(st)
            nodes.append(node)
    return nodes


def encode_object_graph(nodes: List[TrackableObject]) -> bytes:
    """Inverse of parse_object_graph (fixtures/tests)."""
    out = b""
    for node in nodes:
        body = b""
        for nid, lname in node.children:
            ref = wire.field_varint(1, nid) + wire.field_string(2, lname)
            body += wire.field_bytes(1, ref)
        for st in node.attributes:
            enc = wire.field_string(1, st.name)
            if st.full_name:
                enc += wire.field_string(2, st.full_name)
            enc += wire.field_string(3, st.checkpoint_key)
            body += wire.field_bytes(2, enc)
        out += wire.field_bytes(1, body)
    return out


class ObjectGraphIndex:
    """checkpoint-key lookups for variable materialization."""

    def __init__(self, nodes: List[TrackableObject]):
        self.by_full_name: Dict[str, str] = {}
        self.by_path: Dict[str, str] = {}
        # BFS from node 0 building object paths; cycles guarded (TF
        # object graphs can alias nodes — first path wins, like TF)
        seen = set()
        queue: List[Tuple[int, str]] = [(0, "")] if nodes else []
        while queue:
            nid, path = queue.pop(0)
            if nid in seen or nid >= len(nodes):
                continue
            seen.add(nid)
            node = nodes[nid]
            for st in node.attributes:
                key = st.checkpoint_key
                if st.full_name:
                    self.by_full_name.setdefault(st.full_name, key)
                    # graph node names drop the ":0" suffix
                    self.by_full_name.setdefault(
                        st.full_name.split(":")[0], key)
                p = f"{path}/{st.name}" if path else st.name
                self.by_path.setdefault(p, key)
            for cid, lname in node.children:
                cpath = f"{path}/{lname}" if path else lname
                queue.append((cid, cpath))

    def resolve(self, var_name: str) -> Optional[str]:
        """Best-effort checkpoint key for a graph variable name."""
        base = var_name.split(":")[0]
        for cand in (var_name, base):
            if cand in self.by_full_name:
                return self.by_full_name[cand]
        # object-path forms: "a/b/c" variable often maps to path
        # "a/b/c/.ATTRIBUTES/VARIABLE_VALUE"
        for cand in (base, base.replace("/", "/") + "/VARIABLE_VALUE"):
            if cand in self.by_path:
                return self.by_path[cand]
        # suffix match: unique object path ending with the node name
        hits = [k for p, k in self.by_path.items()
                if p == base or p.endswith("/" + base)]
        if len(set(hits)) == 1:
            return hits[0]
        return None


def load_index(bundle) -> Optional[ObjectGraphIndex]:
    """Build the index from a BundleReader, if the checkpoint is
    object-based (TF2)."""
    e = bundle.entries.get(OBJECT_GRAPH_KEY)
    if e is None:
        return None
    raw = bundle._shard(e.shard_id)[e.offset:e.offset + e.size]
    # stored as a DT_STRING scalar: one varint-framed string element
    # (TF serializes the proto INTO a string tensor). A string tensor's
    # payload here is length-delimited: varint len + bytes.
    try:
        ln, pos = wire.read_varint(raw, 0)
        if pos + ln == len(raw):
            raw = raw[pos:pos + ln]
    except Exception:
        pass  # some writers store the proto bytes directly
    return ObjectGraphIndex(parse_object_graph(raw))
