"""Config system: one frozen dataclass, YAML preset + CLI overrides.

(Reference class keeps constants in source; here every BASELINE config
is expressible as a YAML preset under conf/presets/.)
"""
from __future__ import annotations

import argparse
import dataclasses
from dataclasses import dataclass
from typing import Optional

import yaml


@dataclass(frozen=True)
class Config:
    # model
    model_path: str = ""                 # frozen .pb or SavedModel dir ('' = built-in)
    model_name: str = "lenet"            # built-in builder when model_path == ''
    input_name: str = "input"
    output_name: str = "softmax"
    input_height: int = 28
    input_width: int = 28
    input_channels: int = 1
    num_classes: int = 10
    normalize: str = "unit"              # 'unit' | 'inception' | 'raw'
    # engine
    device: str = "auto"                 # 'auto' | 'cpu' | 'cuda'
    dtype: str = "bf16"                  # gpu compute dtype
    use_hip_graphs: bool = True
    batch_sizes: tuple = (1, 8, 32, 64)  # pre-compiled/captured batch sizes
    # serving
    host: str = "0.0.0.0"
    port: int = 5000
    top_k: int = 5
    labels_path: Optional[str] = None
    # batcher
    max_batch: int = 64
    max_wait_ms: float = 2.0
    # data parallel
    gpus: int = 1

    def replace(self, **kw) -> "Config":
        return dataclasses.replace(self, **kw)


def load_config(yaml_path: Optional[str] = None, **overrides) -> Config:
    data = {}
    if yaml_path:
        with open(yaml_path) as fh:
            data = yaml.safe_load(fh) or {}
    data.update({k: v for k, v in overrides.items() if v is not None})
    if "batch_sizes" in data and isinstance(data["batch_sizes"], list):
        data["batch_sizes"] = tuple(data["batch_sizes"])
    return Config(**data)


def add_config_args(p: argparse.ArgumentParser) -> None:
    p.add_argument("--config", default=None, help="YAML preset path")
    p.add_argument("--model", dest="model_path", default=None)
    p.add_argument("--model-name", dest="model_name", default=None)
    p.add_argument("--port", type=int, default=None)
    p.add_argument("--host", default=None)
    p.add_argument("--gpus", type=int, default=None)
    p.add_argument("--device", default=None)
    p.add_argument("--max-batch", dest="max_batch", type=int, default=None)
    p.add_argument("--max-wait-ms", dest="max_wait_ms", type=float, default=None)
    p.add_argument("--labels", dest="labels_path", default=None)


def config_from_args(args: argparse.Namespace) -> Config:
    keys = [f.name for f in dataclasses.fields(Config)]
    overrides = {k: getattr(args, k) for k in keys if hasattr(args, k)}
    return load_config(getattr(args, "config", None), **overrides)
