"""tensorflow_web_deploy_amd — MI355X-native image-model serving framework.

A from-scratch rebuild of the capability surface of
`hetaoaoao/tensorflow_web_deploy` (TF frozen-graph/SavedModel loading +
Flask image-classification serving), engineered MI355X-first:
PyTorch-ROCm execution layer, hand-written CDNA4 HIP kernels for the hot
ops, RCCL-over-xGMI data-parallel replicas, dynamic batching.
No TensorFlow dependency anywhere.
"""
__version__ = "0.1.0"
