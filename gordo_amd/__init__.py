"""gordo_amd — MI355X-native framework for building, deploying and serving
fleets of per-asset time-series anomaly-detection models.

Capability parity target: equinor/gordo-components (v0.x line). The reference
is a Keras/TensorFlow + Kubernetes fleet framework; this is a from-scratch
redesign for a single 8×MI355X node: PyTorch-ROCm estimators backed by
hand-written CDNA4 HIP kernels, RCCL-over-xGMI data-parallel training, and
hipGraph-captured batched inference.

Reference layer map: see SURVEY.md §1 (reference: gordo_components/__init__.py).
"""

__version__ = "0.1.0"

MAJOR_VERSION, MINOR_VERSION = (int(x) for x in __version__.split(".")[:2])
