"""fedtorch_amd — an MI355X-native federated / local-SGD training engine.

A from-scratch re-design of the capabilities of MLOPTPSU/FedTorch
(reference surveyed in SURVEY.md) for AMD Instinct MI355X (gfx950):

* PyTorch-ROCm autograd/models, one process per GPU over RCCL/xGMI.
* All per-parameter Python loops of the reference (aggregation, SGD step,
  quantize/compress — reference `fedtorch/comms/...`) are replaced by a flat
  contiguous parameter **arena** per model replica (`fedtorch_amd/parallel/arena.py`)
  so every hot path is ONE hand-written CDNA4 HIP kernel launch and ONE RCCL
  collective per model, instead of P small launches + P small messages.
* The reference's star topology (gather -> sum at rank 0 -> broadcast,
  `comms/algorithms/federated/fedavg.py:42-78`) is implemented as weighted
  all-reduce over cached communicators — mathematically identical (weights are
  pre-scaled per rank, offline ranks contribute zero) and xGMI-friendly.

Public API parity with the reference (README.md:44-78 of the reference):
`get_args()`, `Client`, `train_and_validate_federated*`, plus the centered
(single-process simulation) mode.
"""

__version__ = "0.1.0"

from fedtorch_amd.parameters import get_args  # noqa: F401


def __getattr__(name):
    # lazy: `from fedtorch_amd import Client` without importing torch at
    # package-import time
    if name in ('Client', 'ClientCentered', 'ServerCentered'):
        import fedtorch_amd.nodes as nodes
        return getattr(nodes, name)
    raise AttributeError(name)
