"""pertgnn — MI355X-native PERT-GNN training framework.

A from-scratch AMD-native implementation of the capabilities of the KDD'23
PERT-GNN reference (handasontam/PERT-GNN-KDD23): end-to-end latency prediction
for microservice applications over Alibaba-2021-style call-graph traces.

Layers (see SURVEY.md for the blueprint):
  - ``pertgnn.data``     : offline ingest (reference preprocess.py parity),
                           synthetic trace generation, dataset assembly,
                           native batch collation (CSR + batch-ptr).
  - ``pertgnn.ops``      : op library. Pure-PyTorch reference (oracle) path and
                           hand-written CDNA4 HIP kernels (gfx950) behind
                           autograd.Function wrappers.
  - ``pertgnn.models``   : SAGEDeterministic graph-transformer (state_dict
                           compatible with the reference model.py).
  - ``pertgnn.parallel`` : Comm abstraction + bucketed DDP gradient engine over
                           RCCL/xGMI (torch.distributed backend "nccl").
  - ``pertgnn.train``    : training loop, metrics, checkpoint/resume.
"""

__version__ = "0.1.0"
