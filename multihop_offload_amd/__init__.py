"""multihop_offload_amd — MI355X-native GNN task-offloading framework.

A from-scratch rebuild of the capabilities of ``zhongyuanzhao/multihop-offload``
(ICASSP'24 congestion-aware task offloading in wireless multi-hop networks),
designed MI355X-first:

* the queueing-network simulator (``env``/``engine``) is vectorised over a
  batch of graphs and runs device-resident (HIP kernels, one workgroup per
  graph, LDS-staged state) — reference analog: ``src/offloading_v3.py``;
* the ChebConv actor + differentiable queueing critic (``agent``/``models``)
  — reference analog: ``src/gnn_offloading_agent.py``;
* data-parallel training over RCCL/xGMI (``parallel.dp``);
* compatible ``AdHoc_train``/``AdHoc_test`` entry points, CSV schema and
  ``model_ChebConv_*`` checkpoint layout (``harness``, ``utils.checkpoint``).
"""

__version__ = "0.1.0"

from .graphs import CaseGraph, JobInstance  # noqa: F401
from .env import AdhocCloudEnv, AdhocCloud  # noqa: F401
from .agent import ACOAgent                 # noqa: F401
from .engine import EpisodeEngine, JobBatch, EpisodeResult  # noqa: F401
