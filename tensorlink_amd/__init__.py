"""tensorlink_amd — an MI355X-native model-sharding and serving framework.

A from-scratch re-design of the capabilities of tensorlink-lab/tensorlink
(reference: /root/reference) for a single 8×MI355X node:

- the reference's peer-to-peer worker network (tensorlink/p2p/, tensorlink/nodes/)
  collapses onto one process per GPU rank with ``torch.distributed`` over
  RCCL/xGMI;
- its auto-sharder (``tensorlink/ml/graphing.py``) becomes
  :mod:`tensorlink_amd.parallel.planner`, sizing pipeline stages for
  288 GB HBM3E per GPU;
- its per-stage HuggingFace eager compute (``tensorlink/ml/worker.py``) becomes
  a native model zoo (:mod:`tensorlink_amd.models`) running hand-written
  CDNA4 HIP kernels (:mod:`tensorlink_amd.ops`);
- its socket/shared-memory transport (``tensorlink/p2p/connection.py``,
  ``tensorlink/nodes/shared_memory.py``) becomes RCCL point-to-point
  send/recv of device-resident activations (:mod:`tensorlink_amd.parallel.comm`);
- its user API (``DistributedModel``, ``create_optimizer`` —
  ``tensorlink/ml/module.py``, ``tensorlink/ml/optim.py``) and REST API
  (``tensorlink/api/node.py``) keep their shapes in
  :mod:`tensorlink_amd.module` and :mod:`tensorlink_amd.api`.
"""

__version__ = "0.1.0"

from tensorlink_amd.models.configs import ModelConfig, PRESETS  # noqa: F401
from tensorlink_amd.parallel.planner import ModelParser, StagePlan  # noqa: F401


def _lazy(name):
    if name == "DistributedModel":
        from tensorlink_amd.module import DistributedModel
        return DistributedModel
    if name == "InferenceEngine":
        from tensorlink_amd.engine.engine import InferenceEngine
        return InferenceEngine
    raise AttributeError(name)


def __getattr__(name):  # PEP 562 lazy imports keep `import tensorlink_amd` cheap
    return _lazy(name)
