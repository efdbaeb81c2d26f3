"""Payload serialization for the data plane.

Two-tier policy (SURVEY §2c "pickle / cloudpickle" row):

* host metadata (task tuples, plain Python args) rides pickle — cloudpickle
  when the closure was defined in an interactive shell;
* **CUDA (ROCm) tensors never ride by value**: they are reduced to HIP IPC
  handles via ``torch.multiprocessing.reductions.reduce_tensor`` so a tensor
  task-arg/result stays device-resident and crosses the process boundary as
  a ~100-byte handle.  Requires ``HSA_ENABLE_IPC_MODE_LEGACY=0`` (dmabuf
  IPC) which the environment exports.

CPU tensors ride by value (safe everywhere; the single-node data plane is a
shared-memory ring, so a by-value copy is one memcpy).
"""

import io
import pickle

try:
    import cloudpickle
except ImportError:  # pragma: no cover - cloudpickle is in the wheelhouse
    cloudpickle = None

_torch = None


def _lazy_torch():
    global _torch
    if _torch is None:
        import torch

        _torch = torch
    return _torch


class _Pickler(pickle.Pickler):
    """Pickler that ships CUDA tensors as HIP IPC handles."""

    def reducer_override(self, obj):
        torch = _lazy_torch() if type(obj).__module__.startswith("torch") else None
        if torch is not None and isinstance(obj, torch.Tensor) and obj.is_cuda:
            from torch.multiprocessing.reductions import reduce_tensor

            return reduce_tensor(obj)
        return NotImplemented


def dumps(obj, interactive=False):
    if interactive:
        if cloudpickle is None:
            raise RuntimeError("cloudpickle required for interactive closures")
        return cloudpickle.dumps(obj)
    buf = io.BytesIO()
    _Pickler(buf, protocol=pickle.HIGHEST_PROTOCOL).dump(obj)
    return buf.getvalue()


def loads(data):
    return pickle.loads(data)


def dumps_closure(obj):
    """Serialize possibly-interactive callables (cloudpickle fallback)."""
    try:
        return dumps(obj)
    except (pickle.PicklingError, AttributeError, TypeError):
        if cloudpickle is None:
            raise
        return cloudpickle.dumps(obj)
