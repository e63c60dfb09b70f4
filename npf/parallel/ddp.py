"""Data-parallel backend: one process per GPU, RCCL over xGMI.

The reference is strictly single-device (no torch.distributed
anywhere); this module is new capability designed for one MI355X node
(8 GPUs, fully-connected point-to-point xGMI, 7 links x ~153 GB/s per GPU):

- NPF models are <1 M params (<4 MB fp32 grads), so gradient reduction is
  LATENCY-bound, not bandwidth-bound.  We therefore keep every parameter's
  gradient as a VIEW into one flat contiguous buffer and issue a single
  all-reduce per step — one RCCL launch, no bucketing overhead, trivially
  capturable inside a hipGraph.
- Meta-batch semantics: each rank draws its own tasks (rank-offset seed), so
  the global batch is world_size x per-rank batch; all-reduce(AVG) of
  gradients is exactly the reference's single-GPU batch-mean loss
  (reference npf/losses.py:74-77 reduces with mean over tasks).
- backend "nccl" IS RCCL on ROCm; "gloo" is used for CPU tests.
"""

import os
from datetime import timedelta

import torch
import torch.distributed as dist

__all__ = [
    "init_distributed",
    "is_distributed",
    "get_rank",
    "get_world_size",
    "barrier",
    "FlatDDP",
    "all_gather_cat",
]


def init_distributed(backend=None, timeout_s=600):
    """Initialize the process group from torchrun env vars; no-op without them.

    Returns (rank, world_size, local_rank).
    """
    if "RANK" not in os.environ or int(os.environ.get("WORLD_SIZE", "1")) <= 1:
        return 0, 1, 0
    if not dist.is_initialized():
        if backend is None:
            backend = "nccl" if torch.cuda.is_available() else "gloo"
        dist.init_process_group(backend, timeout=timedelta(seconds=timeout_s))
    rank = dist.get_rank()
    world = dist.get_world_size()
    local_rank = int(os.environ.get("LOCAL_RANK", rank))
    if torch.cuda.is_available():
        # modulo: an N-rank smoke run may share fewer GPUs (e.g. 2 ranks on
        # one box to exercise RCCL init/collectives)
        local_rank = local_rank % torch.cuda.device_count()
        torch.cuda.set_device(local_rank)
    return rank, world, local_rank


def is_distributed():
    return dist.is_available() and dist.is_initialized() and dist.get_world_size() > 1


def get_rank():
    return dist.get_rank() if is_distributed() else 0


def get_world_size():
    return dist.get_world_size() if is_distributed() else 1


def barrier():
    if is_distributed():
        dist.barrier()


def all_gather_cat(t, dim=0):
    """All-gather variable-size tensors along `dim` (pads to max then trims).

    Used to reassemble rank-sharded per-task eval vectors in original order
    (rank-interleaved shards are inverse-permuted by the caller).
    """
    if not is_distributed():
        return t
    world = dist.get_world_size()
    n = torch.tensor([t.shape[dim]], device=t.device, dtype=torch.long)
    ns = [torch.zeros_like(n) for _ in range(world)]
    dist.all_gather(ns, n)
    ns = [int(x) for x in ns]
    n_max = max(ns)
    if t.shape[dim] < n_max:
        pad_shape = list(t.shape)
        pad_shape[dim] = n_max - t.shape[dim]
        t = torch.cat([t, t.new_zeros(pad_shape)], dim=dim)
    outs = [torch.empty_like(t) for _ in range(world)]
    dist.all_gather(outs, t.contiguous())
    return torch.cat([o.narrow(dim, 0, ns[i]) for i, o in enumerate(outs)], dim=dim)


class FlatDDP:
    """Flat-buffer data parallelism for small models.

    All parameter gradients live in ONE contiguous buffer; `reduce_()` is a
    single all-reduce(AVG).  Parameters are broadcast from rank 0 at
    construction so replicas start identical.
    """

    def __init__(self, module, grad_dtype=None):
        self.module = module
        self.params = [p for p in module.parameters() if p.requires_grad]
        numel = sum(p.numel() for p in self.params)
        device = self.params[0].device if self.params else torch.device("cpu")
        if grad_dtype is None:
            grad_dtype = self.params[0].dtype if self.params else torch.float32
        self.flat_grads = torch.zeros(numel, dtype=grad_dtype, device=device)
        # carve the buffer into per-param gradient views
        offset = 0
        for p in self.params:
            n = p.numel()
            p.grad = self.flat_grads.narrow(0, offset, n).view_as(p)
            offset += n
        self._world = get_world_size()
        if self._world > 1:
            self.broadcast_params_()

    def broadcast_params_(self):
        with torch.no_grad():
            for p in self.params:
                dist.broadcast(p.data, src=0)

    def zero_grad_(self):
        """Zero the flat buffer (never set_to_none: grads must stay views)."""
        self.flat_grads.zero_()

    def reduce_(self):
        """One all-reduce(mean) over the whole gradient buffer."""
        if self._world > 1:
            self.flat_grads.div_(self._world)
            dist.all_reduce(self.flat_grads, op=dist.ReduceOp.SUM)
        return self.flat_grads
