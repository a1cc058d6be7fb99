"""Process-group communication wrapper (RCCL over xGMI on GPU, gloo on CPU).

The reference's entire multi-GPU story is single-process
``torch.nn.DataParallel`` (reference main.py:184). Here: one process per
GPU, ``torch.distributed`` with the nccl backend (RCCL on ROCm), xGMI
point-to-point links underneath (SURVEY.md §5 "Distributed communication
backend").
"""

import datetime
import os
from typing import Optional

import torch
import torch.distributed as dist


def env_rank() -> int:
    return int(os.environ.get('RANK', '0'))


def env_world_size() -> int:
    return int(os.environ.get('WORLD_SIZE', '1'))


def env_local_rank() -> int:
    return int(os.environ.get('LOCAL_RANK', os.environ.get('RANK', '0')))


class Comm:
    """Thin process-group handle. world_size==1 -> all ops are no-ops."""

    def __init__(self, backend: Optional[str] = None, device: Optional[torch.device] = None,
                 timeout_s: int = 600):
        self.world_size = env_world_size()
        self.rank = env_rank()
        self.local_rank = env_local_rank()
        if device is not None:
            self.device = device
        elif torch.cuda.is_available():
            self.device = torch.device('cuda', self.local_rank % torch.cuda.device_count())
        else:
            self.device = torch.device('cpu')

        self.backend = backend
        if self.world_size > 1 and not dist.is_initialized():
            if backend is None:
                # register gloo alongside RCCL so the occasional CPU-tensor
                # collective (checkpoint metadata, host-side merges) is
                # routed to gloo instead of crashing the nccl group
                backend = ('cpu:gloo,cuda:nccl' if self.device.type == 'cuda'
                           else 'gloo')
            self.backend = backend
            if self.device.type == 'cuda':
                torch.cuda.set_device(self.device)
            dist.init_process_group(backend=backend,
                                    timeout=datetime.timedelta(seconds=timeout_s))
        elif dist.is_initialized():
            self.backend = dist.get_backend()

    @property
    def is_distributed(self) -> bool:
        return self.world_size > 1

    def all_reduce_sum(self, t: torch.Tensor) -> torch.Tensor:
        if not self.is_distributed:
            return t
        dist.all_reduce(t, op=dist.ReduceOp.SUM)
        return t

    def all_reduce_sum_async(self, t: torch.Tensor):
        if not self.is_distributed:
            return None
        return dist.all_reduce(t, op=dist.ReduceOp.SUM, async_op=True)

    def all_gather_fixed(self, t: torch.Tensor) -> torch.Tensor:
        """All-gather equal-shaped tensors -> [world, *t.shape] (no host sync)."""
        if not self.is_distributed:
            return t.unsqueeze(0)
        t = t.contiguous()
        if t.dim() == 0:
            t = t.unsqueeze(0)
        out = t.new_empty((self.world_size * t.shape[0],) + tuple(t.shape[1:]))
        dist.all_gather_into_tensor(out, t)
        return out.view((self.world_size,) + tuple(t.shape))

    def all_gather_varlen(self, t: torch.Tensor) -> torch.Tensor:
        """All-gather 1-D tensors of differing lengths (host-syncing; used
        only off the hot path, e.g. OoD eval)."""
        if not self.is_distributed:
            return t
        n = torch.tensor([t.numel()], device=self.device, dtype=torch.int64)
        ns = self.all_gather_fixed(n).flatten().tolist()
        # mx >= 1: a zero-size all_gather_into_tensor is backend-dependent
        # (all ranks empty happens e.g. when no push candidates exist yet)
        mx = max(max(ns), 1)
        buf = t.new_zeros(mx)
        buf[:t.numel()] = t
        out = self.all_gather_fixed(buf)
        return torch.cat([out[r, :ns[r]] for r in range(self.world_size)])

    def broadcast(self, t: torch.Tensor, src: int = 0) -> torch.Tensor:
        if self.is_distributed:
            dist.broadcast(t, src=src)
        return t

    def barrier(self):
        if self.is_distributed:
            dist.barrier()

    def broadcast_module(self, module: torch.nn.Module, src: int = 0):
        """Make parameters+buffers bit-identical across ranks at startup.

        Iterates named_parameters/named_buffers (the LIVE tensors, including
        non-persistent buffers) — state_dict would hand us the memory bank's
        exported logical COPIES, which a broadcast would not write through.
        """
        if not self.is_distributed:
            return
        with torch.no_grad():
            for _, p in list(module.named_parameters()) \
                    + list(module.named_buffers()):
                if torch.is_tensor(p) and p.numel() > 0:
                    self.broadcast(p.data if hasattr(p, 'data') else p, src=src)

    def broadcast_buffers(self, module: torch.nn.Module, src: int = 0):
        """Re-sync buffers only (BatchNorm running stats drift with each
        rank's local batches; gradients use batch stats so training math is
        unaffected, but eval/push read the running stats — rank 0's win,
        DDP's broadcast_buffers convention). Called at epoch end. Buffers
        that are already rank-identical (memory bank) ride along --
        idempotent, and one extra epoch-rate collective is noise."""
        if not self.is_distributed:
            return
        with torch.no_grad():
            for _, b in module.named_buffers():
                if torch.is_tensor(b) and b.numel() > 0 and b.is_floating_point():
                    self.broadcast(b, src=src)
