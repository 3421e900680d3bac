"""Flat-parameter DDP + fused optimizer — the MI355X data-parallel engine.

Role parity: the reference wraps torch DDP (python/ray/train/torch/
train_loop_utils.py:458) whose C++ reducer packs gradients into 25 MiB
buckets with copy kernels. MI355X-first design instead:

  * ALL parameters live as views into ONE flat bf16 buffer; gradients
    accumulate directly into views of ONE flat bf16 grad buffer — there are
    no pack/unpack copies at all (autograd writes land in the communication
    buffer).
  * Buckets are contiguous slices of the flat grad buffer, ordered by
    backward completion (reverse parameter order). When the last grad of a
    bucket lands (post-accumulate-grad hook), an async RCCL all-reduce is
    launched on that slice — overlapping with the rest of backward. On xGMI
    a ring all-reduce is per-link bound (7 x ~153 GB/s point-to-point links),
    so the default bucket is sized large (64 MiB) to amortize launch latency
    while still giving several buckets of overlap; tune with bucket_mb.
  * The optimizer is ONE fused HIP kernel pass over the flat buffers
    (csrc/kernels/adamw.hip): bf16 grad -> fp32 master update -> bf16 param
    write-back, with the 1/world_size averaging folded into grad_scale.
"""
from __future__ import annotations

import logging
from typing import Dict, List, Optional

import torch
import torch.distributed as dist
import torch.nn as nn

from ant_ray_amd import ops

logger = logging.getLogger("antray.parallel")

ALIGN_ELEMS = 16  # bucket/param alignment (kernels vectorize by 8; pad to 16)


def _aligned(n: int) -> int:
    return (n + ALIGN_ELEMS - 1) // ALIGN_ELEMS * ALIGN_ELEMS


class FlatParamManager:
    """Re-homes a module's parameters and gradients into flat buffers
    (bf16 on GPU; pass dtype=torch.float32 for CPU/gloo paths)."""

    def __init__(self, module: nn.Module, device=None, dtype=torch.bfloat16):
        self.module = module
        params = [p for p in module.parameters() if p.requires_grad]
        # reverse registration order approximates backward completion order,
        # so bucket[0] is ready first during backward
        params = params[::-1]
        self.params: List[nn.Parameter] = params
        self.offsets: Dict[int, int] = {}
        total = 0
        for p in params:
            self.offsets[id(p)] = total
            total += _aligned(p.numel())
        self.numel = _aligned(total)
        dev = device or (params[0].device if params else "cpu")
        self.dtype = dtype
        self.flat_param = torch.zeros(self.numel, dtype=dtype, device=dev)
        self.flat_grad = torch.zeros(self.numel, dtype=dtype, device=dev)
        for p in params:
            off = self.offsets[id(p)]
            n = p.numel()
            self.flat_param[off : off + n].copy_(p.data.reshape(-1))
            p.data = self.flat_param[off : off + n].view(p.shape)
            p.grad = self.flat_grad[off : off + n].view(p.shape)

    def grad_slice(self, p: nn.Parameter) -> torch.Tensor:
        off = self.offsets[id(p)]
        return self.flat_grad[off : off + p.numel()]

    def zero_grad(self):
        self.flat_grad.zero_()


class FlatDDP(nn.Module):
    """Data-parallel wrapper with bucketed, overlapped RCCL all-reduce."""

    def __init__(self, module: nn.Module, manager: Optional[FlatParamManager] = None,
                 bucket_mb: int = 64, process_group=None, device=None,
                 auto_sync: bool = False, average_grads: bool = False):
        super().__init__()
        self.module = module
        self.manager = manager or FlatParamManager(module, device=device)
        self.pg = process_group
        self.auto_sync = auto_sync
        # True → divide summed grads by world in finish_grad_sync (torch-DDP
        # semantics for user-supplied optimizers); False → the caller folds
        # 1/world into the optimizer (FlatAdamW grad_scale)
        self.average_grads = average_grads
        self._final_cb_queued = False
        self.world = dist.get_world_size(process_group) if dist.is_initialized() else 1
        self._works: List = []
        self._bucket_of: Dict[int, int] = {}
        self._bucket_ranges: List[tuple] = []
        self._bucket_pending: List[int] = []
        self._bucket_total: List[int] = []
        self._hooks = []
        if self.world > 1:
            self._build_buckets(bucket_mb)
            self._register_hooks()
            # parameter sync at init: bucketed broadcast of the flat buffer
            dist.broadcast(self.manager.flat_param, src=0, group=self.pg)

    # buckets are contiguous slices of the flat buffer in backward order
    def _build_buckets(self, bucket_mb: int):
        bucket_elems = bucket_mb * 1024 * 1024 // 2  # bf16
        mgr = self.manager
        start = 0
        cur_elems = 0
        cur_params = 0
        for i, p in enumerate(mgr.params):
            cur_elems += _aligned(p.numel())
            cur_params += 1
            self._bucket_of[id(p)] = len(self._bucket_ranges)
            last = i == len(mgr.params) - 1
            if cur_elems >= bucket_elems or last:
                self._bucket_ranges.append((start, start + cur_elems))
                self._bucket_total.append(cur_params)
                start += cur_elems
                cur_elems = 0
                cur_params = 0
        self._bucket_pending = list(self._bucket_total)

    def _register_hooks(self):
        for p in self.manager.params:
            h = p.register_post_accumulate_grad_hook(self._on_grad_ready)
            self._hooks.append(h)

    def _on_grad_ready(self, p):
        b = self._bucket_of[id(p)]
        self._bucket_pending[b] -= 1
        if self._bucket_pending[b] == 0:
            s, e = self._bucket_ranges[b]
            work = dist.all_reduce(
                self.manager.flat_grad[s:e], op=dist.ReduceOp.SUM,
                group=self.pg, async_op=True,
            )
            self._works.append(work)

    def forward(self, *args, **kwargs):
        out = self.module(*args, **kwargs)
        if self.auto_sync and self.world > 1 and torch.is_grad_enabled():
            self._attach_final_sync(out)
        return out

    def _attach_final_sync(self, out):
        """Queue finish_grad_sync to run at the END of the next backward, so
        user loops written for torch DDP (no explicit sync call) are correct.
        The hook fires at backward START; queue_callback defers to its end."""
        tensors = [t for t in torch.utils._pytree.tree_leaves(out)
                   if isinstance(t, torch.Tensor) and t.requires_grad]
        if not tensors:
            return

        def _on_backward_start(_grad):
            if not self._final_cb_queued:
                self._final_cb_queued = True
                torch.autograd.Variable._execution_engine.queue_callback(
                    self._final_sync_cb
                )
            return _grad

        tensors[0].register_hook(_on_backward_start)

    def _final_sync_cb(self):
        self._final_cb_queued = False
        self.finish_grad_sync()

    def finish_grad_sync(self):
        """Wait for outstanding bucket all-reduces (call before optimizer)."""
        for w in self._works:
            w.wait()
        if self._works and self.average_grads and self.world > 1:
            self.manager.flat_grad.div_(self.world)
        self._works.clear()
        self._bucket_pending = list(self._bucket_total)

    def state_dict(self, *a, **kw):
        return self.module.state_dict(*a, **kw)

    def load_state_dict(self, *a, **kw):
        return self.module.load_state_dict(*a, **kw)


class FlatAdamW:
    """Fused AdamW over the flat buffers (one kernel launch per step)."""

    def __init__(self, manager: FlatParamManager, lr=3e-4, betas=(0.9, 0.95),
                 eps=1e-8, weight_decay=0.0, world_size: Optional[int] = None):
        self.manager = manager
        self.lr = lr
        self.b1, self.b2 = betas
        self.eps = eps
        self.wd = weight_decay
        self.step_count = 0
        self.world = world_size or (dist.get_world_size() if dist.is_initialized() else 1)
        self.master = manager.flat_param.to(torch.float32)
        self.m = torch.zeros_like(self.master)
        self.v = torch.zeros_like(self.master)

    def step(self, grad_scale: float = 1.0):
        self.step_count += 1
        ops.adamw_step(
            self.master, self.manager.flat_param, self.manager.flat_grad,
            self.m, self.v, lr=self.lr, b1=self.b1, b2=self.b2, eps=self.eps,
            wd=self.wd, step=self.step_count, grad_scale=grad_scale / self.world,
        )

    def zero_grad(self, set_to_none: bool = False):
        self.manager.zero_grad()

    def state_dict(self):
        return {
            "step": self.step_count,
            "master": self.master,
            "m": self.m,
            "v": self.v,
            "lr": self.lr,
        }

    def load_state_dict(self, sd):
        self.step_count = sd["step"]
        self.master.copy_(sd["master"])
        self.m.copy_(sd["m"])
        self.v.copy_(sd["v"])
        self.manager.flat_param.copy_(self.master.to(torch.bfloat16))
