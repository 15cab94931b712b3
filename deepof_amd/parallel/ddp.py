"""Data parallelism over RCCL/xGMI: bucketed all-reduce overlapped with
backward.

The reference has no distributed anything (single tf.Session on
'/gpu:0', /root/reference/version1/trainOF.py:97); this module is the
MI355X-native DP design of SURVEY §5.8: one process per GPU,
torch.distributed with the "nccl" backend (= RCCL on ROCm), gradients
pre-bucketed into flat fp32 buffers whose slices ARE the params' .grad
views, each bucket all-reduced asynchronously the moment its last grad
is accumulated — so the reduce of late (deep) buckets hides under the
backward of early layers.  xGMI is a 7-link point-to-point clique
(≈153 GB/s/link); bucket size defaults to 25 MB so RCCL's pipelined
algorithms keep all links busy without serializing the tail.
"""

from __future__ import annotations

import os

import torch
import torch.distributed as dist


def is_distributed() -> bool:
    return dist.is_available() and dist.is_initialized()


def init_distributed(backend: str | None = None) -> tuple[int, int, int]:
    """Initialize from torchrun env vars; returns (rank, local_rank, world).

    No-op single-process fallback when WORLD_SIZE is absent/1.
    """
    world = int(os.environ.get("WORLD_SIZE", "1"))
    if world <= 1:
        return 0, 0, 1
    if backend is None:
        backend = "nccl" if torch.cuda.is_available() else "gloo"
    if not dist.is_initialized():
        dist.init_process_group(backend=backend)
    rank = dist.get_rank()
    local_rank = int(os.environ.get("LOCAL_RANK", rank % max(torch.cuda.device_count(), 1)))
    if torch.cuda.is_available():
        torch.cuda.set_device(local_rank)
    return rank, local_rank, world


class _Bucket:
    __slots__ = ("flat", "params", "pending", "work")

    def __init__(self, flat, params):
        self.flat = flat
        self.params = params
        self.pending = 0
        self.work = None


class BucketedDataParallel(torch.nn.Module):
    """Wraps a module; gradients are averaged across ranks automatically.

    Usage:
        model = BucketedDataParallel(model)
        loss.backward()
        model.finish_gradient_sync()   # before optimizer.step()
    """

    def __init__(self, module: torch.nn.Module, bucket_cap_mb: float = 25.0,
                 process_group=None):
        super().__init__()
        self.module = module
        self.pg = process_group
        # gradient-accumulation support: while True, backward passes
        # accumulate into the flat buckets WITHOUT all-reducing (the
        # boundary micro-batch clears it and reduces the sums)
        self.accumulate_only = False
        self.world = dist.get_world_size(process_group) if is_distributed() else 1
        self._hooks = []
        self._buckets: list[_Bucket] = []
        self._param_bucket: dict[int, _Bucket] = {}

        if self.world > 1:
            self._broadcast_params()
            self._build_buckets(int(bucket_cap_mb * 1024 * 1024))

    # -- setup ------------------------------------------------------------
    def _broadcast_params(self):
        for t in self.module.state_dict().values():
            if not (isinstance(t, torch.Tensor) and t.numel()):
                continue
            if t.is_contiguous():
                dist.broadcast(t.data, src=0, group=self.pg)
            else:  # channels_last params: collectives want contiguous
                buf = t.data.contiguous()
                dist.broadcast(buf, src=0, group=self.pg)
                t.data.copy_(buf)

    def _build_buckets(self, cap_bytes: int):
        params = [p for p in self.module.parameters() if p.requires_grad]
        # reverse registration order ~ backward completion order
        params = params[::-1]
        group: list[torch.nn.Parameter] = []
        size = 0
        groups = []
        for p in params:
            nbytes = p.numel() * 4  # grads kept fp32
            if group and size + nbytes > cap_bytes:
                groups.append(group)
                group, size = [], 0
            group.append(p)
            size += nbytes
        if group:
            groups.append(group)

        dev = params[0].device if params else torch.device("cpu")
        for g in groups:
            total = sum(p.numel() for p in g)
            flat = torch.zeros(total, dtype=torch.float32, device=dev)
            bucket = _Bucket(flat, g)
            off = 0
            for p in g:
                n = p.numel()
                # autograd accumulates straight into the flat slice; the
                # view's strides must match the param's memory format or
                # AccumulateGrad copies every step (channels_last convs)
                sl = flat[off : off + n]
                if (p.dim() == 4 and
                        p.is_contiguous(memory_format=torch.channels_last)
                        and not p.is_contiguous()):
                    N, C, H, W = p.shape
                    p.grad = sl.view(N, H, W, C).permute(0, 3, 1, 2)
                else:
                    p.grad = sl.view_as(p)
                off += n
                self._param_bucket[id(p)] = bucket
                h = p.register_post_accumulate_grad_hook(self._on_grad_ready)
                self._hooks.append(h)
            self._buckets.append(bucket)
        self._reset_pending()

    def _reset_pending(self):
        for b in self._buckets:
            b.pending = len(b.params)
            b.work = None

    # -- runtime ----------------------------------------------------------
    def _on_grad_ready(self, param):
        b = self._param_bucket[id(param)]
        b.pending -= 1
        if b.pending == 0:
            if self.accumulate_only:
                b.pending = len(b.params)  # re-arm, no reduce this pass
            else:
                b.work = dist.all_reduce(b.flat, op=dist.ReduceOp.SUM,
                                         group=self.pg, async_op=True)

    def forward(self, *args, **kwargs):
        return self.module(*args, **kwargs)

    def finish_gradient_sync(self):
        """Wait for in-flight all-reduces and average; call before step()."""
        if self.world <= 1:
            return
        inv = 1.0 / self.world
        for b in self._buckets:
            if b.work is None and b.pending != len(b.params):
                # partial bucket (grad accumulation edge) — reduce now
                b.work = dist.all_reduce(b.flat, op=dist.ReduceOp.SUM,
                                         group=self.pg, async_op=True)
        for b in self._buckets:
            if b.work is not None:
                b.work.wait()
                b.flat.mul_(inv)
        self._reset_pending()

    def zero_grad_buckets(self):
        for b in self._buckets:
            b.flat.zero_()
