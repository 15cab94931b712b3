"""DataLoader construction: prefetching, pinned host buffers, DDP sharding.

The reference trains off synchronous cv2 reads on the training thread
(flyingChairsTrain.py:173-178); here loading runs in worker processes
with pinned memory so H2D copies overlap compute on a side HIP stream.
"""

from __future__ import annotations

import torch
from torch.utils.data import DataLoader, DistributedSampler


def build_dataloader(dataset, batch_size: int, shuffle: bool = True,
                     num_workers: int = 4, distributed: bool = False,
                     drop_last: bool = True, seed: int = 0):
    sampler = None
    if distributed:
        sampler = DistributedSampler(dataset, shuffle=shuffle, seed=seed,
                                     drop_last=drop_last)
        shuffle = False
    return DataLoader(
        dataset,
        batch_size=batch_size,
        shuffle=shuffle,
        sampler=sampler,
        num_workers=num_workers,
        pin_memory=torch.cuda.is_available(),
        drop_last=drop_last,
        persistent_workers=num_workers > 0,
    )


class CudaPrefetcher:
    """Async H2D prefetch on a dedicated stream (one batch ahead)."""

    def __init__(self, loader, device):
        self.loader = iter(loader)
        self.device = device
        self.stream = torch.cuda.Stream(device)
        self._next = None
        self._preload()

    def _preload(self):
        try:
            batch = next(self.loader)
        except StopIteration:
            self._next = None
            return
        with torch.cuda.stream(self.stream):
            self._next = {
                k: v.to(self.device, non_blocking=True)
                if isinstance(v, torch.Tensor) else v
                for k, v in batch.items()
            }

    def __iter__(self):
        return self

    def __next__(self):
        if self._next is None:
            raise StopIteration
        cur = torch.cuda.current_stream(self.device)
        cur.wait_stream(self.stream)
        batch = self._next
        # tensors were allocated on the side stream but are consumed on
        # the compute stream: tell the caching allocator, or it may hand
        # the memory to a later side-stream H2D copy while compute-stream
        # kernels still read it (standard apex-prefetcher guard)
        for v in batch.values():
            if isinstance(v, torch.Tensor):
                v.record_stream(cur)
        self._preload()
        return batch
