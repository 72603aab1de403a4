"""Synthetic ImageNet-shaped input pipeline.

Replaces the reference's feed_dict/tf.data input path with the MI355X-native
equivalent mandated by BASELINE.json: synthetic 224x224x3 tensors staged in
PINNED host memory and copied H2D with hipMemcpyAsync (torch non_blocking
copy on a dedicated copy stream), double-buffered so step k+1's H2D overlaps
step k's compute.

A small pool of pre-generated batches cycles forever — generation cost is
off the timed path, the per-step H2D copy (the part the reference also pays)
is kept.
"""

from __future__ import annotations

from typing import Tuple

import torch


class SyntheticPipeline:
    def __init__(self, batch_size: int, image_size: int, channels: int,
                 num_classes: int, device: torch.device, dtype: torch.dtype,
                 seed: int = 0, pool: int = 4):
        self.device = device
        g = torch.Generator().manual_seed(seed)
        shape = (batch_size, image_size, image_size, channels)
        pin = device.type == "cuda"
        self.images = []
        self.labels = []
        for _ in range(pool):
            img = torch.rand(shape, generator=g, dtype=torch.float32)
            img = img.mul_(2.0).sub_(1.0).to(dtype)
            lab = torch.randint(0, num_classes, (batch_size,), generator=g)
            if pin:
                img = img.pin_memory()
                lab = lab.pin_memory()
            self.images.append(img)
            self.labels.append(lab)
        self._i = 0
        if pin:
            self.copy_stream = torch.cuda.Stream(device)
            # double device buffers
            self._dev = [(torch.empty(shape, dtype=dtype, device=device),
                          torch.empty((batch_size,), dtype=torch.long,
                                      device=device)) for _ in range(2)]
            self._events = [torch.cuda.Event(), torch.cuda.Event()]
            self._slot = 0
            self._prefetch(self._slot)
        else:
            self.copy_stream = None

    def _prefetch(self, slot: int, wait_compute: bool = False) -> None:
        i = self._i % len(self.images)
        self._i += 1
        dimg, dlab = self._dev[slot]
        if wait_compute:
            # All compute that reads this slot is already enqueued on the
            # current (compute) stream — fence the overwrite behind it.
            ev_free = torch.cuda.Event()
            ev_free.record(torch.cuda.current_stream())
            self.copy_stream.wait_event(ev_free)
        with torch.cuda.stream(self.copy_stream):
            dimg.copy_(self.images[i], non_blocking=True)
            dlab.copy_(self.labels[i], non_blocking=True)
            self._events[slot].record(self.copy_stream)

    def next(self) -> Tuple[torch.Tensor, torch.Tensor]:
        if self.copy_stream is None:
            i = self._i % len(self.images)
            self._i += 1
            return self.images[i], self.labels[i]
        slot = self._slot
        self._slot ^= 1
        # compute stream waits for this slot's H2D; start next H2D at once
        torch.cuda.current_stream().wait_event(self._events[slot])
        out = self._dev[slot]
        self._prefetch(self._slot, wait_compute=True)
        return out
