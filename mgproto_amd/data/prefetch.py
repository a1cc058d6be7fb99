"""Device prefetcher: overlap H2D copies of the next batch with compute.

The trainer's default path does ``tensor.to(device, non_blocking=True)``
at the top of each step, which serializes the copy against the step's
first kernels unless the DMA engine gets a head start. ``DevicePrefetcher``
wraps a DataLoader and stages batch i+1's copies on a side HIP stream
while the model computes on batch i; ``next()`` waits on the copy event
and hands over device tensors.

Opt-in (train.py ``--prefetch``): the synthetic bench path uses
DeviceBatchPool (already resident) and does not need it. On CPU the
wrapper degrades to a plain iterator (no streams).
"""

import torch


class DevicePrefetcher:
    def __init__(self, loader, device):
        self.loader = loader
        self.device = device
        self.use_stream = (device is not None and device.type == 'cuda')
        self.stream = torch.cuda.Stream() if self.use_stream else None

    def __len__(self):
        return len(self.loader)

    def _to_device(self, batch):
        return tuple(t.to(self.device, non_blocking=True)
                     if torch.is_tensor(t) else t for t in batch)

    def __iter__(self):
        if not self.use_stream:
            for batch in self.loader:
                yield self._to_device(batch) if self.device is not None \
                    else batch
            return

        it = iter(self.loader)
        preloaded = None
        event = None
        try:
            host = next(it)
        except StopIteration:
            return
        while True:
            with torch.cuda.stream(self.stream):
                preloaded = self._to_device(host)
            event = torch.cuda.Event()
            event.record(self.stream)
            try:
                host = next(it)
            except StopIteration:
                host = None
            torch.cuda.current_stream().wait_event(event)
            # the consumer uses these tensors on the current stream; tie
            # their lifetime to it so the allocator cannot hand the blocks
            # back to the copy stream while the step still reads them
            for t in preloaded:
                if torch.is_tensor(t):
                    t.record_stream(torch.cuda.current_stream())
            yield preloaded
            if host is None:
                return
