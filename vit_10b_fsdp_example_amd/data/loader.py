"""Double-buffered host->device prefetch loader.

MI355X-native replacement for torch_xla's pl.MpDeviceLoader (reference
run_vit_training.py:74,88).  The XLA loader's job was (a)
background host->device transfer and (b) the per-batch graph-execution
barrier.  Eager HIP needs no barrier, so what remains is classic copy /
compute overlap: batches are staged from the DataLoader's pinned-memory
workers to the GPU with hipMemcpyAsync on a dedicated copy stream, one
batch ahead, and handed to the compute stream with a stream-wait (no
host sync in the steady state).
"""

import torch


class DeviceLoader:
    def __init__(self, loader, device, compute_dtype=None):
        self._loader = loader
        self._device = torch.device(device)
        self._compute_dtype = compute_dtype
        self._use_cuda = self._device.type == "cuda"
        self._copy_stream = torch.cuda.Stream() if self._use_cuda else None

    def __len__(self):
        return len(self._loader)

    @property
    def sampler(self):
        return self._loader.sampler

    def _move(self, batch):
        data, target = batch
        if self._use_cuda:
            with torch.cuda.stream(self._copy_stream):
                data = data.to(self._device, non_blocking=True)
                if self._compute_dtype is not None and data.is_floating_point():
                    data = data.to(self._compute_dtype)
                target = target.to(self._device, non_blocking=True)
        else:
            data = data.to(self._device)
            if self._compute_dtype is not None and data.is_floating_point():
                data = data.to(self._compute_dtype)
            target = target.to(self._device)
        return data, target

    def __iter__(self):
        if not self._use_cuda:
            for batch in self._loader:
                yield self._move(batch)
            return

        it = iter(self._loader)
        prefetched = None
        try:
            prefetched = self._move(next(it))
        except StopIteration:
            return
        for batch in it:
            nxt = self._move(batch)
            # Hand the previously staged batch to the compute stream.
            torch.cuda.current_stream().wait_stream(self._copy_stream)
            # The staged tensors are consumed on the compute stream; make
            # sure the caching allocator doesn't hand their memory back to
            # the copy stream while the compute stream still reads them.
            for t in prefetched:
                if torch.is_tensor(t):
                    t.record_stream(torch.cuda.current_stream())
            yield prefetched
            prefetched = nxt
        torch.cuda.current_stream().wait_stream(self._copy_stream)
        for t in prefetched:
            if torch.is_tensor(t):
                t.record_stream(torch.cuda.current_stream())
        yield prefetched
