"""Growable device arenas (reference: _memory_utility.py DeviceMemory /
HostPinnedMemory, ~L1-350).

A ``DeviceBuffer`` owns one allocation that only grows; communicators keep
two of them (pack source / all-reduce destination) so no allocation happens
on the hot path and async collectives never race the torch caching
allocator (the tensor is owned here, not by autograd).
"""

import torch


class DeviceBuffer(object):

    def __init__(self, device=None, pinned=False):
        self._device = device
        self._pinned = pinned
        self._buf = None  # uint8 backing

    def array(self, numel, dtype=torch.float32):
        """A flat tensor view of ``numel`` elements, growing the backing."""
        nbytes = numel * torch.empty(0, dtype=dtype).element_size()
        if self._buf is None or self._buf.numel() < nbytes:
            if self._pinned:
                self._buf = torch.empty(nbytes, dtype=torch.uint8,
                                        pin_memory=True)
            else:
                self._buf = torch.empty(nbytes, dtype=torch.uint8,
                                        device=self._device)
        return self._buf[:nbytes].view(dtype)
