"""BaguaTensor — a communication-registered view of a torch tensor.

The reference gorilla-patched ``torch.Tensor`` with a proxy/effective-tensor
design (reference: bagua/torch_api/tensor.py:19-263). This build keeps the
same *semantics* — register a proxy (usually the parameter) plus a getter
closure so the backend always reaches the live tensor (usually ``.grad``),
and a setter closure so fusion can re-point the live tensor into bucket
storage — but as an explicit object instead of monkey-patching, which keeps
us compatible with current PyTorch and makes ownership obvious.
"""

from typing import Callable, Optional

import torch


class BaguaTensor:
    """Registration record for one communication tensor.

    Args:
        proxy: the torch tensor the registration hangs off (e.g. the
            parameter). Kept alive by this object.
        name: globally unique name inside one backend.
        getter_closure: maps proxy -> effective tensor (e.g.
            ``lambda p: p.grad``). None means the proxy itself is effective.
        setter_closure: called as ``setter_closure(proxy, new_tensor)`` when
            fusion re-points the effective tensor into bucket storage.
    """

    def __init__(
        self,
        proxy: torch.Tensor,
        name: str,
        getter_closure: Optional[Callable] = None,
        setter_closure: Optional[Callable] = None,
    ):
        if setter_closure is not None and getter_closure is None:
            raise ValueError(
                "must provide getter_closure when setter_closure is set")
        self.proxy = proxy
        self.name = name
        self.getter_closure = getter_closure
        self.setter_closure = setter_closure
        # scheduling state, managed by the backend
        self.ready = False
        self.ready_event: Optional[torch.cuda.Event] = None
        self.bucket = None  # back-pointer set by BaguaBucket
        self._bucket_view: Optional[torch.Tensor] = None

    # ------------------------------------------------------------------
    def tensor(self) -> torch.Tensor:
        """The live (effective) tensor the backend communicates."""
        if self.getter_closure is not None:
            return self.getter_closure(self.proxy)
        return self.proxy

    def materialized(self) -> torch.Tensor:
        """The effective tensor, materializing it (zeros shaped like the
        proxy) when a grad-getter currently returns None — happens when a
        re-bucket lands between zero_grad(set_to_none) and backward."""
        eff = self.tensor()
        if eff is None:
            fresh = torch.zeros_like(self.proxy)
            if self.setter_closure is not None:
                self.setter_closure(self.proxy, fresh)
            eff = self.tensor()
        return eff

    def data_ptr(self) -> int:
        return self.tensor().data_ptr()

    def numel(self) -> int:
        return self.tensor().numel()

    def ensure_grad(self):
        """Allocate ``proxy.grad`` if missing (reference: tensor.py:190-204)."""
        p = self.proxy
        if isinstance(p, torch.Tensor) and p.grad is None:
            p.grad = torch.zeros_like(p)
        return self

    def set_storage(self, flat: torch.Tensor, offset: int):
        """Re-point the effective tensor into ``flat`` at ``offset``
        (reference: tensor.py:239-263). The new view keeps the original
        shape; the setter closure installs it back on the proxy."""
        eff = self.tensor()
        new_view = flat.narrow(0, offset, eff.numel()).view_as(eff)
        new_view.copy_(eff)
        if self.setter_closure is not None:
            self.setter_closure(self.proxy, new_view)
        elif self.getter_closure is None:
            # effective tensor IS the proxy: re-point proxy storage by
            # swapping .data (works for non-leaf-sensitive registrations)
            self.proxy.data = new_view
        else:
            raise RuntimeError(
                "cannot re-point tensor %s: getter without setter" % self.name)
        self._bucket_view = new_view

    def repair_bucket_view(self):
        """Re-pin the effective tensor into bucket storage if something
        (e.g. ``optimizer.zero_grad(set_to_none=True)``) replaced it.

        torch 2.x zero_grad defaults to set_to_none, which drops the fused
        grad view the reference relied on keeping alive
        (reference: tensor.py:239-263). Cheap data_ptr check per call;
        copies only when the aliasing was actually broken."""
        if self._bucket_view is None:
            return
        eff = self.tensor()
        if eff is None:
            if self.setter_closure is not None:
                self._bucket_view.zero_()
                self.setter_closure(self.proxy, self._bucket_view)
            return
        if eff.data_ptr() != self._bucket_view.data_ptr():
            self._bucket_view.copy_(eff.detach())
            if self.setter_closure is not None:
                self.setter_closure(self.proxy, self._bucket_view)
            elif self.getter_closure is None:
                self.proxy.data = self._bucket_view

    def mark_communication_ready(self, backend):
        """Record a ready event on the current stream and notify the
        scheduler (reference: tensor.py:214-226)."""
        if self.tensor().is_cuda:
            ev = backend.event_pool.get()
            ev.record(torch.cuda.current_stream())
            self.ready_event = ev
        backend.mark_communication_ready(self)

    def mark_communication_ready_without_synchronization(self, backend):
        self.ready_event = None
        backend.mark_communication_ready(self)

    def is_bagua_tensor(self) -> bool:
        return True


def ensure_bagua_tensor(
    tensor: torch.Tensor,
    name: str,
    getter_closure: Optional[Callable] = None,
    setter_closure: Optional[Callable] = None,
) -> BaguaTensor:
    """Create (or reuse) the BaguaTensor registration attached to a torch
    tensor (reference: tensor.py:57-137)."""
    state = getattr(tensor, "_bagua_state", None)
    if state is not None and state.name == name:
        state.getter_closure = getter_closure
        state.setter_closure = setter_closure
        return state
    state = BaguaTensor(tensor, name, getter_closure, setter_closure)
    tensor._bagua_state = state
    return state


def to_bagua_tensor(tensor: torch.Tensor, name: str) -> BaguaTensor:
    return ensure_bagua_tensor(tensor, name)
