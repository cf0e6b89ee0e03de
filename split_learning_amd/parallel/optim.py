"""Optimizer wrappers: fused HIP multi-tensor kernels on GPU, torch on CPU.

Semantics match the reference's optimizer choices exactly:
SGD(lr, momentum) for VGG16/MobileNet/ViT (src/train/VGG16.py:62),
AdamW(lr, weight_decay) for BERT/KWT (src/train/BERT.py:69, KWT.py:62).
"""

from __future__ import annotations

from typing import List

import torch



SLK_OPT_CHUNK = 16384


def _build_desc(host_pin, live_params, live_grads, s1, s2, device):
    """Fill the fused-step pointer table (layout = optim.hip make_opt_desc:
    prefix[n+1], numels[n], param/grad/state1/state2 ptrs[n]) into the
    preallocated PINNED host buffer and async-copy it to the device.

    Built in Python (not via the pageable-H2D C++ helper) so it is legal
    INSIDE hipGraph capture: the device tensor comes from the capture pool and
    the H2D copy from pinned memory is recorded as a graph memcpy node that
    re-reads the (kept-alive, constant) pinned buffer on every replay."""
    n = len(live_params)
    vals = [0] * (n + 1)
    numels, pp, gp, b1, b2 = [], [], [], [], []
    chunks = 0
    for i, p in enumerate(live_params):
        vals[i] = chunks
        ne = p.numel()
        numels.append(ne)
        chunks += (ne + SLK_OPT_CHUNK - 1) // SLK_OPT_CHUNK
        pp.append(p.data_ptr())
        gp.append(live_grads[i].data_ptr())
        b1.append(s1[i].data_ptr())
        b2.append(s2[i].data_ptr())
    vals[n] = chunks
    vals += numels + pp + gp + b1 + b2
    m = len(vals)
    host_pin[:m].copy_(torch.tensor(vals, dtype=torch.int64))  # host-side
    desc = torch.empty(m, dtype=torch.int64, device=device)
    desc.copy_(host_pin[:m], non_blocking=True)
    return desc, chunks


def _pinned_host(n_params: int) -> torch.Tensor:
    host = torch.empty(6 * n_params + 1, dtype=torch.int64)
    if torch.cuda.is_available():
        host = host.pin_memory()  # preallocated OUTSIDE any graph capture
    return host


# A captured graph's memcpy node re-reads its pinned source buffer on every
# replay, so a buffer referenced by a live graph must never be overwritten or
# freed.  Rebuilds outside capture therefore take a FRESH pinned buffer and
# retire the old one here (tiny: <= a few hundred int64 each); the
# preallocated per-optimizer buffer is reserved for the one build that can
# happen INSIDE capture, where pinning anew would not be capture-legal.
_RETIRED_HOSTS: List[torch.Tensor] = []


def _host_for_build(opt) -> torch.Tensor:
    """Pinned buffer for a (re)build: inside capture reuse the preallocated
    one (pinning anew is not capture-legal); outside capture take a fresh
    buffer and retire the old so any live graph's memcpy source stays
    intact (see _RETIRED_HOSTS)."""
    if torch.cuda.is_available() and torch.cuda.is_current_stream_capturing():
        return opt._host
    if opt._desc_cache is not None:
        _RETIRED_HOSTS.append(opt._host)
        opt._host = _pinned_host(len(opt.params))
    return opt._host


class FusedSGD:
    def __init__(self, params, lr: float, momentum: float = 0.0,
                 weight_decay: float = 0.0):
        self.params: List[torch.Tensor] = [p for p in params if p.requires_grad]
        self.lr = lr
        self.momentum = momentum
        self.weight_decay = weight_decay
        self.bufs = [torch.zeros_like(p) for p in self.params]
        self.steps = 0
        self._desc_cache = None  # (ptr_signature, desc_tensor, n, chunks)
        self._host = _pinned_host(len(self.params))
        # release_grads=True (default): drop p.grad after the fused step so the
        # next backward MOVES fresh gradients in (AccumulateGrad steals the
        # producing kernel's output tensor — no accumulate-add kernel per
        # parameter).  This also holds under hipGraph capture: grads are None
        # when the captured step runs, the capture-pool grad buffers stay
        # stable across replays, and every backward kernel fully overwrites
        # its gradient output, so no inter-replay zeroing is needed.
        self.release_grads = True

    def _desc(self, live):
        """Cached device descriptor for the single-launch fused step; with
        release_grads the allocator usually hands the same blocks back, so
        the signature (and the table) is stable across eager steps."""
        sig = tuple(p.grad.data_ptr() for p, _ in live) + \
              tuple(p.data_ptr() for p, _ in live)
        if self._desc_cache is None or self._desc_cache[0] != sig:
            params = [p for p, _ in live]
            bufs = [b for _, b in live]
            desc, chunks = _build_desc(_host_for_build(self), params,
                                       [p.grad for p, _ in live], bufs, bufs,
                                       params[0].device)
            self._desc_cache = (sig, desc, len(params), chunks)
        return self._desc_cache[1], self._desc_cache[2], self._desc_cache[3]

    def zero_grad(self):
        """No-op by design: step() zeroes grads in the update kernel itself
        (zero_grad_after), so autograd accumulates each microbatch into
        already-zeroed buffers without extra fill launches."""

    @torch.no_grad()
    def step(self):
        live = [(p, b) for p, b in zip(self.params, self.bufs) if p.grad is not None]
        if not live:
            return
        if live[0][0].is_cuda:
            desc, n, chunks = self._desc(live)
            from ..ops import native
            native().sgd_step_fused(desc, n, chunks, self.lr, self.momentum,
                                    self.weight_decay, self.steps == 0,
                                    not self.release_grads)
            if self.release_grads:
                # next backward MOVES fresh gradients in (no accumulate-add
                # kernel per tensor); the allocator usually hands back the same
                # blocks, so the descriptor cache still hits
                for p, _ in live:
                    p.grad = None
        else:
            for p, buf in live:
                g = p.grad
                if self.weight_decay:
                    g = g + self.weight_decay * p
                if self.momentum:
                    if self.steps == 0:
                        buf.copy_(g)
                    else:
                        buf.mul_(self.momentum).add_(g)
                    g = buf
                p.add_(g, alpha=-self.lr)
                p.grad.zero_()
        self.steps += 1


class FusedAdamW:
    def __init__(self, params, lr: float, weight_decay: float = 0.01,
                 betas=(0.9, 0.999), eps: float = 1e-8):
        self.params = [p for p in params if p.requires_grad]
        self.lr = lr
        self.weight_decay = weight_decay
        self.beta1, self.beta2 = betas
        self.eps = eps
        self.m = [torch.zeros_like(p) for p in self.params]
        self.v = [torch.zeros_like(p) for p in self.params]
        self.steps = 0
        self._desc_cache = None
        self._host = _pinned_host(len(self.params))
        self.release_grads = True  # see FusedSGD

    def _desc(self, live):
        sig = tuple(p.grad.data_ptr() for p, _, _ in live) + \
              tuple(p.data_ptr() for p, _, _ in live)
        if self._desc_cache is None or self._desc_cache[0] != sig:
            params = [p for p, _, _ in live]
            desc, chunks = _build_desc(_host_for_build(self), params,
                                       [p.grad for p, _, _ in live],
                                       [m for _, m, _ in live],
                                       [v for _, _, v in live],
                                       params[0].device)
            self._desc_cache = (sig, desc, len(params), chunks)
        return self._desc_cache[1], self._desc_cache[2], self._desc_cache[3]

    def zero_grad(self):
        """No-op: see FusedSGD.zero_grad."""

    @torch.no_grad()
    def step(self):
        self.steps += 1
        live = [(p, m, v) for p, m, v in zip(self.params, self.m, self.v)
                if p.grad is not None]
        if not live:
            return
        if live[0][0].is_cuda:
            desc, n, chunks = self._desc(live)
            from ..ops import native
            native().adamw_step_fused(desc, n, chunks, self.steps, self.lr,
                                      self.beta1, self.beta2, self.eps,
                                      self.weight_decay, not self.release_grads)
            if self.release_grads:
                for p, _, _ in live:
                    p.grad = None
        else:
            b1, b2 = self.beta1, self.beta2
            bc1 = 1 - b1 ** self.steps
            bc2 = 1 - b2 ** self.steps
            for p, m, v in live:
                p.mul_(1 - self.lr * self.weight_decay)
                m.mul_(b1).add_(p.grad, alpha=1 - b1)
                v.mul_(b2).addcmul_(p.grad, p.grad, value=1 - b2)
                p.addcdiv_(m / bc1, (v / bc2).sqrt().add_(self.eps), value=-self.lr)
                p.grad.zero_()


def make_optimizer(model_name: str, params, learning: dict):
    """Reference mapping: VGG16/MobileNetv1/ViT -> SGD+momentum; BERT/KWT -> AdamW."""
    if model_name in ("BERT", "KWT"):
        return FusedAdamW(params, lr=learning["learning-rate"],
                          weight_decay=learning.get("weight-decay", 0.01))
    return FusedSGD(params, lr=learning["learning-rate"],
                    momentum=learning.get("momentum", 0.0))
