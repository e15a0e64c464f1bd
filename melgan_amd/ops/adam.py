"""Multi-tensor fused Adam.

One HIP kernel launch updates ALL parameter tensors of a model per step
(the generator has ~80 small conv weights; per-tensor eager Adam is
launch-bound).  Tensors are described by a device-side pointer table;
the kernel grid-strides over the concatenated element range.

Falls back to torch.optim.Adam semantics exactly (same state dict
layout: step / exp_avg / exp_avg_sq per param) so checkpoints are
interchangeable with the reference's optimizer state.
"""

from __future__ import annotations

import torch

from melgan_amd.ops import _ext


class FusedAdam(torch.optim.Optimizer):
    def __init__(self, params, lr=1e-4, betas=(0.5, 0.9), eps=1e-8,
                 weight_decay=0.0):
        defaults = dict(lr=lr, betas=betas, eps=eps,
                        weight_decay=weight_decay)
        super().__init__(params, defaults)

    @torch.no_grad()
    def step(self, closure=None):
        loss = None
        if closure is not None:
            with torch.enable_grad():
                loss = closure()
        for group in self.param_groups:
            params, grads, exp_avgs, exp_avg_sqs, steps = [], [], [], [], []
            for p in group["params"]:
                if p.grad is None:
                    continue
                state = self.state[p]
                if len(state) == 0:
                    state["step"] = torch.zeros((), dtype=torch.float32)
                    state["exp_avg"] = torch.zeros_like(p)
                    state["exp_avg_sq"] = torch.zeros_like(p)
                params.append(p)
                grads.append(p.grad)
                exp_avgs.append(state["exp_avg"])
                exp_avg_sqs.append(state["exp_avg_sq"])
                state["step"] += 1
                steps.append(int(state["step"]))
            if not params:
                continue
            beta1, beta2 = group["betas"]
            e = _ext.ext()
            if e is not None and params[0].is_cuda and hasattr(
                    e, "multi_tensor_adam_tbl"):
                # the fused kernel takes ONE bias-correction step count per
                # launch — partition by step value so a param whose grad
                # first appeared later (param added / temporarily grad-less)
                # gets its own correct correction.  In steady state all
                # params share one step -> one launch, cached tables.
                by_step: dict[int, list[int]] = {}
                for i, s in enumerate(steps):
                    by_step.setdefault(s, []).append(i)
                for s, idxs in by_step.items():
                    sel = (lambda lst: [lst[i] for i in idxs])
                    # cache key from the PARTITION (not the step value,
                    # which increments every step and would kill caching)
                    tbl, chunks, nt = self._tables(
                        (id(group), tuple(idxs)), sel(params), sel(grads),
                        sel(exp_avgs), sel(exp_avg_sqs))
                    e.multi_tensor_adam_tbl(
                        tbl, chunks, nt, group["lr"], beta1, beta2,
                        group["eps"], group["weight_decay"], s)
                # the kernel writes params through raw pointers, which
                # does NOT bump torch's _version counter — bump an
                # explicit epoch so weight-norm fold caches (keyed on it)
                # invalidate (frozen-weights bug otherwise!)
                for p in params:
                    p._wn_epoch = getattr(p, "_wn_epoch", 0) + 1
            else:
                # torch's own multi-tensor path (same math)
                torch._foreach_mul_(exp_avgs, beta1)
                torch._foreach_add_(exp_avgs, grads, alpha=1 - beta1)
                torch._foreach_mul_(exp_avg_sqs, beta2)
                torch._foreach_addcmul_(exp_avg_sqs, grads, grads,
                                        value=1 - beta2)
                for p, m, v, t in zip(params, exp_avgs, exp_avg_sqs, steps):
                    bc1 = 1 - beta1 ** t
                    bc2 = 1 - beta2 ** t
                    denom = (v / bc2).sqrt_().add_(group["eps"])
                    if group["weight_decay"]:
                        p.mul_(1 - group["lr"] * group["weight_decay"])
                    p.addcdiv_(m, denom, value=-group["lr"] / bc1)
        return loss

    def _tables(self, key, params, grads, exp_avgs, exp_avg_sqs):
        """Device pointer tables, cached across steps (rebuilt only when a
        pointer changes, e.g. grads reallocated). Keeps the step at ONE
        kernel launch with no per-step H2D, and is hipGraph-capture safe
        once pointers are static."""
        import torch as _t

        CHUNK = 1 << 13  # must match ADAM_CHUNK in kernels.hip
        ptrs = []
        for lst in (params, grads, exp_avgs, exp_avg_sqs):
            ptrs.extend(t.data_ptr() for t in lst)
        cache = getattr(self, "_tbl_cache", {}).get(key)
        if cache is not None and cache[0] == ptrs:
            return cache[1], cache[2], cache[3]
        nt = len(params)
        host = ptrs + [p.numel() for p in params]
        chunks = []
        for i, p in enumerate(params):
            for off in range(0, p.numel(), CHUNK):
                chunks.append((i << 48) | off)
        dev = params[0].device
        tbl = _t.tensor(host, dtype=_t.long).to(dev)
        ch = _t.tensor(chunks, dtype=_t.long).to(dev)
        if not hasattr(self, "_tbl_cache"):
            self._tbl_cache = {}
        self._tbl_cache[key] = (ptrs, tbl, ch, nt)
        return tbl, ch, nt
