"""Fused multi-tensor Adam.

The reference uses tf.train.AdamOptimizer over ~38M FlowNetS params
(/root/reference/flyingChairsTrain.py:124).  On GPU every step runs ONE
hand-written HIP kernel over all tensors (chunked multi-tensor apply:
one launch, grid-strided over a packed pointer table) instead of
hundreds of elementwise launches.  CPU falls back to torch's foreach
Adam math (same update, used as the numerics reference in tests).
"""

from __future__ import annotations

import math

import torch
from torch.optim import Optimizer


class FusedAdam(Optimizer):
    def __init__(self, params, lr=1e-3, betas=(0.9, 0.999), eps=1e-8,
                 weight_decay=0.0):
        defaults = dict(lr=lr, betas=betas, eps=eps,
                        weight_decay=weight_decay)
        super().__init__(params, defaults)
        # cached device chunk table + pinned hyper buffer (hipGraph-safe:
        # the captured step re-reads [lr, bias1, bias2] from pinned host
        # memory at every replay)
        self._tables = {}  # group index -> (sig, table, n_chunks)
        self._hyper_pin = None
        self._hyper_dev = None

    @torch.no_grad()
    def step(self, closure=None):
        loss = None
        if closure is not None:
            with torch.enable_grad():
                loss = closure()

        for gi, group in enumerate(self.param_groups):
            params, grads, exp_avgs, exp_avg_sqs = [], [], [], []
            for p in group["params"]:
                if p.grad is None:
                    continue
                state = self.state[p]
                if not state:
                    state["step"] = 0
                    state["exp_avg"] = torch.zeros_like(p)
                    state["exp_avg_sq"] = torch.zeros_like(p)
                state["step"] += 1
                params.append(p)
                grads.append(p.grad)
                exp_avgs.append(state["exp_avg"])
                exp_avg_sqs.append(state["exp_avg_sq"])
            if not params:
                continue

            beta1, beta2 = group["betas"]
            step = self.state[params[0]]["step"]
            bias1 = 1.0 - beta1**step
            bias2 = 1.0 - beta2**step

            if params[0].is_cuda:
                from ..ops.functional import require_hip

                hip = require_hip()
                # EVERY grad pointer is part of the signature: with
                # zero_grad(set_to_none=True) grads are reallocated each
                # backward and any one of them can move independently.
                # Moment pointers too: load_state_dict() replaces the
                # exp_avg/exp_avg_sq tensors.
                sig = (len(params), params[0].data_ptr(),
                       params[-1].data_ptr(),
                       exp_avgs[0].data_ptr(), exp_avg_sqs[-1].data_ptr(),
                       tuple(g.data_ptr() for g in grads))
                cached = self._tables.get(gi)
                if cached is None or cached[0] != sig:
                    table = hip.build_adam_table(
                        params, grads, exp_avgs, exp_avg_sqs)
                    cached = (sig, table, table.numel() // 40)  # 40 = sizeof(ChunkInfo)
                    self._tables[gi] = cached
                    if self._hyper_pin is None:
                        self._hyper_pin = torch.zeros(3, pin_memory=True)
                        self._hyper_dev = torch.zeros(
                            3, device=params[0].device)
                self._hyper_pin[0] = group["lr"]
                self._hyper_pin[1] = bias1
                self._hyper_pin[2] = bias2
                self._hyper_dev.copy_(self._hyper_pin, non_blocking=True)
                hip.fused_adam_table(
                    cached[1], cached[2], self._hyper_dev,
                    beta1, beta2, group["eps"], group["weight_decay"],
                )
            else:
                if group["weight_decay"] != 0.0:  # L2 (pre-moment, like TF)
                    grads = torch._foreach_add(grads, params,
                                               alpha=group["weight_decay"])
                torch._foreach_mul_(exp_avgs, beta1)
                torch._foreach_add_(exp_avgs, grads, alpha=1 - beta1)
                torch._foreach_mul_(exp_avg_sqs, beta2)
                torch._foreach_addcmul_(exp_avg_sqs, grads, grads,
                                        value=1 - beta2)
                step_size = group["lr"] / bias1
                denom = torch._foreach_sqrt(exp_avg_sqs)
                torch._foreach_div_(denom, math.sqrt(bias2))
                torch._foreach_add_(denom, group["eps"])
                torch._foreach_addcdiv_(params, exp_avgs, denom,
                                        value=-step_size)
        return loss
