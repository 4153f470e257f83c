"""FusedAdamW: AdamW with moments + decoupled decay + update fused into one
CDNA4 kernel launch per 48-tensor bucket (csrc/adamw_kernels.hip).

Torch-eager AdamW issues ~8 kernels per parameter; a BERT-large step spends
several ms in them.  Drop-in for torch.optim.AdamW (state_dict-compatible);
CPU or non-fp32 parameters take a correct eager fallback with a one-time
warning, like FusedSGD.
"""
import warnings

import torch

from horovod_amd import _core


class FusedAdamW(torch.optim.Optimizer):
    def __init__(self, params, lr=1e-3, betas=(0.9, 0.999), eps=1e-8,
                 weight_decay=1e-2):
        if not 0.0 <= betas[0] < 1.0 or not 0.0 <= betas[1] < 1.0:
            raise ValueError(f"invalid betas: {betas}")
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
            # bucket by bias-correction step: the fused launch applies ONE
            # step factor, and params restored mid-run can disagree
            by_step = {}
            for p in group["params"]:
                if p.grad is None:
                    continue
                st = self.state.setdefault(p, {})
                if "step" not in st:
                    st["step"] = 0
                    st["exp_avg"] = torch.zeros_like(p)
                    st["exp_avg_sq"] = torch.zeros_like(p)
                st["step"] += 1
                if not p.is_cuda or p.dtype != torch.float32 or \
                        p.grad.is_sparse:
                    if not getattr(self, "_warned_eager_fallback", False):
                        self._warned_eager_fallback = True
                        warnings.warn(
                            "FusedAdamW: parameter is not CUDA fp32; using "
                            "the eager (unfused) update for such parameters")
                    self._eager_update(p, group, st)
                    continue
                b = by_step.setdefault(st["step"], ([], [], [], []))
                b[0].append(p)
                b[1].append(p.grad)
                b[2].append(st["exp_avg"])
                b[3].append(st["exp_avg_sq"])
            b1, b2 = group["betas"]
            for step_t, (params, grads, avgs, sqs) in by_step.items():
                _core.fused_adamw_step(params, grads, avgs, sqs,
                                       group["lr"], b1, b2, group["eps"],
                                       group["weight_decay"], step_t)
        return loss

    def _eager_update(self, p, group, st):
        b1, b2 = group["betas"]
        g = p.grad
        if g.is_sparse:
            g = g.to_dense()
        p.mul_(1 - group["lr"] * group["weight_decay"])
        st["exp_avg"].mul_(b1).add_(g, alpha=1 - b1)
        st["exp_avg_sq"].mul_(b2).addcmul_(g, g, value=1 - b2)
        bc1 = 1 - b1 ** st["step"]
        bc2 = 1 - b2 ** st["step"]
        denom = (st["exp_avg_sq"].sqrt() / (bc2 ** 0.5)).add_(group["eps"])
        p.addcdiv_(st["exp_avg"], denom, value=-group["lr"] / bc1)
