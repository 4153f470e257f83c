"""FusedSGD: SGD with momentum/weight-decay/update fused into one CDNA4
kernel launch over all parameter buckets (csrc/sgd_kernels.hip).

Torch-eager SGD issues 3-4 kernels per parameter (161 params x 4 = ~600
launches per ResNet-50 step); FusedSGD issues ceil(161/48) = 4.  Combine
with hvd.DistributedOptimizer exactly like torch.optim.SGD.

NOTE: the fused kernel path covers CUDA fp32 parameters only.  CPU or
non-fp32 params (e.g. bf16 master weights) take a correct eager fallback
and a one-time warning is emitted so a benchmarked configuration cannot
silently lose the fusion.
"""
import warnings
import torch

from horovod_amd import _core


class FusedSGD(torch.optim.Optimizer):
    def __init__(self, params, lr=1e-3, momentum=0.0, weight_decay=0.0,
                 dampening=0.0, nesterov=False):
        if nesterov and (momentum <= 0 or dampening != 0):
            raise ValueError("nesterov requires momentum > 0, dampening 0")
        defaults = dict(lr=lr, momentum=momentum, weight_decay=weight_decay,
                        dampening=dampening, nesterov=nesterov)
        super().__init__(params, defaults)

    @torch.no_grad()
    def step(self, closure=None):
        loss = None
        if closure is not None:
            with torch.enable_grad():
                loss = closure()
        for group in self.param_groups:
            params, grads, momenta = [], [], []
            use_momentum = group["momentum"] != 0
            for p in group["params"]:
                if p.grad is None:
                    continue
                if not p.is_cuda or p.dtype != torch.float32:
                    # CPU / non-fp32 fallback: eager update (warn once)
                    if not getattr(self, "_warned_eager_fallback", False):
                        self._warned_eager_fallback = True
                        warnings.warn(
                            "FusedSGD: parameter is not CUDA fp32; using the "
                            "eager (unfused) update for such parameters")
                    self._eager_update(p, group)
                    continue
                params.append(p)
                grads.append(p.grad)
                if use_momentum:
                    st = self.state.setdefault(p, {})
                    if "momentum_buffer" not in st:
                        st["momentum_buffer"] = torch.zeros_like(p)
                    momenta.append(st["momentum_buffer"])
            if params:
                _core.fused_sgd_step(params, grads,
                                     momenta if use_momentum else [],
                                     group["lr"], group["momentum"],
                                     group["weight_decay"],
                                     group["dampening"], group["nesterov"])
        return loss

    def _eager_update(self, p, group):
        g = p.grad
        if group["weight_decay"]:
            g = g.add(p, alpha=group["weight_decay"])
        if group["momentum"]:
            st = self.state.setdefault(p, {})
            buf = st.get("momentum_buffer")
            if buf is None:
                buf = st["momentum_buffer"] = torch.clone(g).detach()
            else:
                buf.mul_(group["momentum"]).add_(g, alpha=1 - group["dampening"])
            g = g.add(buf, alpha=group["momentum"]) if group["nesterov"] else buf
        p.add_(g, alpha=-group["lr"])
