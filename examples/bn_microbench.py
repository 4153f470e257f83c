"""Per-shape microbench: FusedBNReLU vs MIOpen BN + separate ReLU
(training fwd+bwd, bf16 channels_last, ResNet-50 shapes at batch 64)."""
import os, sys
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch, time
import torch.nn.functional as F
from horovod_amd.ops.fused_bn import FusedBNReLU

shapes = [(64,64,112,112),(64,256,56,56),(64,64,56,56),(64,512,28,28),
          (64,128,28,28),(64,1024,14,14),(64,256,14,14),(64,2048,7,7),
          (64,512,7,7)]
if os.environ.get("BN_SHAPES"):  # e.g. BN_SHAPES=64x64x112x112,64x256x56x56
    shapes = [tuple(int(v) for v in s.split("x"))
              for s in os.environ["BN_SHAPES"].split(",")]
ITERS = int(os.environ.get("BN_ITERS", "20"))
for (n,c,h,w) in shapes:
    x = torch.randn(n,c,h,w, device="cuda", dtype=torch.bfloat16) \
        .to(memory_format=torch.channels_last).requires_grad_(True)
    fused = FusedBNReLU(c).cuda().train()
    ref = torch.nn.BatchNorm2d(c).cuda().train()
    def run(mod, use_relu):
        y = mod(x) if not use_relu else F.relu(ref(x))
        y.backward(torch.ones_like(y))
        x.grad = None
    for label, fn in (("fused", lambda: run(fused, False)),
                      ("miopen+relu", lambda: run(None, True))):
        for _ in range(5): fn()
        torch.cuda.synchronize(); t0=time.perf_counter()
        for _ in range(ITERS): fn()
        torch.cuda.synchronize()
        print(f"{n}x{c}x{h}x{w} {label:12s} {(time.perf_counter()-t0)/ITERS*1e3:7.3f} ms")
