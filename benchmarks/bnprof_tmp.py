import sys, torch
sys.path.insert(0, "/root/repo")
from torchmpi_amd.ops.batchnorm import FusedBatchNorm2d
dev = torch.device("cuda", 0)
for shape in [(256, 64, 112, 112), (256, 2048, 7, 7)]:
    x = (torch.randn(shape, device=dev, dtype=torch.bfloat16)
         .contiguous(memory_format=torch.channels_last).requires_grad_(True))
    g = torch.randn_like(x)
    m = FusedBatchNorm2d(shape[1]).to(dev)
    for _ in range(10):
        y = m(x); y.backward(g); x.grad = None
torch.cuda.synchronize()
print("done")
