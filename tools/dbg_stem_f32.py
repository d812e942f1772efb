import sys
import torch
import torch.nn.functional as F
sys.path.insert(0, ".")
from dist_mnist_amd import ops

DEV = "cuda:0"
g = torch.Generator().manual_seed(92)
x = torch.randn(16, 1, 28, 28, generator=g).to(DEV)
w = (torch.randn(10, 1, 5, 5, generator=g) * 0.3).to(DEV).requires_grad_(True)
b = torch.randn(10, generator=g).to(DEV).requires_grad_(True)

p = ops.conv_pool_stem(x, w, b)
dp = torch.randn(16, 10, 12, 12, generator=g).to(DEV)
p.backward(dp)

wr = w.detach().clone().requires_grad_(True)
br = b.detach().clone().requires_grad_(True)
ref = F.relu(F.max_pool2d(F.conv2d(x, wr, br), 2))
ref.backward(dp)

def rep(name, a, r):
    d = (a - r).abs()
    print(f"{name}: max|d|={d.max().item():.3e} rel={d.max().item()/(r.abs().max().item()+1e-12):.3e} "
          f"mism={(d > 1e-4).sum().item()}/{d.numel()}")
    if d.max().item() > 1e-4:
        i = d.flatten().argmax().item()
        print("  worst idx", i, "got", a.flatten()[i].item(), "ref", r.flatten()[i].item())

rep("p ", p, ref)
rep("dw", w.grad, wr.grad)
rep("db", b.grad, br.grad)
# fwd-only pieces
conv_ref = F.conv2d(x, wr.detach(), br.detach())
pool_ref, idx_ref = F.max_pool2d(conv_ref, 2, return_indices=True)
print("pool_ref>0 frac", (pool_ref > 0).float().mean().item())

# tie-aware check: rebuild dy_conv from OUR p/idx routing, then dw via torch
import torch.nn.grad as G
pk, idxk = ops.native.ext().conv_pool_stem_f32(x, w.detach(), b.detach())
N, CO, OH2, OW2 = pk.shape
dyc = torch.zeros(N, CO, 24, 24, device=DEV)
q = (torch.arange(24, device=DEV)[:, None] % 2) * 2 + torch.arange(24, device=DEV)[None, :] % 2
qmap = q[None, None].expand(N, CO, 24, 24)
idx_up = idxk.repeat_interleave(2, 2).repeat_interleave(2, 3)
p_up = pk.repeat_interleave(2, 2).repeat_interleave(2, 3)
dp_up = dp.repeat_interleave(2, 2).repeat_interleave(2, 3)
take = (idx_up.long() == qmap) & (p_up > 0)
dyc[take] = dp_up[take]
dw2 = G.conv2d_weight(x, w.shape, dyc)
d = (w.grad - dw2).abs()
print(f"dw vs OUR-routing ref: max|d|={d.max().item():.3e} mism={(d>1e-4).sum().item()}/250")
