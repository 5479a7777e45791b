import sys, os
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from tensor2robot_amd.ops import fused_bn
def _bn_reference(x32, gamma, beta, eps, relu=True):
  mean = x32.mean(dim=0); var = x32.var(dim=0, unbiased=False)
  xhat = (x32 - mean) / torch.sqrt(var + eps)
  y = xhat * gamma + beta
  return torch.relu(y) if relu else y
torch.manual_seed(1)
M, C = 4096, 64
x32 = torch.randn(M, C, device="cuda").to(torch.bfloat16).float().requires_grad_(True)
gamma32 = (torch.rand(C, device="cuda") + 0.5).requires_grad_(True)
beta32 = torch.randn(C, device="cuda").requires_grad_(True)
ref = _bn_reference(x32, gamma32, beta32, 1e-3)
dy = torch.randn_like(ref)
ref.backward(dy)
x_bf = x32.detach().to(torch.bfloat16).requires_grad_(True)
gamma = gamma32.detach().clone().requires_grad_(True)
beta = beta32.detach().clone().requires_grad_(True)
y = fused_bn._FusedBNReLUFunction.apply(x_bf, gamma, beta, None, None, 1e-3, 0.003, True)
print("y diff max:", (y.float()-ref).abs().max().item())
y.backward(dy.to(torch.bfloat16))
print("gamma diff:", (gamma.grad-gamma32.grad).abs().max().item(), "norm", gamma32.grad.abs().max().item())
print("beta diff:", (beta.grad-beta32.grad).abs().max().item())
rel = (x_bf.grad.float() - x32.grad).abs().max() / x32.grad.abs().max().clamp(min=1e-6)
print("dx rel:", rel.item())
