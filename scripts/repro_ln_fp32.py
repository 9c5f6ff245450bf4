import torch, sys
sys.path.insert(0, "/root/repo")
from turboprune_amd.ops.norm_act import FusedLayerNorm
torch.manual_seed(0)
ln = FusedLayerNorm(384, eps=1e-6).to("cuda:0")
x = torch.randn(256, 197, 384, device="cuda:0", requires_grad=True)  # fp32
y = ln(x)
dy = torch.randn_like(y)
y.backward(dy)
torch.cuda.synchronize()
ref = torch.nn.LayerNorm(384, eps=1e-6).to("cuda:0")
ref.load_state_dict(ln.state_dict())
x2 = x.detach().requires_grad_()
y_ref = ref(x2); y_ref.backward(dy)
print("y", (y - y_ref).abs().max().item())
print("dx", (x.grad - x2.grad).abs().max().item())
print("dg", (ln.weight.grad - ref.weight.grad).abs().max().item())
