import sys, torch
sys.path.insert(0, "/root/repo")
from oktopk_amd import _hip_ops
torch.manual_seed(0)
# fused attention at BERT-base shape
qkv = (torch.randn(8, 128, 3*12*64) * 0.5).bfloat16().cuda()
mask = torch.zeros(8, 128).bfloat16().cuda()
for _ in range(20):
    _hip_ops.attn_fwd(qkv, mask, 12, 0.1, True, True)
# fused linear+gelu at BERT MLP shape
x = (torch.randn(1024, 768) * 0.5).bfloat16().cuda()
w = (torch.randn(3072, 768) * 0.5).bfloat16().cuda()
b = torch.randn(3072).bfloat16().cuda()
for _ in range(20):
    _hip_ops.linear_gelu(x, w, b, True, False)
torch.cuda.synchronize()
print("done")
