import sys, torch
import torch.nn.functional as F
sys.path.insert(0,'/root/repo')
from fedtorch_amd import ops
CL = torch.channels_last
empty = torch.empty(0, device='cuda')
torch.manual_seed(0)
C,H,W,N = 32,16,16,256
x = torch.randn(N,C,H,W,device='cuda').to(memory_format=CL).bfloat16()
w = (torch.randn(C,C,3,3,device='cuda')/(3*C)**0.5).to(memory_format=CL).bfloat16()
a = torch.rand(C,device='cuda')+0.5
b = torch.randn(C,device='cuda')*0.1
r = torch.randn(N,C,H,W,device='cuda').to(memory_format=CL).bfloat16()
xt32 = (a.view(1,C,1,1)*x.float()+b.view(1,C,1,1)+r.float()).relu()
xt = xt32.bfloat16()
ref2 = F.conv2d(xt.float(), w.float(), None,1,1)
y2 = ops._C.conv3x3_bn_fwd(x, w, empty, a, b, r, True)
d = (y2.float()-ref2).abs()
i = d.argmax()
n_,c_,h_,w_ = torch.unravel_index(i, d.shape)
print('max err %.4f at n=%d c=%d h=%d w=%d (H=%d W=%d)' % (d.max(), n_,c_,h_,w_,H,W))
print('err at borders mean %.5f, interior mean %.5f' % (
    d[:,:, [0,H-1],:].mean(), d[:,:,1:H-1,1:W-1].mean()))
# check kernel transform vs torch: run conv with w=identity-ish? use stats-free compare of xt
# cast-mode probe: does (bf16)(float) truncate?
t = torch.tensor([1.0009765625], device='cuda')  # halfway between bf16 steps
print('torch bf16 of 1.0009765625 ->', t.bfloat16().float().item())
# row0 pattern: compare a column strip
print('per-row mean err:', [float(d[:,:,hh,:].mean()) for hh in range(H)])
