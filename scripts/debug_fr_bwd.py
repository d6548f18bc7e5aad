"""Stage-by-stage FR backward comparison vs the bf16-faithful oracle."""
import os, sys
sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))
import torch
import torch.nn.functional as F
from torch.nn.grad import conv2d_input, conv2d_weight
from torchbeast_amd.ops import functional as tbf
import torchbeast_amd.ops as om

ext = om.require_ext()
bf = lambda x: x.to(torch.bfloat16).float()
N, shape = 6, (3, 210, 160)
torch.manual_seed(3)
c1 = torch.nn.Conv2d(3, 32, 8, stride=4).cuda()
c2 = torch.nn.Conv2d(32, 64, 4, stride=2).cuda()
c3 = torch.nn.Conv2d(64, 64, 3, stride=1).cuda()
frames = torch.randint(0, 256, (N, *shape), dtype=torch.uint8, device="cuda")
w1p, w2p, w3p = tbf._pack_trunk_weights(c1.weight, c2.weight, c3.weight)
out3, a1, a2 = ext.conv_trunk_fwd(frames, w1p, c1.bias.detach().contiguous(),
                                  w2p, c2.bias.detach().contiguous(), w3p,
                                  c3.bias.detach().contiguous(), True)
nfeat = out3.shape[1]
d_out = torch.randn(N, nfeat, device="cuda")

d3m = ext.conv_trunk_mask_d3(d_out.contiguous(), out3)
w3r = (c3.weight.detach().flip(2, 3).permute(1, 2, 3, 0)
       .reshape(64, -1).to(torch.bfloat16).contiguous())
w2r = (c2.weight.detach().flip(2, 3).permute(1, 2, 3, 0)
       .reshape(32, -1).to(torch.bfloat16).contiguous())
d2 = ext.conv_trunk_dgrad3(d3m, w3r, a2)
d1 = ext.conv_trunk_dgrad2(d2, w2r, a1)
dw1p, db1 = ext.conv_trunk_wgrad1(frames, d1)

# oracle chain
a1f = a1.permute(0, 3, 1, 2).float()
a2f = a2.permute(0, 3, 1, 2).float()
oh3, ow3 = a2.shape[1] - 2, a2.shape[2] - 2
d3_o = bf((d_out * (out3 > 0)).view(N, 64, oh3, ow3))
d2_o = bf(conv2d_input(a2f.shape, bf(c3.weight), d3_o) * (a2f > 0))
d1_o = bf(conv2d_input(a1f.shape, bf(c2.weight), d2_o, stride=2) * (a1f > 0))
dw1_o = conv2d_weight(bf(frames.float() / 255.0), c1.weight.shape, d1_o, stride=4)

def cmp(name, ours_nhwc, theirs_nchw):
    ours = ours_nhwc.permute(0, 3, 1, 2).float()
    err = (ours - theirs_nchw).abs().max() / theirs_nchw.abs().max().clamp_min(1e-5)
    print(f"{name}: rel-max {err.item():.5f}")

cmp("d3m", d3m, d3_o)
cmp("d2 ", d2, d2_o)
cmp("d1 ", d1, d1_o)
kwcp = dw1p.shape[-1]
dw1 = dw1p.view(8, 32, kwcp)[..., :24].reshape(8, 32, 3, 8).permute(1, 2, 0, 3)
err = (dw1 - dw1_o).abs().max() / dw1_o.abs().max()
print(f"dw1: rel-max {err.item():.5f}")
# also wgrad1 from the ORACLE's d1 (isolates wgrad from dgrad chain):
d1o_nhwc = d1_o.permute(0, 2, 3, 1).to(torch.bfloat16).contiguous()
dw1p2, _ = ext.conv_trunk_wgrad1(frames, d1o_nhwc)
dw1b = dw1p2.view(8, 32, kwcp)[..., :24].reshape(8, 32, 3, 8).permute(1, 2, 0, 3)
err = (dw1b - dw1_o).abs().max() / dw1_o.abs().max()
print(f"dw1(oracle d1): rel-max {err.item():.5f}")
