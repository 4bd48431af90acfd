"""Stage-by-stage parity debug of the channel-last generator path."""
import sys, torch
sys.path.insert(0, '.')
from sonata_amd.models.config import QUALITY_PRESETS, VitsArchitecture
from sonata_amd.models.vits import Generator, LRELU_SLOPE
from sonata_amd.ops.functional import (leaky_conv1d_cl,
                                       leaky_convtranspose1d_cl)

dev = "cuda:0"


def rel(a, b):
    a = a.float().cpu(); b = b.float().cpu()
    return float((a - b).abs().max() / b.abs().max().clamp_min(1e-6))


# individual 16-channel shapes
torch.manual_seed(3)
for Cin, Cout, k, dil in [(16, 16, 3, 1), (16, 16, 11, 5), (16, 1, 7, 1)]:
    x = (torch.randn(2, 99, Cin) / 4).to(torch.bfloat16)
    w = (torch.randn(Cout, Cin, k) / (Cin * k) ** 0.5).to(torch.bfloat16)
    b = torch.randn(Cout) / 10
    pad = (k - 1) * dil // 2
    got = leaky_conv1d_cl(x.to(dev), w.to(dev), b.to(dev), padding=pad,
                          dilation=dil, pre_lrelu=0.1)
    ref = leaky_conv1d_cl(x.float(), w.float(), b, padding=pad, dilation=dil,
                          pre_lrelu=0.1)
    print(f"conv {Cin}->{Cout} k{k}d{dil}: {rel(got, ref):.4f}")

for Cin, Cout, k, s in [(32, 16, 4, 2), (64, 32, 4, 2)]:
    x = (torch.randn(2, 99, Cin) / 4).to(torch.bfloat16)
    w = (torch.randn(Cin, Cout, k) / (Cin * k) ** 0.5).to(torch.bfloat16)
    b = torch.randn(Cout) / 10
    got = leaky_convtranspose1d_cl(x.to(dev), w.to(dev), b.to(dev), s,
                                   (k - s) // 2, pre_lrelu=0.1)
    ref = leaky_convtranspose1d_cl(x.float(), w.float(), b, s, (k - s) // 2,
                                   pre_lrelu=0.1)
    print(f"convT {Cin}->{Cout} k{k}s{s}: {rel(got, ref):.4f}")

# stage-by-stage generator
torch.manual_seed(11)
arch = VitsArchitecture(**QUALITY_PRESETS["x_low"]["arch"])
gen = Generator(arch).eval()
geng = Generator(arch).eval()
geng.load_state_dict(gen.state_dict())
geng = geng.to(dev, torch.bfloat16)
B, F = 3, 61
z = torch.randn(B, arch.inter_channels, F) / 2
lens = torch.tensor([F, 40, 23])

with torch.no_grad():
    # oracle channel-first fp32
    xr = z.float()
    from sonata_amd.ops.functional import conv_mod, mask_tail_
    xr = conv_mod(gen.conv_pre, xr)
    lr = lens.clone()
    mask_tail_(xr, lr)
    # cl path
    xg = z.to(dev, torch.bfloat16).transpose(1, 2).contiguous()
    lg = lens.to(dev)
    xg = leaky_conv1d_cl(xg, geng.conv_pre.weight, geng.conv_pre.bias,
                         padding=3, out_lens=lg)
    print("after conv_pre:", rel(xg.transpose(1, 2), xr))
    from sonata_amd.ops.functional import leaky_conv1d, leaky_convtranspose1d
    for i, (upr, upg) in enumerate(zip(gen.ups, geng.ups)):
        import torch.nn.functional as Fnn
        xr = Fnn.conv_transpose1d(Fnn.leaky_relu(xr, LRELU_SLOPE), upr.weight,
                                  upr.bias, stride=upr.stride[0],
                                  padding=upr.padding[0])
        lr = lr * upr.stride[0]
        mask_tail_(xr, lr)
        lg = lg * upg.stride[0]
        xg = leaky_convtranspose1d_cl(xg, upg.weight, upg.bias,
                                      stride=upg.stride[0],
                                      padding=upg.padding[0],
                                      pre_lrelu=LRELU_SLOPE, out_lens=lg)
        print(f"after up{i}:", rel(xg.transpose(1, 2), xr))
        xsr = None
        xsg = None
        for j in range(gen.num_kernels):
            rbr = gen.resblocks[i * gen.num_kernels + j]
            outr = xr
            for c1, c2 in zip(rbr.convs1, rbr.convs2):
                xt = Fnn.conv1d(Fnn.leaky_relu(outr, LRELU_SLOPE), c1.weight,
                                c1.bias, padding=c1.padding[0],
                                dilation=c1.dilation[0])
                mask_tail_(xt, lr)
                outr = Fnn.conv1d(Fnn.leaky_relu(xt, LRELU_SLOPE), c2.weight,
                                  c2.bias, padding=c2.padding[0]) + outr
                mask_tail_(outr, lr)
            xsr = outr if xsr is None else xsr + outr
            outg = geng.resblocks[i * gen.num_kernels + j].forward_cl(xg, lg)
            xsg = outg if xsg is None else xsg + outg
            print(f"  rb{j}:", rel(outg.transpose(1, 2), outr))
        xr = xsr / gen.num_kernels
        xg = xsg / gen.num_kernels
    xr = torch.tanh(Fnn.conv1d(Fnn.leaky_relu(xr, LRELU_SLOPE),
                               gen.conv_post.weight, None, padding=3))
    xg = leaky_conv1d_cl(xg, geng.conv_post.weight, None, padding=3,
                         pre_lrelu=LRELU_SLOPE, post_tanh=True, out_lens=lg)
    print("after conv_post:", rel(xg.transpose(1, 2), xr))
