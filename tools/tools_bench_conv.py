"""Microbench: time leaky_conv1d / leaky_convtranspose1d on the VITS hot
shapes; prints TFLOP/s and GB/s per shape."""
import sys, time, torch
sys.path.insert(0, '.')
from sonata_amd.ops.functional import leaky_conv1d, leaky_convtranspose1d

B = 64
F = 256
shapes = [
    # (name, Cin, Cout, T, k, dil, pre)
    ("res@256 k3d1", 256, 256, 8*F, 3, 1, 0.1),
    ("res@256 k11d5", 256, 256, 8*F, 11, 5, 0.1),
    ("res@128 k3d1", 128, 128, 64*F, 3, 1, 0.1),
    ("res@64 k3d1", 64, 64, 128*F, 3, 1, 0.1),
    ("res@32 k3d1", 32, 32, 256*F, 3, 1, 0.1),
    ("conv_pre k7", 192, 512, F, 7, 1, 0.0),
    ("wn k5 192->384", 192, 384, F, 5, 1, 0.0),
    ("proj1x1 192", 192, 192, F, 1, 1, 0.0),
]
dev = "cuda:0"
for name, Cin, Cout, T, k, dil, pre in shapes:
    x = (torch.randn(B, Cin, T)/4).to(torch.bfloat16).to(dev)
    w = (torch.randn(Cout, Cin, k)/(Cin*k)**0.5).to(torch.bfloat16).to(dev)
    bias = (torch.randn(Cout)/10).to(dev)
    pad = (k-1)*dil//2
    for _ in range(3):
        y = leaky_conv1d(x, w, bias, padding=pad, dilation=dil, pre_lrelu=pre)
    torch.cuda.synchronize()
    t0 = time.perf_counter(); N = 10
    for _ in range(N):
        y = leaky_conv1d(x, w, bias, padding=pad, dilation=dil, pre_lrelu=pre)
    torch.cuda.synchronize()
    dt = (time.perf_counter()-t0)/N
    fl = 2*B*Cin*Cout*k*T
    gb = 2*B*T*(Cin+Cout)  # bf16 in+out
    print(f"{name:18s} T={T:6d} {dt*1e3:8.3f} ms  {fl/dt/1e12:7.1f} TF  {gb/dt/1e9:7.0f} GB/s")

# convT shapes
for name, Cin, Cout, k, s, T in [("up0 512->256", 512, 256, 16, 8, F),
                                 ("up1 256->128", 256, 128, 16, 8, 8*F),
                                 ("up3 64->32", 64, 32, 4, 2, 128*F)]:
    x = (torch.randn(B, Cin, T)/4).to(torch.bfloat16).to(dev)
    w = (torch.randn(Cin, Cout, k)/(Cin*k)**0.5).to(torch.bfloat16).to(dev)
    bias = (torch.randn(Cout)/10).to(dev)
    for _ in range(3):
        y = leaky_convtranspose1d(x, w, bias, s, (k-s)//2, pre_lrelu=0.1)
    torch.cuda.synchronize()
    t0 = time.perf_counter(); N = 10
    for _ in range(N):
        y = leaky_convtranspose1d(x, w, bias, s, (k-s)//2, pre_lrelu=0.1)
    torch.cuda.synchronize()
    dt = (time.perf_counter()-t0)/N
    fl = 2*B*Cin*Cout*(k//s)*T*s
    gb = 2*B*(T*Cin + T*s*Cout)
    print(f"{name:18s} T={T:6d} {dt*1e3:8.3f} ms  {fl/dt/1e12:7.1f} TF  {gb/dt/1e9:7.0f} GB/s")

# ---- channel-last kernels + T-sweep diagnostics --------------------------
from sonata_amd.ops.functional import leaky_conv1d_cl, leaky_convtranspose1d_cl

print("--- channel-last ---")
for name, Cin, Cout, T, k, dil, pre in shapes:
    x = (torch.randn(B, T, Cin)/4).to(torch.bfloat16).to(dev)
    w = (torch.randn(Cout, Cin, k)/(Cin*k)**0.5).to(torch.bfloat16).to(dev)
    bias = (torch.randn(Cout)/10).to(dev)
    pad = (k-1)*dil//2
    for _ in range(3):
        y = leaky_conv1d_cl(x, w, bias, padding=pad, dilation=dil, pre_lrelu=pre)
    torch.cuda.synchronize()
    t0 = time.perf_counter(); N = 10
    for _ in range(N):
        y = leaky_conv1d_cl(x, w, bias, padding=pad, dilation=dil, pre_lrelu=pre)
    torch.cuda.synchronize()
    dt = (time.perf_counter()-t0)/N
    fl = 2*B*Cin*Cout*k*T
    gb = 2*B*T*(Cin+Cout)
    print(f"cl {name:15s} T={T:6d} {dt*1e3:8.3f} ms  {fl/dt/1e12:7.1f} TF  {gb/dt/1e9:7.0f} GB/s")

for name, Cin, Cout, k, s, T in [("up0 512->256", 512, 256, 16, 8, F),
                                 ("up1 256->128", 256, 128, 16, 8, 8*F),
                                 ("up2 128->64", 128, 64, 4, 2, 64*F),
                                 ("up3 64->32", 64, 32, 4, 2, 128*F)]:
    x = (torch.randn(B, T, Cin)/4).to(torch.bfloat16).to(dev)
    w = (torch.randn(Cin, Cout, k)/(Cin*k)**0.5).to(torch.bfloat16).to(dev)
    bias = (torch.randn(Cout)/10).to(dev)
    for _ in range(3):
        y = leaky_convtranspose1d_cl(x, w, bias, s, (k-s)//2, pre_lrelu=0.1)
    torch.cuda.synchronize()
    t0 = time.perf_counter(); N = 10
    for _ in range(N):
        y = leaky_convtranspose1d_cl(x, w, bias, s, (k-s)//2, pre_lrelu=0.1)
    torch.cuda.synchronize()
    dt = (time.perf_counter()-t0)/N
    fl = 2*B*Cin*Cout*(k//s)*T*s
    gb = 2*B*(T*Cin + T*s*Cout)
    print(f"cl {name:15s} T={T:6d} {dt*1e3:8.3f} ms  {fl/dt/1e12:7.1f} TF  {gb/dt/1e9:7.0f} GB/s")

# channel-FIRST convT T-sweep: localize the up1 anomaly (34 TF at T=2048)
print("--- convT chfirst T-sweep (Cin=256 Cout=128 k16 s8) ---")
for T in [256, 512, 1024, 2048]:
    x = (torch.randn(B, 256, T)/4).to(torch.bfloat16).to(dev)
    w = (torch.randn(256, 128, 16)/(256*16)**0.5).to(torch.bfloat16).to(dev)
    bias = (torch.randn(128)/10).to(dev)
    for _ in range(3):
        y = leaky_convtranspose1d(x, w, bias, 8, 4, pre_lrelu=0.1)
    torch.cuda.synchronize()
    t0 = time.perf_counter(); N = 10
    for _ in range(N):
        y = leaky_convtranspose1d(x, w, bias, 8, 4, pre_lrelu=0.1)
    torch.cuda.synchronize()
    dt = (time.perf_counter()-t0)/N
    fl = 2*B*256*128*2*T*8
    print(f"cf up1 T={T:6d} {dt*1e3:8.3f} ms  {fl/dt/1e12:7.1f} TF")

# up1 flakiness probe: same shape, fresh tensors each trial
print("--- cl up1 flakiness probe (5 fresh-tensor trials) ---")
for trial in range(5):
    x = (torch.randn(B, 2048, 256)/4).to(torch.bfloat16).to(dev)
    w = (torch.randn(256, 128, 16)/(256*16)**0.5).to(torch.bfloat16).to(dev)
    bias = (torch.randn(128)/10).to(dev)
    for _ in range(3):
        y = leaky_convtranspose1d_cl(x, w, bias, 8, 4, pre_lrelu=0.1)
    torch.cuda.synchronize()
    t0 = time.perf_counter(); N = 10
    for _ in range(N):
        y = leaky_convtranspose1d_cl(x, w, bias, 8, 4, pre_lrelu=0.1)
    torch.cuda.synchronize()
    dt = (time.perf_counter()-t0)/N
    fl = 2*B*256*128*2*2048*8
    print(f"trial {trial}: {dt*1e3:8.3f} ms  {fl/dt/1e12:7.1f} TF  x_ptr={x.data_ptr()%(1<<21)} w_ptr={w.data_ptr()%(1<<21)}")
    del x, w, bias, y

# fused resblock pair vs two convs
from sonata_amd.ops.functional import resblock_pair_cl
print("--- fused resblock pair (k3d1 + k3d1 equivalents) ---")
for C, T, k, d in [(256, 8*F, 3, 1), (128, 64*F, 3, 1), (64, 128*F, 3, 1),
                   (32, 256*F, 3, 1), (128, 64*F, 11, 5), (32, 256*F, 11, 5)]:
    x = (torch.randn(B, T, C)/4).to(torch.bfloat16).to(dev)
    w1 = (torch.randn(C, C, k)/(C*k)**0.5).to(torch.bfloat16).to(dev)
    w2 = (torch.randn(C, C, k)/(C*k)**0.5).to(torch.bfloat16).to(dev)
    b1 = (torch.randn(C)/10).to(dev); b2 = (torch.randn(C)/10).to(dev)
    for _ in range(3):
        y = resblock_pair_cl(x, w1, b1, w2, b2, dilation=d)
    torch.cuda.synchronize()
    t0 = time.perf_counter(); N = 10
    for _ in range(N):
        y = resblock_pair_cl(x, w1, b1, w2, b2, dilation=d)
    torch.cuda.synchronize()
    dt = (time.perf_counter()-t0)/N
    fl = 2*2*B*C*C*k*T
    gb = 2*B*T*(2*C + C)  # x read(+resid reread), out write
    print(f"rbpair C={C:4d} k{k}d{d} T={T:6d} {dt*1e3:8.3f} ms  {fl/dt/1e12:7.1f} TF  {gb/dt/1e9:7.0f} GB/s")

# direct (zero-LDS) conv vs staged cl conv
from sonata_amd.ops import hip_ext
from sonata_amd.ops.functional import _conv_weight_mfma, _bias_f32
ext = hip_ext(required=True)
print("--- direct vs staged cl conv ---")
for name, Cin, Cout, T, k, dil in [("res128 k3", 128, 128, 16384, 3, 1),
                                   ("res128 k11d5", 128, 128, 16384, 11, 5),
                                   ("res64 k3", 64, 64, 32768, 3, 1),
                                   ("res32 k3", 32, 32, 65536, 3, 1),
                                   ("res32 k11d5", 32, 32, 65536, 11, 5),
                                   ("res256 k3", 256, 256, 2048, 3, 1)]:
    x = (torch.randn(B, T, Cin)/4).to(torch.bfloat16).to(dev)
    w = (torch.randn(Cout, Cin, k)/(Cin*k)**0.5).to(torch.bfloat16).to(dev)
    bias = (torch.randn(Cout)/10).to(dev)
    pad = (k-1)*dil//2
    wp = _conv_weight_mfma(w); bf = _bias_f32(bias)
    ref = leaky_conv1d_cl(x, w, bias, padding=pad, dilation=dil, pre_lrelu=0.1)
    got = ext.conv1d_direct_cl(x, wp, bf, Cout, k, pad, dil, 0.1, 0, 0.0, None, None)
    err = (got.float()-ref.float()).abs().max().item()/max(ref.float().abs().max().item(),1e-6)
    for fn, tag in [(lambda: leaky_conv1d_cl(x, w, bias, padding=pad, dilation=dil, pre_lrelu=0.1), "staged"),
                    (lambda: ext.conv1d_direct_cl(x, wp, bf, Cout, k, pad, dil, 0.1, 0, 0.0, None, None), "direct")]:
        for _ in range(3): y = fn()
        torch.cuda.synchronize()
        t0 = time.perf_counter(); N = 10
        for _ in range(N): y = fn()
        torch.cuda.synchronize()
        dt = (time.perf_counter()-t0)/N
        fl = 2*B*Cin*Cout*k*T
        print(f"{tag} {name:14s} {dt*1e3:8.3f} ms  {fl/dt/1e12:7.1f} TF" + (f"  err={err:.4f}" if tag=="direct" else ""))

# hybrid: A staged, W direct-from-L2
print("--- hybrid W-direct vs staged cl conv ---")
for name, Cin, Cout, T, k, dil in [("res128 k3", 128, 128, 16384, 3, 1),
                                   ("res128 k11d5", 128, 128, 16384, 11, 5),
                                   ("res64 k3", 64, 64, 32768, 3, 1),
                                   ("res32 k3", 32, 32, 65536, 3, 1),
                                   ("res32 k11d5", 32, 32, 65536, 11, 5),
                                   ("res256 k3", 256, 256, 2048, 3, 1)]:
    x = (torch.randn(B, T, Cin)/4).to(torch.bfloat16).to(dev)
    w = (torch.randn(Cout, Cin, k)/(Cin*k)**0.5).to(torch.bfloat16).to(dev)
    bias = (torch.randn(Cout)/10).to(dev)
    pad = (k-1)*dil//2
    wp = _conv_weight_mfma(w); bf = _bias_f32(bias)
    ref = leaky_conv1d_cl(x, w, bias, padding=pad, dilation=dil, pre_lrelu=0.1)
    got = ext.conv1d_cl_wdirect(x, wp, bf, Cout, k, pad, dil, 0.1, 0, 0.0, None, None)
    err = (got.float()-ref.float()).abs().max().item()/max(ref.float().abs().max().item(),1e-6)
    for fn, tag in [(lambda: leaky_conv1d_cl(x, w, bias, padding=pad, dilation=dil, pre_lrelu=0.1), "staged"),
                    (lambda: ext.conv1d_cl_wdirect(x, wp, bf, Cout, k, pad, dil, 0.1, 0, 0.0, None, None), "wdirect")]:
        for _ in range(3): y = fn()
        torch.cuda.synchronize()
        t0 = time.perf_counter(); N = 10
        for _ in range(N): y = fn()
        torch.cuda.synchronize()
        dt = (time.perf_counter()-t0)/N
        fl = 2*B*Cin*Cout*k*T
        print(f"{tag:8s} {name:14s} {dt*1e3:8.3f} ms  {fl/dt/1e12:7.1f} TF" + (f"  err={err:.4f}" if tag=="wdirect" else ""))
