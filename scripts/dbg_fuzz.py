import sys, os
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import os, torch
import melgan_amd._C as C

torch.manual_seed(0)
B, Cin, Cout, K, S, G, T = 2, 2, 2, 1, 4, 2, 67
x = torch.randn(B, Cin, T, device="cuda").to(torch.bfloat16)
w = (torch.randn(Cout, Cin//G, K, device="cuda") * 0.3).to(torch.bfloat16)
To = (T - 1) // S + 1
gz = torch.randn(B, Cout, To, device="cuda").to(torch.bfloat16)

# composite reference
xf = x.float().requires_grad_(False)
wf = w.float().requires_grad_(True)
y = torch.nn.functional.conv1d(xf, wf, stride=S, groups=G)
y.backward(gz.float())
print("ref gw:", wf.grad.flatten().tolist())

gw = C.conv1d_bwd_weight(gz, x, S, 1, 0, False, G, K, False, 0.0)
print("mfma gw:", gw.flatten().tolist())

os.environ["MELGAN_AMD_NO_MFMA"] = "1"
# fresh process would be needed for the static flag; emulate VALU via fp32
gw32 = C.conv1d_bwd_weight(gz.float(), x.float(), S, 1, 0, False, G, K, False, 0.0)
print("valu gw:", gw32.flatten().tolist())

# also check a few more fuzz-like shapes on the mfma16 path
for (cin, cout, k, s, g, t) in [(4,4,1,4,2,67),(2,2,1,1,2,67),(8,4,3,2,2,128),(2,2,1,4,2,64)]:
    x2 = torch.randn(2, cin, t, device="cuda").to(torch.bfloat16)
    w2 = (torch.randn(cout, cin//g, k, device="cuda")*0.3).to(torch.bfloat16)
    pad = 0
    to2 = (t + 2*pad - ((k-1)+1))//s + 1
    gz2 = torch.randn(2, cout, to2, device="cuda").to(torch.bfloat16)
    wf2 = w2.float().requires_grad_(True)
    y2 = torch.nn.functional.conv1d(x2.float(), wf2, stride=s, padding=pad, groups=g)
    y2.backward(gz2.float())
    gwk = C.conv1d_bwd_weight(gz2, x2, s, 1, pad, False, g, k, False, 0.0)
    rel = (gwk - wf2.grad).abs().max()/(wf2.grad.abs().max()+1e-6)
    print(f"case {cin}x{cout}k{k}s{s}g{g}t{t}: rel {float(rel):.4f}")
