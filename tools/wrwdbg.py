import sys, os
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from tensorflowonspark_amd.ops import get_ext
ext = get_ext(required=True)

def mk(t): return t.cuda().to(torch.bfloat16).contiguous(memory_format=torch.channels_last)

# --- probe A: 1x1, single one-hot dy; dW row should equal x[m0,:] ---
N, Cin, H, W, Cout = 1, 64, 8, 8, 64
x = torch.arange(N*H*W*Cin, dtype=torch.float32).reshape(N, H, W, Cin) % 97
xc = mk(x.permute(0,3,1,2))
for m0, co0 in [(0,0), (1,0), (5,17), (37,63)]:
    dy = torch.zeros(N, H, W, Cout)
    dy[0, m0//W, m0%W, co0] = 1.0
    dyc = mk(dy.permute(0,3,1,2))
    dw = ext.conv_wrw2(dyc, xc, 1, 1, 1, 0).cpu()
    ref = x.reshape(-1, Cin)[m0]
    errs = (dw[co0] - ref).abs()
    other = dw[torch.arange(Cout) != co0].abs().max()
    print(f"m0={m0} co={co0}: maxerr={errs.max():.3f} nan={torch.isnan(dw).sum().item()} other_rows_max={other:.3f}")
    if errs.max() > 0.5:
        bad = (errs > 0.5).nonzero().flatten()
        print("   bad cins:", bad[:12].tolist())
        print("   got:", dw[co0][bad[:6]].tolist(), " want:", ref[bad[:6]].tolist())

# --- probe B: full-random small 1x1, error structure ---
torch.manual_seed(0)
x = torch.randn(2, 16, 16, 64); dy = torch.randn(2, 16, 16, 64)
dw = ext.conv_wrw2(mk(dy.permute(0,3,1,2)), mk(x.permute(0,3,1,2)), 1, 1, 1, 0).cpu()
ref = torch.einsum('mc,md->cd', dy.reshape(-1,64).to(torch.bfloat16).float(),
                   x.reshape(-1,64).to(torch.bfloat16).float())
err = (dw - ref).abs()
print("probe B: maxerr", err.max().item(), "nan", torch.isnan(dw).sum().item())
# error structure by 16x16 block
eb = err.reshape(4,16,4,16).amax(dim=(1,3))
print(eb)
