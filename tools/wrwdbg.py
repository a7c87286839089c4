import sys, os
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from tensorflowonspark_amd.ops import get_ext
ext = get_ext(required=True)
def mk(t): return t.cuda().to(torch.bfloat16).contiguous(memory_format=torch.channels_last)

for H in (10, 16, 20):
    N, C = 1, 64
    x = mk(torch.ones(N, C, H, H))
    dy = mk(torch.ones(N, C, H, H))
    dw = ext.conv_wrw2(dy, x, 3, 3, 1, 1).cpu()   # counts per tap
    print(f"H={H} M={H*H}:")
    bad = False
    for tap in range(9):
        r, s = tap // 3, tap % 3
        oh_n = H - (1 if r != 1 else 0)
        ow_n = H - (1 if s != 1 else 0)
        want = oh_n * ow_n
        col = dw[:, tap * C:(tap + 1) * C]
        mn, mx = col.min().item(), col.max().item()
        if abs(mn - want) > 0.5 or abs(mx - want) > 0.5:
            print(f"  tap{tap} (r{r}s{s}): want {want} got range [{mn}, {mx}]")
            bad = True
    if not bad:
        print("  all taps exact")
