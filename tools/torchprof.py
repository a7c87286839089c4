import sys, os
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from tensorflowonspark_amd.models import resnet50
from tensorflowonspark_amd.ops.modules import BucketSGD, softmax_cross_entropy
from tensorflowonspark_amd.parallel import DDPEngine

model = resnet50().cuda().to(memory_format=torch.channels_last); model.train()
engine = DDPEngine(model); opt = BucketSGD(engine, lr=0.05, momentum=0.9)
x = torch.randn(1024, 3, 224, 224, device="cuda").to(torch.bfloat16).contiguous(memory_format=torch.channels_last)
y = torch.randint(0, 1000, (1024,), device="cuda")

def step():
    opt.zero_grad()
    with torch.autocast("cuda", dtype=torch.bfloat16):
        loss = softmax_cross_entropy(model(x), y)
    loss.backward(); engine.finalize_backward(); opt.step()

for _ in range(3): step()
torch.cuda.synchronize()
from torch.profiler import profile, ProfilerActivity
with profile(activities=[ProfilerActivity.CPU, ProfilerActivity.CUDA]) as prof:
    for _ in range(3): step()
    torch.cuda.synchronize()
print(prof.key_averages().table(sort_by="self_cuda_time_total", row_limit=28, max_name_column_width=60))
