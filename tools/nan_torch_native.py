"""Control experiment: SAME capture/replay harness, but pure torch-native ops
(nn.Conv2d/BatchNorm2d/F.cross_entropy/torch.optim.SGD on aten/MIOpen kernels,
none of our HIP extension).  If the intermittent graph-replay divergence
reproduces here, the issue is platform-level (HIP graphs / torch-ROCm), not
the split_learning_amd kernels."""
import sys, torch
import torch.nn as nn

steps = int(sys.argv[1]) if len(sys.argv) > 1 else 1024
dev = torch.device("cuda:0")
torch.manual_seed(1234)

def make_vgg_head():  # stage-1-like: convs at 32x32
    return nn.Sequential(
        nn.Conv2d(3, 64, 3, 1, 1), nn.BatchNorm2d(64), nn.ReLU(),
        nn.Conv2d(64, 64, 3, 1, 1), nn.BatchNorm2d(64), nn.ReLU(),
        nn.MaxPool2d(2, 2)).to(dev).train()

def make_vgg_tail():
    layers = []
    c_in = 64
    for block in [[128, 128], [256, 256, 256], [512, 512, 512], [512, 512, 512]]:
        for c in block:
            layers += [nn.Conv2d(c_in, c, 3, 1, 1), nn.BatchNorm2d(c), nn.ReLU()]
            c_in = c
        layers.append(nn.MaxPool2d(2, 2))
    layers += [nn.Flatten(), nn.Dropout(0.5), nn.Linear(512, 4096), nn.ReLU(),
               nn.Dropout(0.5), nn.Linear(4096, 4096), nn.ReLU(),
               nn.Linear(4096, 10)]
    return nn.Sequential(*layers).to(dev).train()

s1, s2 = make_vgg_head(), make_vgg_tail()
o1 = torch.optim.SGD(s1.parameters(), lr=5e-4, momentum=0.5)
o2 = torch.optim.SGD(s2.parameters(), lr=5e-4, momentum=0.5)
x_buf = torch.zeros(32, 3, 32, 32, device=dev)
y_buf = torch.zeros(32, dtype=torch.int64, device=dev)
nan_flag = torch.zeros((), dtype=torch.bool, device=dev)

def step():
    o1.zero_grad(set_to_none=False)
    o2.zero_grad(set_to_none=False)
    out1 = s1(x_buf)
    act = out1.detach().requires_grad_(True)
    loss = nn.functional.cross_entropy(s2(act), y_buf)
    nan_flag.copy_(nan_flag | torch.isnan(loss))
    loss.backward()
    o2.step()
    out1.backward(gradient=act.grad)
    o1.step()

s = torch.cuda.Stream(); s.wait_stream(torch.cuda.current_stream())
with torch.cuda.stream(s):
    for _ in range(3):
        step()
torch.cuda.current_stream().wait_stream(s)
g = torch.cuda.CUDAGraph()
with torch.cuda.graph(g):
    step()
print("captured", flush=True)

gcpu = torch.Generator().manual_seed(5)
proj = torch.randn(3072, 10, generator=gcpu).to(dev)
xs = torch.randn(256, 32, 3, 32, 32, generator=gcpu).to(dev)
ys = (xs.reshape(256 * 32, -1) @ proj).argmax(-1).reshape(256, 32)
bad = 0
for i in range(steps):
    x_buf.copy_(xs[i % 256]); y_buf.copy_(ys[i % 256])
    g.replay()
    if i % 128 == 127:
        torch.cuda.synchronize()
        if bool(nan_flag.item()):
            bad += 1
            nan_flag.zero_()
print(f"torch-native graph: {steps} replays, NaN windows={bad}", flush=True)
