import sys, os, torch
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import bench as B
from split_learning_amd.ops import functional as hf

mode = sys.argv[1] if len(sys.argv) > 1 else "full"   # full | nobwd | noopt
trials = int(sys.argv[2]) if len(sys.argv) > 2 else 1500
dev = torch.device("cuda:0")
torch.manual_seed(0)

s1m, s1o = B.build_stage([0, B.CUT], dev)
s2m, s2o = B.build_stage([B.CUT, -1], dev)
x_buf = torch.randn(B.BATCH, 3, 32, 32, device=dev)
y_buf = torch.randint(0, 10, (B.BATCH,), device=dev)

holder = {}
def step():
    out1 = s1m(x_buf)
    act = out1.detach().requires_grad_(True)
    logits = s2m(act)
    loss = hf.cross_entropy(logits, y_buf)
    if mode != "nobwd":
        loss.backward()
        if mode == "full":
            s2o.step()
        out1.backward(gradient=act.grad)
        if mode == "full":
            s1o.step()
    holder["loss"] = loss
    holder["act"] = act
    holder["out1"] = out1

# warmup
s = torch.cuda.Stream(); s.wait_stream(torch.cuda.current_stream())
with torch.cuda.stream(s):
    for _ in range(3):
        step()
torch.cuda.current_stream().wait_stream(s)
g = torch.cuda.CUDAGraph()
with torch.cuda.graph(g):
    step()
print("captured", flush=True)

params1 = [(n, p) for n, p in s1m.named_parameters()]
for i in range(trials):
    if mode != "full":
        for _, p in params1:
            if p.grad is not None: p.grad.zero_()
        for p in s2m.parameters():
            if p.grad is not None: p.grad.zero_()
    g.replay()
    if i % 25 == 24 or i == trials - 1:
        torch.cuda.synchronize()
        bad = []
        if torch.isnan(holder["loss"]).any(): bad.append("loss")
        if holder["act"].grad is not None and torch.isnan(holder["act"].grad).any():
            bad.append("act.grad")
        for n, p in params1:
            if torch.isnan(p).any(): bad.append(f"w:{n}")
            if p.grad is not None and torch.isnan(p.grad).any(): bad.append(f"g:{n}")
        if bad:
            print(f"[{mode}] NaN at replay<={i}: {bad[:8]}", flush=True)
            sys.exit(1)
print(f"[{mode}] clean after {trials} replays", flush=True)
