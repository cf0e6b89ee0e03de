import sys, os, torch
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import bench as B
dev = torch.device("cuda:0")
pipe = B.ColocatedPipeline(dev, use_graphs=False)
xs, ys = B.make_batches(dev, 768, seed=1)
for i in range(768):
    pipe.x_buf.copy_(xs[i]); pipe.y_buf.copy_(ys[i])
    pipe._step()
    if i % 64 == 0 or i == 767:
        # peek loss by recomputing forward on the same batch
        with torch.no_grad():
            act = pipe.s1_model(pipe.x_buf)
            logits = pipe.s2_model(act)
            loss = torch.nn.functional.cross_entropy(logits, pipe.y_buf)
            wmax = max(p.abs().max().item() for p in pipe.s2_model.parameters())
        print(f"step {i:4d} loss {float(loss):9.4f} max|w2| {wmax:9.3f}", flush=True)
print("nan_flag:", bool(pipe.nan_flag.item()))
