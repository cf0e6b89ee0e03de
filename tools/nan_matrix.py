import sys, os, torch
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import bench as B

mode = sys.argv[1]         # eager | graphsync | graph
steps = int(sys.argv[2])
dev = torch.device("cuda:0")
pipe = B.ColocatedPipeline(dev, use_graphs=(mode != "eager"))
xs, ys = B.make_batches(dev, 256, seed=3)
if pipe.use_graphs:
    pipe.run(8)  # trigger capture
nan_steps = []
for i in range(steps):
    pipe.x_buf.copy_(xs[i % 256])
    pipe.y_buf.copy_(ys[i % 256])
    if pipe.graph is not None:
        pipe.graph.replay()
    else:
        pipe._step()
    if mode == "graphsync":
        torch.cuda.synchronize()
    if i % 128 == 127:
        torch.cuda.synchronize()
        if bool(pipe.nan_flag.item()):
            nan_steps.append(i)
            pipe.nan_flag.zero_()
            # reset weights to recover and keep hunting frequency
            pipe.s1_model, pipe.s1_opt = B.build_stage([0, B.CUT], dev)
            pipe.s2_model, pipe.s2_opt = B.build_stage([B.CUT, -1], dev)
            if pipe.graph is not None:
                pipe.graph = None
                pipe._capture()
torch.cuda.synchronize()
print(f"[{mode}] {steps} steps, NaN windows: {len(nan_steps)} at {nan_steps[:10]}", flush=True)
