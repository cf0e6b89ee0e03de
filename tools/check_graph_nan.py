import sys, os, torch
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import bench as B
dev = torch.device("cuda:0")
pipe = B.ColocatedPipeline(dev, use_graphs=True)
xs, ys = B.make_batches(dev, 64, seed=1)
# trigger capture via one short run
pipe.run(8)
print("captured:", pipe.graph is not None, flush=True)

def weights_nan():
    for name, mod in (("s1", pipe.s1_model), ("s2", pipe.s2_model)):
        for pname, p in mod.named_parameters():
            if torch.isnan(p).any():
                return f"{name}.{pname}"
    return None

for trial in range(3):
    pipe.nan_flag.zero_()
    for i in range(512):
        pipe.x_buf.copy_(xs[i % 64])
        pipe.y_buf.copy_(ys[i % 64])
        pipe.graph.replay()
        if i % 64 == 63:
            torch.cuda.synchronize()
            if bool(pipe.nan_flag.item()):
                w = weights_nan()
                print(f"trial {trial} NaN flag at step<= {i}; first NaN weight: {w}", flush=True)
                pipe.nan_flag.zero_()
    torch.cuda.synchronize()
    print(f"trial {trial} done; flag={bool(pipe.nan_flag.item())} weightsNaN={weights_nan()}", flush=True)
