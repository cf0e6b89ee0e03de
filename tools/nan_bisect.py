"""Graph-replay divergence bisection: run the bench step capture with selected
op families routed to torch (SLK_DBG_TORCH) / optimizer+CE toggles, fresh
process per trial (run via the shell loop)."""
import os, sys, torch
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import bench as B
from split_learning_amd.parallel.optim import FusedSGD

use_torch_sgd = os.environ.get("DBG_TORCH_SGD") == "1"
use_torch_ce = os.environ.get("DBG_TORCH_CE") == "1"
steps = int(sys.argv[1]) if len(sys.argv) > 1 else 1024
dev = torch.device("cuda:0")

class TorchOpt:
    def __init__(self, params):
        self.o = torch.optim.SGD([p for p in params if p.requires_grad],
                                 lr=B.LR, momentum=B.MOMENTUM)
    def zero_grad(self):
        pass
    def step(self):
        self.o.step()
        self.o.zero_grad(set_to_none=False)

pipe = B.ColocatedPipeline(dev, use_graphs=True)
if use_torch_sgd:
    pipe.s1_opt = TorchOpt(pipe.s1_model.parameters())
    pipe.s2_opt = TorchOpt(pipe.s2_model.parameters())
if use_torch_ce:
    pipe._ce = lambda lg, lb: torch.nn.functional.cross_entropy(lg, lb)

xs, ys = B.make_batches(dev, 256, seed=3)
pipe.run(8)
bad = 0
for i in range(steps):
    pipe.x_buf.copy_(xs[i % 256]); pipe.y_buf.copy_(ys[i % 256])
    pipe.graph.replay()
    if i % 128 == 127:
        torch.cuda.synchronize()
        if bool(pipe.nan_flag.item()):
            bad += 1
            pipe.nan_flag.zero_()
print(f"cfg SLK_DBG_TORCH={os.environ.get('SLK_DBG_TORCH','')} sgd={use_torch_sgd} "
      f"ce={use_torch_ce}: NaN windows={bad}", flush=True)
