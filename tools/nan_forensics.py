import sys, os, torch
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import bench as B

dev = torch.device("cuda:0")

def tensors_of(pipe):
    out = []
    for mname, model, opt in (("s1", pipe.s1_model, pipe.s1_opt),
                              ("s2", pipe.s2_model, pipe.s2_opt)):
        for n, p in model.named_parameters():
            out.append((f"{mname}.{n}.w", p))
            if p.grad is not None:
                out.append((f"{mname}.{n}.g", p.grad))
        for n, b in model.named_buffers():
            out.append((f"{mname}.{n}.buf", b))
        for i, mb in enumerate(opt.bufs):
            out.append((f"{mname}.mom{i}", mb))
    return out

for attempt in range(12):
    pipe = B.ColocatedPipeline(dev, use_graphs=True)
    xs, ys = B.make_batches(dev, 256, seed=3 + attempt)
    pipe.run(8)
    bad = False
    for i in range(1024):
        pipe.x_buf.copy_(xs[i % 256]); pipe.y_buf.copy_(ys[i % 256])
        pipe.graph.replay()
        if i % 64 == 63:
            torch.cuda.synchronize()
            if bool(pipe.nan_flag.item()):
                bad = True
                break
    torch.cuda.synchronize()
    if not bad:
        print(f"attempt {attempt}: clean", flush=True)
        del pipe
        continue
    print(f"attempt {attempt}: NaN at step<= {i}", flush=True)
    rows = []
    for name, t in tensors_of(pipe):
        tf = t.detach().float()
        isn = torch.isnan(tf) | torch.isinf(tf)
        cnt = int(isn.sum())
        fi = int(isn.view(-1).nonzero()[0]) if cnt else -1
        li = int(isn.view(-1).nonzero()[-1]) if cnt else -1
        rows.append((t.data_ptr(), name, t.numel() * t.element_size(), cnt, fi, li, t.numel()))
    rows.sort()
    for j, (ptr, name, nb, cnt, fi, li, ne) in enumerate(rows):
        mark = " <== BAD" if cnt else ""
        gap = ""
        if j > 0:
            prev = rows[j-1]
            gap = f" gap_from_prev={ptr - (prev[0]+prev[2])}"
        if cnt or (j+1 < len(rows) and rows[j+1][3]) or (j > 0 and rows[j-1][3]):
            print(f"  0x{ptr:x} {name:34s} bytes={nb:9d} bad={cnt:8d} "
                  f"first={fi} last={li} numel={ne}{gap}{mark}", flush=True)
    break
