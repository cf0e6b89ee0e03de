#!/usr/bin/env python3
"""Accuracy / convergence evidence for the split-learning engine (round-1
VERDICT missing #2: "no accuracy number exists anywhere").

There is no network access on the GPU boxes, so real CIFAR-10 is unavailable;
instead this trains on a LEARNABLE synthetic task of the exact headline shape:
x ~ N(0,1) [3,32,32], label = argmax(P @ x) for a fixed random projection P
(10 classes).  Three arms train VGG16_CIFAR10 with identical seeds, data
order, and SGD(lr 5e-4, momentum 0.5):

  mono-native : monolithic model, our HIP kernels
  split-native: cut=7 split pipeline (stage-1 no-grad fwd, stage-2 step,
                stage-1 recompute+step — the production scheduler's order)
  mono-torch  : monolithic model, stock torch/MIOpen GPU kernels
                (SLK_DBG_TORCH routing), the independent reference

Reported: train loss trajectory + held-out accuracy per arm.  The claim
checked is (a) the engine LEARNS (acc >> 10% chance), and (b) split == mono
within tolerance (the pipeline changes no math at control-count 1).

Usage (GPU box):  python tools/convergence_acc.py [--steps 1500] [--json]
"""

import argparse
import json
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

BATCH = 32
LR = 5e-4
MOMENTUM = 0.5


def make_data(n_train=4096, n_test=1024, seed=42, device="cuda",
              sigma=3.0, label_noise=0.1):
    """Class-template task: x = template[y] + sigma*noise, 10 random
    templates.  (A random linear-projection rule was tried first and does
    NOT generalize for a CNN — all arms including stock torch sat at chance
    test accuracy while train loss fell; templates give the conv stack real
    spatial structure to learn.)  10% train-label noise keeps the fit
    non-trivial; test labels are clean."""
    g = torch.Generator().manual_seed(seed)
    n = n_train + n_test
    templates = torch.randn(10, 3, 32, 32, generator=g)
    y = torch.randint(0, 10, (n,), generator=g)
    x = templates[y] + sigma * torch.randn(n, 3, 32, 32, generator=g)
    y_train = y[:n_train].clone()
    flip = torch.rand(n_train, generator=g) < label_noise
    y_train[flip] = torch.randint(0, 10, (int(flip.sum()),), generator=g)
    x = x.to(device)
    return (x[:n_train], y_train.to(device)), (x[n_train:], y[n_train:].to(device))


def batches(x, y, steps, seed):
    g = torch.Generator().manual_seed(seed)
    n = len(x)
    for _ in range(steps):
        idx = torch.randint(0, n, (BATCH,), generator=g)
        yield x[idx], y[idx]


@torch.no_grad()
def accuracy(model_fn, x, y):
    correct = 0
    for i in range(0, len(x), 256):
        logits = model_fn(x[i:i + 256])
        correct += (logits.argmax(-1) == y[i:i + 256]).sum().item()
    return correct / len(x)


def train_mono(train, test, steps, device, log_every=0):
    from split_learning_amd.models import build_partition
    torch.manual_seed(1234)
    model = build_partition("VGG16", "CIFAR10", [0, 0]).to(device).train()
    opt = torch.optim.SGD(model.parameters(), lr=LR, momentum=MOMENTUM)
    losses = []
    for i, (xb, yb) in enumerate(batches(*train, steps, seed=99)):
        opt.zero_grad()
        loss = torch.nn.functional.cross_entropy(model(xb), yb)
        loss.backward()
        opt.step()
        if log_every and (i + 1) % log_every == 0:
            losses.append(round(float(loss.detach()), 4))
    model.eval()
    acc = accuracy(model, *test)
    return acc, losses


def train_split(train, test, steps, device, cut=7, log_every=0):
    """Production scheduler order at control-count 1: stage-1 fwd no-grad,
    stage-2 fwd/bwd/step, stage-1 recompute fwd/bwd/step."""
    from split_learning_amd.models import build_partition
    torch.manual_seed(1234)
    full = build_partition("VGG16", "CIFAR10", [0, 0])
    s1 = build_partition("VGG16", "CIFAR10", [0, cut])
    s2 = build_partition("VGG16", "CIFAR10", [cut, -1])
    # identical init to the monolithic arm
    s1.load_state_dict({k: v for k, v in full.state_dict().items()
                        if k in s1.state_dict()})
    s2.load_state_dict({k: v for k, v in full.state_dict().items()
                        if k in s2.state_dict()})
    s1, s2 = s1.to(device).train(), s2.to(device).train()
    o1 = torch.optim.SGD(s1.parameters(), lr=LR, momentum=MOMENTUM)
    o2 = torch.optim.SGD(s2.parameters(), lr=LR, momentum=MOMENTUM)
    losses = []
    for i, (xb, yb) in enumerate(batches(*train, steps, seed=99)):
        with torch.no_grad():
            act = s1(xb)
        act = act.detach().requires_grad_(True)
        o2.zero_grad()
        loss = torch.nn.functional.cross_entropy(s2(act), yb)
        loss.backward()
        o2.step()
        o1.zero_grad()
        out1 = s1(xb)
        out1.backward(gradient=act.grad)
        o1.step()
        if log_every and (i + 1) % log_every == 0:
            losses.append(round(float(loss.detach()), 4))
    s1.eval()
    s2.eval()
    acc = accuracy(lambda xb: s2(s1(xb)), *test)
    return acc, losses


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--steps", type=int, default=1500)
    ap.add_argument("--json", action="store_true")
    ap.add_argument("--skip-torch", action="store_true")
    args = ap.parse_args()
    device = "cuda" if torch.cuda.is_available() else "cpu"
    train, test = make_data(device=device)
    le = max(1, args.steps // 10)

    results = {}
    acc, tr = train_mono(train, test, args.steps, device, log_every=le)
    results["mono_native"] = {"test_acc": acc, "loss_trajectory": tr}
    print(f"mono-native : acc={acc:.4f} loss={tr}", flush=True)

    acc, tr = train_split(train, test, args.steps, device, log_every=le)
    results["split_native"] = {"test_acc": acc, "loss_trajectory": tr}
    print(f"split-native: acc={acc:.4f} loss={tr}", flush=True)

    if not args.skip_torch:
        os.environ["SLK_DBG_TORCH"] = "conv,bn,linear,pool,dropout,relu,attn"
        # re-import not needed: modules read the env at import; spawn a child
        import subprocess
        p = subprocess.run(
            [sys.executable, os.path.abspath(__file__), "--steps",
             str(args.steps), "--json", "--skip-torch", "--mono-only"],
            capture_output=True, text=True,
            env={**os.environ, "SLK_DBG_TORCH": "conv,bn,linear,pool,dropout,relu,attn"})
        try:
            child = json.loads(p.stdout.strip().splitlines()[-1])
            results["mono_torch"] = child["mono_native"]
            print(f"mono-torch  : acc={child['mono_native']['test_acc']:.4f}",
                  flush=True)
        except Exception:
            print(f"mono-torch arm failed:\n{p.stdout[-800:]}\n{p.stderr[-800:]}",
                  flush=True)

    if args.json:
        print(json.dumps(results), flush=True)
    return results


if __name__ == "__main__":
    if "--mono-only" in sys.argv:
        sys.argv = [a for a in sys.argv if a != "--mono-only"]
        ap = argparse.ArgumentParser()
        ap.add_argument("--steps", type=int, default=1500)
        ap.add_argument("--json", action="store_true")
        ap.add_argument("--skip-torch", action="store_true")
        a = ap.parse_args()
        device = "cuda" if torch.cuda.is_available() else "cpu"
        tr_d, te_d = make_data(device=device)
        acc, tr = train_mono(tr_d, te_d, a.steps, device,
                             log_every=max(1, a.steps // 10))
        print(json.dumps({"mono_native": {"test_acc": acc,
                                          "loss_trajectory": tr}}))
    else:
        main()
