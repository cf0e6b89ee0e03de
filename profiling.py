#!/usr/bin/env python3
"""Device profiler entrypoint (reference-compatible CLI, profiling.py:14-17):

    python profiling.py --model VGG16 [--data CIFAR10] --size 4

Writes profiling.json with per-unit exe_time (ns, x3 safety factor like the
reference), per-unit activation byte sizes, device speed, and a transport
bandwidth estimate — consumed by client REGISTER and the server's auto
cut-point search.
"""

import argparse

from split_learning_amd.profiling import write_profiling_json
from split_learning_amd.utils import print_with_color

_DEFAULT_DATA = {"VGG16": "CIFAR10", "BERT": "AGNEWS", "KWT": "SPEECHCOMMANDS",
                 "MobileNetv1": "CIFAR10", "ViT": "CIFAR10"}


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--model", default="VGG16")
    ap.add_argument("--data", default=None)
    ap.add_argument("--size", type=int, default=4, help="profiling batch size")
    ap.add_argument("--out", default="profiling.json")
    args = ap.parse_args()

    data = args.data or _DEFAULT_DATA[args.model]
    prof = write_profiling_json(args.out, args.model, data, args.size)
    print_with_color(
        f"wrote {args.out}: {len(prof['exe_time'])} units, "
        f"speed={prof['speed']:.3e} batch/ns, network={prof['network']:.3e} B/ns",
        "green")


if __name__ == "__main__":
    main()
