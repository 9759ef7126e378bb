"""Microbench for PMC runs on the ResNet bf16 conv kernels.
Usage: python tests/perf/resnet_micro.py [l1|l2|l3|all] [iters]
Exercises conv2d_fwd + conv2d_bwd on the 3x3 s1 p1 layer shapes (the
tap-resident kernels)."""

import os
import sys

sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)),
                                '..', '..'))
import torch  # noqa: E402

from rlr_amd.ops import ext  # noqa: E402

SHAPES = {
    'l1': (256, 64, 32, 64),    # Nb, C, HW, Kout
    'l2': (256, 128, 16, 128),
    'l3': (256, 256, 8, 256),
}


def main():
    which = sys.argv[1] if len(sys.argv) > 1 else 'all'
    iters = int(sys.argv[2]) if len(sys.argv) > 2 else 20
    torch.manual_seed(0)
    E = ext()
    runs = SHAPES.items() if which == 'all' else [(which, SHAPES[which])]
    work = []
    for name, (nb, c, hw, ko) in runs:
        x = (torch.randn(nb, c, hw, hw, device='cuda:0')
             .to(torch.bfloat16)
             .contiguous(memory_format=torch.channels_last))
        w = torch.randn(ko, c, 3, 3, device='cuda:0') * 0.05
        y = E.conv2d_fwd(x, w, None, 1, 1, False)
        dy = torch.randn_like(y)
        work.append((x, w, dy))
    torch.cuda.synchronize()
    for _ in range(iters):
        for x, w, dy in work:
            E.conv2d_fwd(x, w, None, 1, 1, False)
            E.conv2d_bwd(x, w, dy, 1, 1, False, True)
    torch.cuda.synchronize()
    print("done", which, iters)


if __name__ == '__main__':
    main()
