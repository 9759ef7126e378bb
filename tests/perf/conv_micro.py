"""Microbench driver for PMC counter runs on the conv kernels.
Usage: python tests/perf/conv_micro.py [fwd|bwd_data|bwd_weight|all] [iters]
Runs the CNN_MNIST conv2 shape (the dominant kernel in the FL step)."""

import os
import sys

sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)),
                                '..', '..'))
import torch  # noqa: E402

from rlr_amd.ops import ext  # noqa: E402


def main():
    which = sys.argv[1] if len(sys.argv) > 1 else 'all'
    iters = int(sys.argv[2]) if len(sys.argv) > 2 else 30
    dev = 'cuda:0'
    torch.manual_seed(0)
    # conv2 of CNN_MNIST: (256,32,26,26) -> (256,64,24,24), 3x3 s1 p0
    x = torch.randn(256, 32, 26, 26, device=dev)
    w = torch.randn(64, 32, 3, 3, device=dev) * 0.1
    b = torch.randn(64, device=dev)
    y = ext().conv2d_fwd(x, w, b, 1, 0, False)
    dy = torch.randn_like(y)
    torch.cuda.synchronize()

    for _ in range(iters):
        if which in ('fwd', 'all'):
            ext().conv2d_fwd(x, w, b, 1, 0, False)
        if which in ('bwd_data', 'bwd_weight', 'all'):
            ext().conv2d_bwd(x, w, dy, 1, 0, False,
                             which != 'bwd_weight')
    torch.cuda.synchronize()
    print("done", which, iters)


if __name__ == '__main__':
    main()
