"""Per-shape timing of the fused BN(+res+ReLU) kernels vs the autocast
fp32 BN path (cast + batch_norm + add + relu), on ResNet-50 layer shapes."""
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

SHAPES = [  # (N, C, H, W) resnet50 b64 representative layers
    (64, 64, 112, 112),
    (64, 256, 56, 56),
    (64, 512, 28, 28),
    (64, 1024, 14, 14),
    (64, 2048, 7, 7),
]


def timeit(fn, steps=20, warmup=5):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(steps):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / steps * 1e3


def main():
    from kungfu_amd.ops.fused_bn import FusedBNReLU2d

    for shape in SHAPES:
        n, c, h, w = shape
        x = torch.randn(*shape, device="cuda").to(torch.bfloat16)
        x = x.contiguous(memory_format=torch.channels_last).requires_grad_()
        res = torch.randn_like(x).contiguous(
            memory_format=torch.channels_last)
        m = FusedBNReLU2d(c, relu=True).to("cuda")
        bn = torch.nn.BatchNorm2d(c).to("cuda")
        gbytes = x.numel() * 2 / 1e9

        y = m(x, res)
        g = torch.randn_like(y)

        def fused_fwd():
            return m(x, res)

        def fused_fwdbwd():
            x.grad = None
            m(x, res).backward(g)

        def eager_fwd():
            xf = x.float()
            return torch.relu(
                torch.nn.functional.batch_norm(
                    xf, bn.running_mean, bn.running_var, bn.weight,
                    bn.bias, True, 0.1, 1e-5) + res.float()).to(
                        torch.bfloat16)

        def eager_fwdbwd():
            x.grad = None
            eager_fwd().backward(g)

        tf = timeit(fused_fwd)
        tfb = timeit(fused_fwdbwd)
        te = timeit(eager_fwd)
        teb = timeit(eager_fwdbwd)
        # fwd traffic: read x, res; write y (3 streams of bf16)
        print("%s: fused fwd %.3fms (%.0f GB/s) bwd+fwd %.3fms | "
              "eager fwd %.3fms bwd+fwd %.3fms | speedup fwd %.2fx "
              "full %.2fx" %
              (shape, tf, 3 * gbytes / tf * 1e3, tfb, te, teb, te / tf,
               teb / tfb))


if __name__ == "__main__":
    main()
