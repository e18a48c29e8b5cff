"""Localize ResNet-50 single-GPU step cost: fwd / fwd+bwd / full step,
NCHW vs channels_last, find-mode sensitivity. Prints ms per phase."""
import argparse
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch


def timeit(fn, steps=10, warmup=5):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(steps):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / steps * 1000


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--batch", type=int, default=64)
    p.add_argument("--dtype", default="bf16")
    p.add_argument("--channels-last", type=int, default=1)
    p.add_argument("--benchmark-mode", type=int, default=1)
    p.add_argument("--steps", type=int, default=10)
    p.add_argument("--autocast", type=int, default=0,
                   help="fp32 model under torch.autocast(bf16)")
    p.add_argument("--fp32-bn", type=int, default=0,
                   help="bf16 model but BatchNorm modules kept fp32")
    args = p.parse_args()

    torch.backends.cudnn.benchmark = bool(args.benchmark_mode)
    dtype = torch.bfloat16 if args.dtype == "bf16" else torch.float32
    from kungfu_amd.models import resnet50

    model = resnet50()
    if args.autocast:
        model = model.to("cuda")
    else:
        model = model.to("cuda", dtype)
        if args.fp32_bn:
            for m in model.modules():
                if isinstance(m, torch.nn.BatchNorm2d):
                    m.float()
    x = torch.randn(args.batch, 3, 224, 224, device="cuda",
                    dtype=torch.float32 if args.autocast else dtype)
    y = torch.randint(0, 1000, (args.batch,), device="cuda")
    if args.channels_last:
        model = model.to(memory_format=torch.channels_last)
        x = x.contiguous(memory_format=torch.channels_last)

    opt = torch.optim.SGD(model.parameters(), lr=0.1, momentum=0.9,
                          foreach=True)

    import contextlib

    def amp():
        if args.autocast:
            return torch.autocast("cuda", dtype=torch.bfloat16)
        return contextlib.nullcontext()

    def fwd():
        with torch.no_grad(), amp():
            model(x)

    def fwd_bwd():
        for prm in model.parameters():
            prm.grad = None
        with amp():
            out = model(x)
        loss = torch.nn.functional.cross_entropy(out.float(), y)
        loss.backward()

    def full():
        opt.zero_grad(set_to_none=False)
        with amp():
            out = model(x)
        loss = torch.nn.functional.cross_entropy(out.float(), y)
        loss.backward()
        opt.step()

    t_fwd = timeit(fwd, args.steps)
    t_fb = timeit(fwd_bwd, args.steps)
    t_full = timeit(full, args.steps)
    print("batch=%d dtype=%s cl=%d ac=%d fbn=%d | fwd=%.1fms fwd+bwd=%.1fms "
          "full=%.1fms opt=%.1fms img/s=%.0f" %
          (args.batch, args.dtype, args.channels_last, args.autocast,
           args.fp32_bn, t_fwd, t_fb, t_full, t_full - t_fb,
           args.batch / t_full * 1000))


if __name__ == "__main__":
    main()
