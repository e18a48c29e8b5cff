"""All-reduce microbenchmark.

Reference parity: `python -m kungfu.tensorflow.v1.benchmarks --method
CPU|NCCL|NCCL+CPU` (srcs/python/kungfu/tensorflow/v1/benchmarks/) and
tests/go/cmd/kungfu-bench-allreduce. Here:

  python -m kungfu_amd.benchmarks --method CPU|RCCL|RCCL+CPU \
      --model resnet50-imagenet --steps 20

runs fused all-reduce over the fake-model tensor sizes and reports
algorithmic bandwidth (2*(n-1)/n * bytes / time) per step. Launch under
kungfu-run (CPU method) or torchrun (RCCL method).
"""
import argparse
import time

import torch

import kungfu_amd as kf
from kungfu_amd.models.fakemodel import model_sizes


def main():
    p = argparse.ArgumentParser("kungfu_amd.benchmarks")
    p.add_argument("--method", default="CPU",
                   choices=["CPU", "RCCL", "RCCL+CPU"])
    p.add_argument("--model", default="resnet50-imagenet")
    p.add_argument("--steps", type=int, default=20)
    p.add_argument("--warmup", type=int, default=5)
    p.add_argument("--dtype", default="f32", choices=["f32", "bf16"])
    p.add_argument("--fuse", type=int, default=1)
    p.add_argument("--sweep-bucket-mb", default="",
                   help="comma list, e.g. 4,8,16,32,64: slice the fused "
                        "buffer into buckets of each size and report "
                        "algbw per size (xGMI ring tuning, SURVEY 5.8)")
    args = p.parse_args()

    # CPU method must not bring up the GPU backend: on a box with fewer
    # GPUs than ranks the RCCL bootstrap would fail on duplicate devices
    kf.init(with_torch=args.method.startswith("RCCL"))
    sizes = model_sizes(args.model)
    dtype = torch.float32 if args.dtype == "f32" else torch.bfloat16
    use_gpu = args.method.startswith("RCCL")
    if use_gpu and not torch.cuda.is_available():
        raise SystemExit("RCCL method requires a GPU")
    device = "cuda" if use_gpu else "cpu"
    if args.fuse:
        tensors = [torch.ones(sum(sizes), dtype=dtype, device=device)]
    else:
        tensors = [torch.ones(s, dtype=dtype, device=device)
                   for s in sizes]
    total_bytes = sum(t.numel() * t.element_size() for t in tensors)

    from kungfu_amd.ops import (all_reduce, cpu_staged_all_reduce,
                                hierarchical_all_reduce)
    from kungfu_amd.ops import rccl as rccl_ops

    if use_gpu and not rccl_ops.active():
        # the benchmark must exercise the NATIVE layer even at np=1
        # (kf.init skips GPU backends for single-process clusters)
        rccl_ops.init_gpu()

    def run_once():
        for i, t in enumerate(tensors):
            name = "bench/%d" % i
            if args.method == "CPU":
                if use_gpu:
                    cpu_staged_all_reduce(t, name=name)
                else:
                    all_reduce(t, name=name)
            elif args.method == "RCCL":
                if rccl_ops.active():
                    # no world-size short-circuit: measures the real RCCL
                    # launch+completion even on a communicator of 1
                    rccl_ops.all_reduce(t, name=name)
                else:
                    all_reduce(t, name=name)
            else:  # RCCL+CPU hierarchical
                hierarchical_all_reduce(t, name=name)
        if use_gpu:
            torch.cuda.synchronize()

    if args.sweep_bucket_mb and use_gpu:
        # bucket-size sweep over the fused buffer: how large must each
        # collective be before the xGMI rings saturate?
        flat = torch.ones(sum(sizes), dtype=dtype, device=device)
        for mb in [float(x) for x in args.sweep_bucket_mb.split(",")]:
            per = max(1, int(mb * (1 << 20)) // flat.element_size())
            slices = [flat[i:i + per] for i in range(0, flat.numel(), per)]
            for _ in range(args.warmup):
                for j, sl in enumerate(slices):
                    rccl_ops.all_reduce(sl, name="sw%d" % j)
                torch.cuda.synchronize()
            kf.run_barrier()
            t0 = time.perf_counter()
            for _ in range(args.steps):
                handles = [rccl_ops.all_reduce_async(sl, name="sw%d" % j)
                           for j, sl in enumerate(slices)]
                for h in handles:
                    rccl_ops.wait(h)
                torch.cuda.synchronize()
            dt = (time.perf_counter() - t0) / args.steps
            n = kf.size()
            af = 2.0 * (n - 1) / n if n > 1 else 1.0
            total = flat.numel() * flat.element_size()
            if kf.rank() == 0:
                print("bucket=%.0fMiB nslices=%d ms=%.3f "
                      "algbw=%.2f GB/s"
                      % (mb, len(slices), dt * 1e3,
                         af * total / dt / 1e9), flush=True)
        kf.finalize()
        return

    for _ in range(args.warmup):
        run_once()
    kf.run_barrier()
    t0 = time.perf_counter()
    for _ in range(args.steps):
        run_once()
    elapsed = time.perf_counter() - t0
    n = kf.size()
    algo_factor = 2.0 * (n - 1) / n if n > 1 else 0.0
    per_step = elapsed / args.steps
    if kf.rank() == 0:
        print("method=%s model=%s np=%d fuse=%d bytes=%d "
              "ms/step=%.3f algbw=%.2f GB/s" %
              (args.method, args.model, n, args.fuse, total_bytes,
               per_step * 1e3,
               algo_factor * total_bytes / per_step / 1e9))
    kf.finalize()


if __name__ == "__main__":
    main()
