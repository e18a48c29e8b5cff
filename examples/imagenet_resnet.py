"""Flagship example: ResNet-50 data-parallel training on MI355X GPUs with
the full KungFu-AMD feature set — elastic resize schedule, heartbeat
monitoring (auto-recovery compatible), gradient-noise-scale monitoring,
checkpointing, synthetic ImageNet-shaped data.

Launch (8 GPUs, elastic, auto-recovery):
  python -m kungfu_amd.run -np 8 -w -auto-recover 30s \
      python examples/imagenet_resnet.py --epochs 2 --schedule 100:4
"""
import argparse

import torch

import kungfu_amd as kf
from kungfu_amd.cmd import (monitor_batch_begin, monitor_batch_end,
                            monitor_epoch_end, monitor_train_end)
from kungfu_amd.datasets import elastic_loader, synthetic_imagenet
from kungfu_amd.models import resnet50
from kungfu_amd.optimizers import SynchronousSGDOptimizer
from kungfu_amd.parallel.elastic import (ElasticTrainer, load_checkpoint,
                                         save_checkpoint)


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--epochs", dest="epochs", type=int, default=1)
    p.add_argument("--n-epochs", dest="epochs", type=int)
    p.add_argument("--batch-size", type=int, default=64)
    p.add_argument("--samples", type=int, default=512)
    p.add_argument("--image-size", type=int, default=176)
    p.add_argument("--lr", type=float, default=0.1)
    p.add_argument("--schedule", default="", help="step:size,... resizes")
    p.add_argument("--ckpt", default="/tmp/kungfu_resnet_ckpt")
    p.add_argument("--restart", type=int, default=0)
    args = p.parse_args()

    kf.init()
    use_cuda = torch.cuda.is_available()
    device = torch.device("cuda" if use_cuda else "cpu")

    torch.manual_seed(0)
    model = resnet50(fused_bn=use_cuda)
    if use_cuda:
        model = model.to(device).to(memory_format=torch.channels_last)

    def make_opt(m):
        return SynchronousSGDOptimizer(
            torch.optim.SGD(m.parameters(), lr=args.lr, momentum=0.9,
                            weight_decay=1e-4),
            fused_step=use_cuda)

    trainer = ElasticTrainer(model, make_opt, schedule=args.schedule)
    if args.restart:
        step, _ = load_checkpoint(args.ckpt, model, trainer.optimizer,
                                  map_location=str(device))
        trainer.step = max(trainer.step, step)

    data = synthetic_imagenet(args.samples, size=args.image_size)
    loader, sampler = elastic_loader(data, args.batch_size)

    for epoch in range(args.epochs):
        sampler.epoch = epoch
        for x, y in loader:
            monitor_batch_begin()
            if use_cuda:
                x = x.to(device, non_blocking=True).contiguous(
                    memory_format=torch.channels_last)
                y = y.to(device, non_blocking=True)
            trainer.optimizer.zero_grad()
            with torch.autocast("cuda", dtype=torch.bfloat16,
                                enabled=use_cuda):
                out = model(x)
            loss = torch.nn.functional.cross_entropy(out.float(), y)
            loss.backward()
            trainer.optimizer.step()
            monitor_batch_end()
            if not trainer.after_step():
                print("DETACHED at step %d" % trainer.step, flush=True)
                kf.finalize()
                return
        save_checkpoint(args.ckpt, model, trainer.optimizer,
                        step=trainer.step)
        monitor_epoch_end()
        print("epoch=%d step=%d size=%d loss=%.3f" %
              (epoch, trainer.step, kf.size(), float(loss.detach())),
                  flush=True)
    monitor_train_end()
    print("DONE rank=%d steps=%d" % (kf.rank(), trainer.step), flush=True)
    kf.finalize()


if __name__ == "__main__":
    main()
