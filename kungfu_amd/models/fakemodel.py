"""Fake-model fixtures: per-tensor parameter-size lists for collective
benchmarks and tests.

Reference parity: tests/go/fakemodel/ (resnet50-imagenet, vgg16-imagenet,
slp-mnist, bert size arrays). Instead of hard-coding the arrays, we derive
them from the real model definitions in kungfu_amd.models on the meta
device, which guarantees they match what training actually all-reduces.
"""
import torch


def _param_sizes(build):
    with torch.device("meta"):
        model = build()
    return [p.numel() for p in model.parameters() if p.requires_grad]


def model_sizes(name):
    from kungfu_amd import models

    builders = {
        "resnet50-imagenet": lambda: models.resnet50(),
        "vgg16-imagenet": lambda: models.vgg16(),
        "bert": lambda: models.bert_base(),
        "slp-mnist": lambda: models.SLP(),
        "inception-v3": lambda: models.inception_v3(),
    }
    if name not in builders:
        raise KeyError("unknown fake model %r (have %s)" %
                       (name, sorted(builders)))
    return _param_sizes(builders[name])


def total_params(name):
    return sum(model_sizes(name))
