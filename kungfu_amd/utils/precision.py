"""Mixed-precision model conversion for the bf16-master training mode.

MI355X-native alternative to autocast: compute-heavy modules (conv/linear/
embedding) hold bf16 weights and produce bf16 activations directly, while
normalization layers (incl. the fused gfx950 BN/LN modules, whose kernels
require f32 params) stay f32. The fused optimizer keeps an f32 master copy
per low-precision bucket (sync_sgd._build_fused_step), so there are no
per-step bf16<->f32 weight-cast kernels at all (autocast spends ~0.5 ms/
step on them for ResNet-50 b64) and gradient all-reduce payloads halve.
"""
import torch

_BF16_MODULES = (torch.nn.Conv2d, torch.nn.Linear, torch.nn.Embedding)


def convert_bf16_master(model):
    """Cast conv/linear/embedding weights to bf16 in place; leave every
    normalization layer (and anything else) in f32. Returns the model."""
    for m in model.modules():
        if isinstance(m, _BF16_MODULES):
            m.to(dtype=torch.bfloat16)
    return model
