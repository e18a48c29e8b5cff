"""Sweep the BN reduction-kernel launch geometry on hardware.

The launcher reads KF_BN_CAPKB / KF_BN_MAXBLK / KF_BN_ILP once per
process, so each config runs in a subprocess. Shapes are the distinct
ResNet-50 b64 BN sites. Prints per-shape: stats (us, TB/s read) and
bwd_reduce (us, TB/s read of dy+x).
"""
import json
import os
import subprocess
import sys

SHAPES = [(64 * 112 * 112, 64), (64 * 56 * 56, 64), (64 * 56 * 56, 256),
          (64 * 28 * 28, 128), (64 * 28 * 28, 512), (64 * 14 * 14, 256),
          (64 * 14 * 14, 1024), (64 * 7 * 7, 512), (64 * 7 * 7, 2048)]

CHILD = r'''
import json, time, torch
from kungfu_amd import _hip
shapes = %s
torch.cuda.set_device(0)
s = torch.cuda.current_stream().cuda_stream
out = {}
for M, C in shapes:
    x = torch.randn(M * C, device="cuda").to(torch.bfloat16)
    dy = torch.randn_like(x)
    sums = torch.zeros(16 * C, dtype=torch.float32, device="cuda")
    mean = torch.zeros(C, dtype=torch.float32, device="cuda")
    rstd = torch.ones(C, dtype=torch.float32, device="cuda")
    mask = torch.full((M * (C // 8),), 255, dtype=torch.uint8,
                      device="cuda")
    for _ in range(3):
        _hip.bn_stats(x.data_ptr(), M, C, sums.data_ptr(), s)
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(20):
        _hip.bn_stats(x.data_ptr(), M, C, sums.data_ptr(), s)
    torch.cuda.synchronize()
    st = (time.perf_counter() - t0) / 20
    for _ in range(3):
        _hip.bn_bwd_reduce(dy.data_ptr(), x.data_ptr(), mask.data_ptr(),
                           mean.data_ptr(), rstd.data_ptr(), M, C,
                           sums.data_ptr(), s)
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(20):
        _hip.bn_bwd_reduce(dy.data_ptr(), x.data_ptr(), mask.data_ptr(),
                           mean.data_ptr(), rstd.data_ptr(), M, C,
                           sums.data_ptr(), s)
    torch.cuda.synchronize()
    bw = (time.perf_counter() - t0) / 20
    gb = M * C * 2 / 1e9
    out["%%dx%%d" %% (M, C)] = [round(st * 1e6, 1), round(gb / st / 1e3, 2),
                                round(bw * 1e6, 1),
                                round(2 * gb / bw / 1e3, 2)]
print(json.dumps(out))
'''

CONFIGS = [
    {"KF_BN_ILP_STATS": "2", "KF_BN_ILP_BWD": "2", "KF_BN_OCT": "1"},  # r1
    {"KF_BN_ILP_STATS": "4", "KF_BN_ILP_BWD": "2", "KF_BN_OCT": "1"},
    {"KF_BN_ILP_STATS": "2", "KF_BN_ILP_BWD": "2", "KF_BN_OCT": "2"},
    {"KF_BN_ILP_STATS": "4", "KF_BN_ILP_BWD": "2", "KF_BN_OCT": "2"},
    {"KF_BN_ILP_STATS": "4", "KF_BN_ILP_BWD": "4", "KF_BN_OCT": "2"},
    {"KF_BN_ILP_STATS": "4", "KF_BN_ILP_BWD": "2", "KF_BN_OCT": "2",
     "KF_BN_CAPKB": "128"},
    {"KF_BN_ILP_STATS": "4", "KF_BN_ILP_BWD": "2", "KF_BN_OCT": "2",
     "KF_BN_CAPKB": "32"},
]


def main():
    for cfg in CONFIGS:
        env = dict(os.environ)
        env.update(cfg)
        r = subprocess.run([sys.executable, "-c", CHILD % SHAPES], env=env,
                           capture_output=True, text=True)
        print(json.dumps(cfg), r.stdout.strip() or r.stderr[-400:],
              flush=True)


if __name__ == "__main__":
    main()
