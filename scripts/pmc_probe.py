"""PMC-counter probe: loops the three hot MFMA kernels at
production shapes so rocprofv3 --pmc gets a clean per-kernel read.

Run (on a GPU box; counters must NOT be combined with trace domains):

    cd /tmp && export TMPDIR=/tmp
    rocprofv3 --pmc SQ_VALU_MFMA_BUSY_CYCLES SQ_WAVE_CYCLES SQ_WAIT_ANY \
        --kernel-trace -d gpurun_out/pmc -o pmc \
        -- python /root/repo/scripts/pmc_probe.py

Shapes are the ResNet-50 b=64 / BERT-base b=16 hot layers the serving
benchmark actually runs (see profiles/RESULTS.md).
"""
import os
import sys
import time

sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)),
                                ".."))

import torch  # noqa: E402

from tfservingcache_amd.engine import _tfsc_engine as ext  # noqa: E402

ITERS = int(os.environ.get("PMC_ITERS", "20"))


def bf(t):
    return t.to(torch.bfloat16).contiguous()


def main():
    dev = "cuda:0"
    torch.manual_seed(0)

    # 1. GEMM — BERT-base FFN up-proj at b=16: [16*128, 3072] x [3072, 768]
    M, N, K = 2048, 3072, 768
    A = bf(torch.randn(M, K, device=dev) * 0.3)
    Bt = bf(torch.randn(N, K, device=dev) * 0.3)
    bias = bf(torch.randn(N, device=dev) * 0.1)
    C = torch.empty(M, N, device=dev, dtype=torch.bfloat16)
    gemm = ext.ExecPlan([(ext.K_GEMM,
                          [A.data_ptr(), Bt.data_ptr(), bias.data_ptr(),
                           0, C.data_ptr()],
                          [M, N, K, ext.ACT_GELU], [1.0])])

    # 2. conv_igemm — ResNet-50 conv3_x 3x3 at b=64: 28x28x128 -> 128
    N_, H, W, Cc, R, S, Kc, st = 64, 28, 28, 128, 3, 3, 128, 1
    x = bf(torch.randn(N_, H, W, Cc, device=dev) * 0.3)
    Kp = ((R * S * Cc + 63) // 64) * 64
    w = torch.randn(R, S, Cc, Kc, device=dev) * 0.1
    wt = bf(torch.nn.functional.pad(
        w.reshape(R * S * Cc, Kc).t(), (0, Kp - R * S * Cc)))
    cbias = bf(torch.randn(Kc, device=dev) * 0.1)
    zeros = torch.zeros(64, device=dev, dtype=torch.uint8)
    pad = (R - 1) // 2
    out = torch.empty(N_, H, W, Kc, device=dev, dtype=torch.bfloat16)
    conv = ext.ExecPlan([(ext.K_CONV,
                          [x.data_ptr(), wt.data_ptr(), cbias.data_ptr(),
                           0, zeros.data_ptr(), out.data_ptr()],
                          [N_, H, W, Cc, Kc, R, S, st, st, pad, pad,
                           H, W, Kp, ext.ACT_RELU], [])])

    # 3. fused attention — BERT-base at b=16: B=16 S=128 H=12 D=64
    B, Sq, Hh, D = 16, 128, 12, 64
    q = bf(torch.randn(B, Sq, Hh, D, device=dev) * 0.3)
    k = bf(torch.randn(B, Sq, Hh, D, device=dev) * 0.3)
    v = bf(torch.randn(B, Sq, Hh, D, device=dev) * 0.3)
    o = torch.empty(B, Sq, Hh, D, device=dev, dtype=torch.bfloat16)
    attn = ext.ExecPlan([(ext.K_ATTENTION,
                          [q.data_ptr(), k.data_ptr(), v.data_ptr(),
                           o.data_ptr()], [B, Sq, Hh, D],
                          [1.0 / D ** 0.5])])

    for name, plan, flops in [
            ("gemm_2048x3072x768", gemm, 2 * M * N * K),
            ("conv_igemm_28x28x128", conv,
             2 * N_ * H * W * Kc * R * S * Cc),
            ("attention_b16_s128", attn,
             4 * B * Hh * Sq * Sq * D)]:
        plan.run()                    # warm
        torch.cuda.synchronize()
        t0 = time.perf_counter()
        for _ in range(ITERS):
            plan.run()
        torch.cuda.synchronize()
        dt = (time.perf_counter() - t0) / ITERS
        print(f"{name}: {dt * 1e6:.1f} us  "
              f"{flops / dt / 1e12:.1f} TFLOP/s")


if __name__ == "__main__":
    main()
