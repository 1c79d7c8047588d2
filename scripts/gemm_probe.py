"""Direct kernel probes on a real GPU — fast triage before the full
pytest suite. Checks each kernel against torch fp32 with ASYMMETRIC
operands (transpose-detecting, guide §5.4 rule 16)."""
import sys
import os
sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))

import numpy as np
import torch

from tfservingcache_amd.engine import _tfsc_engine as ext


def bfbuf(t):
    return t.to(torch.bfloat16).contiguous()


def check(name, got, want, rtol=0.05, atol=0.05):
    got = got.float().cpu().numpy()
    want = want.float().cpu().numpy()
    err = np.abs(got - want)
    denom = np.abs(want) + 1e-3
    rel = (err / denom).max()
    ok = np.allclose(got, want, rtol=rtol, atol=atol)
    print(f"{name:28s} max_abs={err.max():.4g} max_rel={rel:.4g} "
          f"{'OK' if ok else 'FAIL'}")
    if not ok:
        bad = np.unravel_index(err.argmax(), err.shape)
        print(f"   worst at {bad}: got {got[bad]:.5f} want {want[bad]:.5f}")
    return ok


def run_plan(calls):
    plan = ext.ExecPlan(calls)
    plan.run()
    torch.cuda.synchronize()


def main():
    dev = "cuda:0"
    torch.manual_seed(0)
    ok = True

    # --- GEMM: C = A @ B^T (B passed [N][K]) -----------------------------
    for (M, N, K) in [(128, 128, 64), (128, 128, 128), (200, 300, 192),
                      (33, 64, 256), (512, 1000, 2048)]:
        A = torch.randn(M, K, device=dev) * 0.5
        Bt = torch.randn(N, K, device=dev) * 0.5
        bias = torch.randn(N, device=dev) * 0.2
        Ab, Btb, biasb = bfbuf(A), bfbuf(Bt), bfbuf(bias)
        C = torch.empty(M, N, device=dev, dtype=torch.bfloat16)
        run_plan([(ext.K_GEMM,
                   [Ab.data_ptr(), Btb.data_ptr(), biasb.data_ptr(), 0,
                    C.data_ptr()], [M, N, K, ext.ACT_NONE], [1.0])])
        want = Ab.float() @ Btb.float().t() + biasb.float()
        ok &= check(f"gemm {M}x{N}x{K}", C, want, rtol=0.08, atol=0.08)

    # --- GEMM + relu + residual ------------------------------------------
    M, N, K = 96, 160, 128
    A = bfbuf(torch.randn(M, K, device=dev) * 0.5)
    Bt = bfbuf(torch.randn(N, K, device=dev) * 0.5)
    res = bfbuf(torch.randn(M, N, device=dev))
    C = torch.empty(M, N, device=dev, dtype=torch.bfloat16)
    run_plan([(ext.K_GEMM, [A.data_ptr(), Bt.data_ptr(), 0, res.data_ptr(),
                            C.data_ptr()], [M, N, K, ext.ACT_RELU], [1.0])])
    want = torch.relu(A.float() @ Bt.float().t() + res.float())
    ok &= check("gemm+res+relu", C, want, rtol=0.08, atol=0.08)

    # --- batched GEMM trans_b (QK^T-like) --------------------------------
    bat, M, N, K = 6, 128, 128, 64
    A = bfbuf(torch.randn(bat, M, K, device=dev) * 0.5)
    B = bfbuf(torch.randn(bat, N, K, device=dev) * 0.5)
    C = torch.empty(bat, M, N, device=dev, dtype=torch.bfloat16)
    run_plan([(ext.K_BGEMM, [A.data_ptr(), B.data_ptr(), C.data_ptr()],
               [bat, M, N, K, M * K, N * K, M * N, 1], [1.0])])
    want = A.float() @ B.float().transpose(-1, -2)
    ok &= check("bgemm trans_b", C, want, rtol=0.08, atol=0.08)

    # --- batched GEMM no-trans (P@V-like) --------------------------------
    bat, M, N, K = 6, 128, 64, 128
    A = bfbuf(torch.randn(bat, M, K, device=dev).softmax(-1))
    B = bfbuf(torch.randn(bat, K, N, device=dev) * 0.5)
    C = torch.empty(bat, M, N, device=dev, dtype=torch.bfloat16)
    run_plan([(ext.K_BGEMM, [A.data_ptr(), B.data_ptr(), C.data_ptr()],
               [bat, M, N, K, M * K, K * N, M * N, 0], [1.0])])
    want = A.float() @ B.float()
    ok &= check("bgemm no-trans", C, want, rtol=0.08, atol=0.08)

    # --- softmax ----------------------------------------------------------
    X = bfbuf(torch.randn(300, 1000, device=dev) * 3)
    Y = torch.empty_like(X)
    run_plan([(ext.K_SOFTMAX, [X.data_ptr(), Y.data_ptr()], [300, 1000],
               [])])
    ok &= check("softmax", Y, X.float().softmax(-1), rtol=0.05, atol=1e-3)

    # --- layernorm ---------------------------------------------------------
    X = bfbuf(torch.randn(128, 768, device=dev) * 2)
    g = bfbuf(torch.randn(768, device=dev) * 0.5 + 1)
    b = bfbuf(torch.randn(768, device=dev) * 0.2)
    Y = torch.empty_like(X)
    run_plan([(ext.K_LAYERNORM, [X.data_ptr(), g.data_ptr(), b.data_ptr(),
                                 Y.data_ptr()], [128, 768], [1e-5])])
    want = torch.nn.functional.layer_norm(
        X.float(), (768,), g.float(), b.float(), 1e-5)
    ok &= check("layernorm", Y, want, rtol=0.1, atol=0.05)

    # --- eltwise broadcast -------------------------------------------------
    Xa = bfbuf(torch.randn(4, 8, 16, device=dev))
    Xb = bfbuf(torch.randn(16, device=dev))
    Y = torch.empty_like(Xa)
    n = 4 * 8 * 16
    run_plan([(ext.K_ELT_BINARY,
               [Xa.data_ptr(), Xb.data_ptr(), Y.data_ptr()],
               [n, ext.ELT_ADD, 3, 4, 8, 16, 128, 16, 1, 0, 0, 1], [])])
    ok &= check("eltwise bcast add", Y, Xa.float() + Xb.float(),
                rtol=0.02, atol=0.02)

    # --- im2col -> conv ---------------------------------------------------
    N_, H, W, C, R, S, Kc = 2, 16, 16, 64, 3, 3, 64
    x = bfbuf(torch.randn(N_, H, W, C, device=dev) * 0.5)
    w = torch.randn(R, S, C, Kc, device=dev) * 0.2
    Kp = ((R * S * C + 63) // 64) * 64
    wt = bfbuf(torch.nn.functional.pad(
        w.reshape(R * S * C, Kc).t(), (0, Kp - R * S * C)))
    bias = bfbuf(torch.zeros(Kc, device=dev))
    scratch = torch.empty(N_ * H * W * Kp, device=dev, dtype=torch.bfloat16)
    out = torch.empty(N_, H, W, Kc, device=dev, dtype=torch.bfloat16)
    run_plan([
        (ext.K_IM2COL, [x.data_ptr(), scratch.data_ptr()],
         [N_, H, W, C, R, S, 1, 1, 1, 1, H, W, Kp], []),
        (ext.K_GEMM, [scratch.data_ptr(), wt.data_ptr(), bias.data_ptr(),
                      0, out.data_ptr()],
         [N_ * H * W, Kc, Kp, ext.ACT_NONE], [1.0]),
    ])
    want = torch.nn.functional.conv2d(
        x.float().permute(0, 3, 1, 2), w.cuda().permute(3, 2, 0, 1),
        padding=1).permute(0, 2, 3, 1)
    ok &= check("conv3x3 (im2col+gemm)", out, want, rtol=0.1, atol=0.1)

    # --- fused implicit-GEMM conv ----------------------------------------
    for (N_, H, W, C, R, S, Kc, st) in [(2, 16, 16, 64, 3, 3, 64, 1),
                                        (2, 15, 15, 64, 3, 3, 128, 2),
                                        (1, 14, 14, 256, 3, 3, 256, 1)]:
        x = bfbuf(torch.randn(N_, H, W, C, device=dev) * 0.5)
        w = torch.randn(R, S, C, Kc, device=dev) * 0.2
        Kp = ((R * S * C + 63) // 64) * 64
        wt = bfbuf(torch.nn.functional.pad(
            w.reshape(R * S * C, Kc).t(), (0, Kp - R * S * C)))
        bias = bfbuf(torch.randn(Kc, device=dev) * 0.1)
        zeros = torch.zeros(64, device=dev, dtype=torch.uint8)
        pad = (R - 1) // 2
        Ho = (H + 2 * pad - R) // st + 1
        Wo = (W + 2 * pad - S) // st + 1
        out = torch.empty(N_, Ho, Wo, Kc, device=dev, dtype=torch.bfloat16)
        run_plan([(ext.K_CONV,
                   [x.data_ptr(), wt.data_ptr(), bias.data_ptr(), 0,
                    zeros.data_ptr(), out.data_ptr()],
                   [N_, H, W, C, Kc, R, S, st, st, pad, pad, Ho, Wo, Kp,
                    ext.ACT_RELU], [])])
        want = torch.relu(torch.nn.functional.conv2d(
            x.float().permute(0, 3, 1, 2), w.cuda().permute(3, 2, 0, 1),
            bias=bias.float(), padding=pad, stride=st
        ).permute(0, 2, 3, 1))
        ok &= check(f"conv_igemm {H}x{W}x{C}->{Kc} s{st}", out, want,
                    rtol=0.1, atol=0.1)

    # --- fused attention (flash-style) -----------------------------------
    for (B, S, H) in [(2, 128, 4), (1, 100, 2), (3, 64, 12)]:
        D = 64
        q = bfbuf(torch.randn(B, S, H, D, device=dev) * 0.5)
        k = bfbuf(torch.randn(B, S, H, D, device=dev) * 0.5)
        vv = bfbuf(torch.randn(B, S, H, D, device=dev) * 0.5)
        o = torch.empty(B, S, H, D, device=dev, dtype=torch.bfloat16)
        scale = 1.0 / (D ** 0.5)
        run_plan([(ext.K_ATTENTION,
                   [q.data_ptr(), k.data_ptr(), vv.data_ptr(),
                    o.data_ptr()], [B, S, H, D], [scale])])
        qf = q.float().permute(0, 2, 1, 3)     # [B,H,S,D]
        kf = k.float().permute(0, 2, 1, 3)
        vf = vv.float().permute(0, 2, 1, 3)
        want = torch.softmax(qf @ kf.transpose(-1, -2) * scale, dim=-1) @ vf
        want = want.permute(0, 2, 1, 3)
        ok &= check(f"attention B{B} S{S} H{H}", o, want, rtol=0.08,
                    atol=0.02)

    # --- pool -------------------------------------------------------------
    x = bfbuf(torch.randn(2, 16, 16, 32, device=dev))
    out = torch.empty(2, 8, 8, 32, device=dev, dtype=torch.bfloat16)
    run_plan([(ext.K_POOL, [x.data_ptr(), out.data_ptr()],
               [1, 2, 16, 16, 32, 8, 8, 2, 2, 2, 2, 0, 0], [])])
    want = torch.nn.functional.max_pool2d(
        x.float().permute(0, 3, 1, 2), 2, 2).permute(0, 2, 3, 1)
    ok &= check("maxpool 2x2", out, want, rtol=0.02, atol=0.02)

    print("ALL OK" if ok else "FAILURES PRESENT")
    return 0 if ok else 1


if __name__ == "__main__":
    sys.exit(main())
