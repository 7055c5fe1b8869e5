#!/usr/bin/env python
"""Kernel microbenchmarks on the MI355X box: our HIP kernels vs rocBLAS
(torch.matmul) on the BERT hot shapes. Writes one JSON line per entry."""
import json
import sys
import time

import torch

sys.path.insert(0, ".")
from pdnlp_amd.ops import ext  # noqa: E402

DEV = "cuda:0"


def timeit(fn, iters=50, warmup=10):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters * 1e6  # us


def gemm_bench():
    e = ext()
    torch.manual_seed(0)
    # (name, M, N, K) — fwd shapes for bert-base bs32 seq128 and large seq512
    shapes = [
        ("qkv_base", 4096, 2304, 768),
        ("attnout_base", 4096, 768, 768),
        ("ffn_up_base", 4096, 3072, 768),
        ("ffn_down_base", 4096, 768, 3072),
        ("qkv_large", 8192, 3072, 1024),
        ("ffn_up_large", 8192, 4096, 1024),
    ]
    import os
    sweep = os.environ.get("SWEEP_TILES", "0") == "1"
    for name, M, N, K in shapes:
        A = torch.randn(M, K, device=DEV, dtype=torch.bfloat16)
        W = torch.randn(N, K, device=DEV, dtype=torch.bfloat16)
        b = torch.randn(N, device=DEV, dtype=torch.bfloat16)
        flops = 2.0 * M * N * K
        tiles = {}
        if sweep:
            ref = torch.matmul(A.float(), W.t().float())
            for tile in ["128x128", "64x128", "128x64", "64x64"]:
                os.environ["PDNLP_GEMM_TILE"] = tile
                out = e.gemm_nt_fwd(A, W, b, "none")[0]
                ok = torch.allclose(out.float(), ref + b.float(),
                                    rtol=3e-2, atol=3e-2)
                us = timeit(lambda: e.gemm_nt_fwd(A, W, b, "none"))
                tiles[tile] = {"us": round(us, 2),
                               "tflops": round(flops / us / 1e6, 1),
                               "ok": bool(ok)}
            del os.environ["PDNLP_GEMM_TILE"]
        us_ours = timeit(lambda: e.gemm_nt_fwd(A, W, b, "none"))
        us_blas = timeit(lambda: torch.matmul(A, W.t()))
        print(json.dumps({
            "bench": "gemm_nt", "shape": name, "M": M, "N": N, "K": K,
            "ours_us": round(us_ours, 2), "rocblas_us": round(us_blas, 2),
            "ours_tflops": round(flops / us_ours / 1e6, 1),
            "rocblas_tflops": round(flops / us_blas / 1e6, 1),
            **({"tiles": tiles} if tiles else {})}), flush=True)
        # backward dgemm shapes (rocBLAS path): dX = dy@W, dW = dy^T@x
        dy = torch.randn(M, N, device=DEV, dtype=torch.bfloat16)
        us_dx = timeit(lambda: torch.matmul(dy, W))
        us_dw = timeit(lambda: torch.matmul(dy.t(), A))
        print(json.dumps({
            "bench": "dgemm", "shape": name,
            "dx_us": round(us_dx, 2), "dx_tflops": round(flops / us_dx / 1e6, 1),
            "dw_us": round(us_dw, 2), "dw_tflops": round(flops / us_dw / 1e6, 1)}),
            flush=True)


def attn_bench():
    e = ext()
    import math
    for name, B, S, nh in [("base_bs32", 32, 128, 12), ("large_bs16", 16, 512, 16)]:
        H = nh * 64
        qkv = torch.randn(B, S, 3 * H, device=DEV, dtype=torch.bfloat16)
        mask = torch.zeros(B, 1, 1, S, device=DEV, dtype=torch.bfloat16)
        scale = 1.0 / math.sqrt(64)
        seed = torch.tensor([1], dtype=torch.int64, device=DEV)
        o, lse = e.flash_attn_qkv_fwd(qkv, mask, nh, scale, 0.1, seed, 1)
        dout = torch.randn_like(o)
        us_f = timeit(lambda: e.flash_attn_qkv_fwd(qkv, mask, nh, scale, 0.1, seed, 1))
        us_b = timeit(lambda: e.flash_attn_qkv_bwd(dout, qkv, o, lse, mask, nh,
                                                   scale, 0.1, seed, 1))
        flops_f = 2 * 2.0 * B * nh * S * S * 64
        print(json.dumps({
            "bench": "flash_attn", "shape": name,
            "fwd_us": round(us_f, 2), "bwd_us": round(us_b, 2),
            "fwd_tflops": round(flops_f / us_f / 1e6, 1),
            "bwd_tflops": round(2.5 * flops_f / us_b / 1e6, 1)}), flush=True)


if __name__ == "__main__":
    gemm_bench()
    attn_bench()
