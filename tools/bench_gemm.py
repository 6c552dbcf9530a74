"""Micro-benchmark: sutro gemm_tn vs hipBLASLt (torch.mm) on the Qwen3-32B
decode GEMM shapes at batch 1024.

Run on a GPU box:
    python tools/bench_gemm.py [--check-only]
Writes a summary to gpurun_out/gemm_tn_sweep.txt when GPURUN_OUT is set.
"""

import argparse
import os
import sys

import torch

sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))
from sutro_amd import _C  # noqa: E402

# (name, M, N, K) — qkv/o/gate_up/down per layer + lm_head per step
SHAPES = [
    ("qkv", 1024, 10240, 5120),
    ("o", 1024, 5120, 8192),
    ("gate_up", 1024, 51200, 5120),
    ("down", 1024, 5120, 25600),
    ("lm_head", 1024, 151936, 5120),
]
# M-scaling of the library GEMMs (is a bigger decode batch more efficient?)
M_SWEEP = [256, 512, 1024, 2048, 4096]
CONFIGS = [(256, 256), (128, 256), (128, 128), (256, 128)]


def time_fn(fn, iters=20, warmup=5):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    start = torch.cuda.Event(enable_timing=True)
    end = torch.cuda.Event(enable_timing=True)
    start.record()
    for _ in range(iters):
        fn()
    end.record()
    torch.cuda.synchronize()
    return start.elapsed_time(end) / iters


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--check-only", action="store_true")
    ap.add_argument("--iters", type=int, default=20)
    ap.add_argument("--m", type=int, default=None,
                    help="override M for every shape (e.g. 2048 for the "
                         "batch-2048 operating point)")
    args = ap.parse_args()

    torch.manual_seed(0)
    dev = "cuda"
    lines = []

    def emit(s):
        print(s, flush=True)
        lines.append(s)

    for name, M, N, K in SHAPES:
        if args.m:
            M = args.m
        x = (torch.randn(M, K, device=dev) * 0.5).bfloat16()
        w = (torch.randn(N, K, device=dev) * 0.02).bfloat16()
        ref = x @ w.t()
        reff = x.float() @ w.t().float()
        t_blas = None if args.check_only else time_fn(lambda: x @ w.t(), args.iters)
        flops = 2.0 * M * N * K
        row = [f"{name:9s} M={M} N={N} K={K}"]
        if t_blas is not None:
            row.append(f"hipblaslt {t_blas*1000:8.1f}us {flops/t_blas/1e9:7.1f} TF")
        for bm, bn in CONFIGS:
            if M % bm:
                continue
            for swz in (1, 2):
                out = _C.gemm_tn(x, w, None, bm, bn, swz, 0)
                # bf16 GEMM vs its own bf16 ref; also bound error vs fp32
                err_ref = (out.float() - ref.float()).abs().max().item()
                err_f32 = (out.float() - reff).abs().max().item()
                rel = err_f32 / reff.abs().max().item()
                ok = rel < 2e-2 and torch.isfinite(out.float()).all().item()
                tag = f"bm{bm} bn{bn} swz{swz}"
                if not ok:
                    row.append(f"{tag}: FAIL rel={rel:.3e} dref={err_ref:.3e}")
                    continue
                if args.check_only:
                    row.append(f"{tag}: ok rel={rel:.1e}")
                else:
                    t = time_fn(lambda: _C.gemm_tn(x, w, None, bm, bn, swz, 0),
                                args.iters)
                    t2 = time_fn(lambda: _C.gemm_tn(x, w, None, bm, bn, swz, 1),
                                 args.iters)
                    row.append(f"{tag}: {flops/t/1e9:7.1f} TF | xcd {flops/t2/1e9:7.1f} TF")
        emit("\n    ".join(row))

    # fused residual epilogue check
    M, N, K = 1024, 5120, 8192
    x = (torch.randn(M, K, device=dev) * 0.5).bfloat16()
    w = (torch.randn(N, K, device=dev) * 0.02).bfloat16()
    res = torch.randn(M, N, device=dev).bfloat16()
    out = _C.gemm_tn(x, w, res, 128, 128, 2, 0)
    ref = (x.float() @ w.t().float() + res.float())
    rel = (out.float() - ref).abs().max().item() / ref.abs().max().item()
    emit(f"residual epilogue: rel={rel:.3e} {'ok' if rel < 2e-2 else 'FAIL'}")

    if not args.check_only:
        for name, _, N, K in SHAPES:
            row = [f"M-sweep {name:9s} N={N} K={K} (hipblaslt)"]
            for m in M_SWEEP:
                x = (torch.randn(m, K, device=dev) * 0.5).bfloat16()
                w = (torch.randn(N, K, device=dev) * 0.02).bfloat16()
                t = time_fn(lambda: x @ w.t(), args.iters)
                row.append(f"M={m}: {2.0*m*N*K/t/1e9:7.1f} TF")
            emit("  ".join(row))

    outd = os.environ.get("GPURUN_OUT", "gpurun_out")
    os.makedirs(outd, exist_ok=True)
    with open(os.path.join(outd, "gemm_tn_sweep.txt"), "w") as f:
        f.write("\n".join(lines) + "\n")


if __name__ == "__main__":
    main()
