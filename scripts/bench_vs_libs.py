#!/usr/bin/env python3
"""Hand-written kernels vs the ROCm library routes (torch -> rocSPARSE /
hipBLASLt) on the model's actual shapes. Within-process A/B."""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch


def timeit(fn, rounds=5):
    s, e = torch.cuda.Event(True), torch.cuda.Event(True)
    fn()
    torch.cuda.synchronize()
    ts = []
    for _ in range(rounds):
        s.record()
        fn()
        e.record()
        torch.cuda.synchronize()
        ts.append(s.elapsed_time(e))
    return sorted(ts)[len(ts) // 2]


def main():
    from roc_amd.graph import synthetic_graph
    from roc_amd import _C

    if not torch.cuda.is_available():
        print("bench_vs_libs: GPU-only A/B (torch->rocSPARSE/hipBLASLt "
              "vs roc_amd kernels); run on an MI355X box")
        return

    dev = "cuda:0"
    print("== SpMM: ours vs torch.sparse (rocSPARSE), fp32, Reddit shape ==")
    g = synthetic_graph(232965, 114848857, seed=1)
    rowptr = g.rowptr.to(dev)
    colidx = g.colidx.to(dev)
    for D in (64, 256):
        x = torch.randn(g.num_nodes, D, device=dev)
        out = torch.empty_like(x)
        ours = timeit(lambda: _C.spmm(out, x, rowptr, colidx, None, None, None))
        try:
            A = torch.sparse_csr_tensor(
                rowptr, colidx.to(torch.int64),
                torch.ones(g.num_edges, device=dev),
                size=(g.num_nodes, g.num_nodes))
            lib = timeit(lambda: torch.sparse.mm(A, x))
            print(f"  D={D}: ours {ours:.2f} ms | torch.sparse.mm {lib:.2f} ms"
                  f"  ({lib/ours:.2f}x)", flush=True)
        except Exception as ex:
            print(f"  D={D}: ours {ours:.2f} ms | torch.sparse.mm failed: "
                  f"{ex!r}", flush=True)
        del x, out
    del rowptr, colidx
    torch.cuda.empty_cache()

    print("== GEMM: ours vs torch.matmul (hipBLASLt), bf16, model shapes ==")
    shapes = [(232965, 608, 256), (232965, 256, 64), (1569960, 200, 256),
              (232965, 512, 512)]
    for M, K, N in shapes:
        A = torch.randn(M, K, dtype=torch.bfloat16, device=dev)
        B = torch.randn(K, N, dtype=torch.bfloat16, device=dev)
        Bt = B.t().contiguous()
        C = torch.empty(M, N, dtype=torch.bfloat16, device=dev)
        ours = timeit(lambda: _C.gemm_rr(C, A, Bt, False))
        lib = timeit(lambda: torch.matmul(A, B))
        tf = 2 * M * K * N / 1e12
        print(f"  {M}x{K}x{N}: ours {ours:.3f} ms ({tf/ours*1e3:.0f} TF) | "
              f"torch.matmul {lib:.3f} ms ({tf/lib*1e3:.0f} TF)", flush=True)
        del A, B, Bt, C
        torch.cuda.empty_cache()

    print("== dW GEMM (A^T B, fp32 out): ours vs torch.matmul ==")
    for R, Ka, N in [(232965, 608, 256), (232965, 256, 64)]:
        A = torch.randn(R, Ka, dtype=torch.bfloat16, device=dev)
        B = torch.randn(R, N, dtype=torch.bfloat16, device=dev)
        C = torch.zeros(Ka, N, dtype=torch.float32, device=dev)
        ours = timeit(lambda: _C.gemm_atb(C, A, B))
        lib = timeit(lambda: torch.matmul(A.t().float(), B.float()))
        lib_bf = timeit(lambda: torch.matmul(A.t(), B))
        tf = 2 * R * Ka * N / 1e12
        print(f"  {R}x{Ka}x{N}: ours {ours:.3f} ms ({tf/ours*1e3:.0f} TF) | "
              f"matmul-fp32cast {lib:.3f} | matmul-bf16 {lib_bf:.3f} ms",
              flush=True)


if __name__ == "__main__":
    main()
