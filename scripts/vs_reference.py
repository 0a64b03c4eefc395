#!/usr/bin/env python3
"""The reference repo's only concrete performance numbers are two doxygen
CLI transcripts:
  A) CPD rank-30, 1857x4721x6328, 20,607 nnz, 12 iters:
     CPD total 0.194 s (~16 ms/iter), THREADS=3 (unspecified CPU)
  B) MPI CPD rank-10, 45981x11537x2504, 229,906 nnz, 12 iters,
     4 nodes x 4 threads: best 0.48 s/iter (coarse), CPD 3.361 s
This script runs the SAME shapes/ranks/iteration counts here and prints
CPD-loop seconds for (1) the C++ CPU core and (2) the HIP engine
(if a GPU is visible), for profiles/vs_reference_transcript.md."""
import os
import time

import torch

import splatt_amd as sp


def run(name, dims, nnz, rank, iters, seed):
    t = sp.SpTensor.synthetic(dims, nnz, seed=seed).fixed(dedup=True)
    opts = sp.CpdOptions(max_iters=iters, tolerance=0.0)
    tic = time.time()
    k = sp.cpd_als_cpu_native(t, rank, opts)
    cpu_s = time.time() - tic
    print(f"{name}: C++ CPU core  ({os.cpu_count()} threads avail): "
          f"CPD {cpu_s:.4f} s  ({cpu_s / iters * 1e3:.2f} ms/iter)  "
          f"fit {k.fit:.4f}", flush=True)
    if torch.cuda.is_available():
        cs = sp.csf_alloc(t.to("cuda"), "all")
        sp.cpd_als(cs, rank, sp.CpdOptions(max_iters=2, tolerance=0.0))
        torch.cuda.synchronize()
        tic = time.time()
        kg = sp.cpd_als(cs, rank, opts)
        torch.cuda.synchronize()
        gpu_s = time.time() - tic
        print(f"{name}: HIP engine (1x MI355X):      "
              f"CPD {gpu_s:.4f} s  ({gpu_s / iters * 1e3:.2f} ms/iter)  "
              f"fit {kg.fit:.4f}", flush=True)


run("A (30running.dox)", [1857, 4721, 6328], 20_607, 30, 12, 40)
run("B (50mpi.dox)", [45_981, 11_537, 2504], 229_906, 10, 12, 41)
