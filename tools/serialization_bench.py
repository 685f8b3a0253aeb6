#!/usr/bin/env python3
"""Serialization microbenchmark — the successor of the reference's
Serialization-timing.ipynb.

The reference measured pickle-vs-msgpack dump/load + zlib framing of numpy
arrays on the host, because its wire format WAS host pickle bytes
(mpi_comms.py:186-193).  This framework's wire format is the flat device
buffer itself, so the comparison here is:

  host path (reference-style): tensor -> .cpu().numpy() -> pickle.dumps
                               -> frame -> pickle.loads -> torch
  device path (this repo):     codec.encode into a fixed-capacity device
                               wire tensor (identity = flat copy)

Run on CPU it benchmarks the host path only; on a GPU box both.
Writes gpurun_out/serialization_bench.json.
"""

import json
import os
import pickle
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

from pytorch_ps_mpi_amd import codecs


def t_host(fn, reps):
    t0 = time.perf_counter()
    for _ in range(reps):
        fn()
    return (time.perf_counter() - t0) / reps


def t_dev(fn, reps):
    for _ in range(3):
        fn()
    torch.cuda.synchronize()
    s = torch.cuda.Event(enable_timing=True)
    e = torch.cuda.Event(enable_timing=True)
    s.record()
    for _ in range(reps):
        fn()
    e.record()
    torch.cuda.synchronize()
    return s.elapsed_time(e) / reps * 1e-3


def main():
    results = {}
    sizes = [10, 10_000, 1_000_000, 25_000_000]
    have_gpu = torch.cuda.is_available()
    print(f"{'n':>10} {'host pickle us':>15} {'host rtrip us':>14} "
          f"{'dev identity us':>16} {'dev quant8 us':>14}")
    for n in sizes:
        reps = max(3, min(200, 20_000_000 // n))
        x = torch.randn(n)
        row = {}

        def host_dump():
            return pickle.dumps(x.numpy(), protocol=4)

        blob = host_dump()
        row["host_pickle_dump_s"] = t_host(host_dump, reps)
        row["host_roundtrip_s"] = t_host(
            lambda: torch.from_numpy(pickle.loads(blob).copy()), reps)
        row["bytes"] = len(blob)

        if have_gpu:
            xg = x.to("cuda:0").bfloat16()
            ident = codecs.Identity()
            wire = torch.zeros(n, dtype=torch.bfloat16, device="cuda:0")
            row["dev_identity_s"] = t_dev(lambda: ident.encode(xg, wire), reps)
            q = codecs.QuantInt8()
            wq = torch.zeros(q.wire_numel(n), dtype=torch.uint8,
                             device="cuda:0")
            row["dev_quant8_s"] = t_dev(lambda: q.encode(xg, wq), reps)
            # reference-style host round trip of the SAME gpu tensor
            row["host_path_of_gpu_tensor_s"] = t_host(
                lambda: pickle.dumps(xg.float().cpu().numpy(), protocol=4),
                max(3, reps // 10))
        results[n] = row
        print(f"{n:10d} {row['host_pickle_dump_s']*1e6:15.1f} "
              f"{row['host_roundtrip_s']*1e6:14.1f} "
              f"{row.get('dev_identity_s', 0)*1e6:16.1f} "
              f"{row.get('dev_quant8_s', 0)*1e6:14.1f}")

    os.makedirs("gpurun_out", exist_ok=True)
    with open("gpurun_out/serialization_bench.json", "w") as f:
        json.dump(results, f, indent=1)


if __name__ == "__main__":
    main()
