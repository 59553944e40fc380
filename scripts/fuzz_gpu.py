"""Randomized GPU-vs-oracle differential fuzz (round-end hardening).

Each iteration draws a random cluster shape (pods, pools, nodes, GPU
fraction, selector width, criticals, zero-requests) and asserts the
full GPU decide equals the CPU oracle bit-for-bit. Every third
iteration additionally simulates a W-rank sharded decide ON ONE GPU:
pods split by uid-hash, per-shard segment rows built through the real
kernels, concatenated like an all-gather, and merged by
decide_rows_device — the complete multi-GPU decision math minus the
wire transfer.
"""
import os
import random
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import numpy as np
import torch

from mi355x_autoscaler.models.agent_pool import group_nodes_into_pools
from mi355x_autoscaler.ops.engine_gpu import GpuEngine
from mi355x_autoscaler.sim.engine import CpuEngine
from mi355x_autoscaler.sim.synth import burst_trace, wrap
from mi355x_autoscaler.snapshot.encode import encode_snapshot

ITERS = int(__import__("os").environ.get("FUZZ_ITERS", "60"))

POOL_MENU = [
    ("agentpool1", "Standard_D2_v2"),
    ("bigpool", "Standard_D4_v2"),
    ("gpupool", "Standard_ND96_MI355X_v1"),
    ("gpub", "Standard_ND96_MI355X_v1"),
]


def check_equal(d_gpu, d_cpu, ctx):
    assert d_gpu.units_needed == d_cpu.units_needed, (ctx, d_gpu.units_needed, d_cpu.units_needed)
    assert d_gpu.pods_placed_existing == d_cpu.pods_placed_existing, ctx
    assert d_gpu.pods_placed_new == d_cpu.pods_placed_new, ctx
    assert d_gpu.pods_unfulfillable == d_cpu.pods_unfulfillable, ctx
    np.testing.assert_array_equal(d_gpu.pool_residual_totals, d_cpu.pool_residual_totals)


def main(iters=None):
    eng = GpuEngine("cuda:0")
    cpu = CpuEngine()
    rng = random.Random(12345)
    n_iters = ITERS if iters is None else iters
    for it in range(n_iters):
        npools = rng.randint(1, 4)
        pools = tuple(POOL_MENU[:npools])
        sizes = tuple(rng.randint(1, 12) for _ in range(npools))
        n_pods = rng.choice([0, 3, 64, 500, 2000, 8000])
        gpufrac = rng.choice([0.0, 0.3, 1.0])
        seed = rng.randint(0, 10**6)
        pods_json, nodes_json = burst_trace(
            n_pods, n_nodes_per_pool=sizes, pools=pools, gpu_fraction=gpufrac,
            seed=seed, bound_pods_per_node=rng.randint(0, 4))
        # mutations: wide selectors, zero requests, critical pods
        for p in pods_json:
            r = rng.random()
            if r < 0.05:
                # wide widths straddle the 32-pair lattice edge: >32 takes
                # the poison (matches-nothing) path on both engines
                w = rng.choice([9, 14, 31, 32, 33, 40])
                p["spec"]["nodeSelector"] = {f"k{i}": f"v{i}" for i in range(w)}
            elif r < 0.10:
                p["spec"]["containers"][0]["resources"]["requests"] = {}
            elif r < 0.15:
                p.setdefault("metadata", {}).setdefault("annotations", {})[
                    "scheduler.alpha.kubernetes.io/critical-pod"] = "true"
        pods, nodes = wrap(pods_json, nodes_json)
        snap = encode_snapshot(pods, nodes, group_nodes_into_pools(nodes))
        ctx = f"iter {it}: P={n_pods} pools={npools} sizes={sizes} gpufrac={gpufrac} seed={seed}"
        ref = cpu.decide(snap)
        check_equal(eng.decide(snap), ref, ctx)

        if it % 3 == 0 and n_pods > 0:
            # simulated W-rank shard: uid-hash split, per-shard rows
            # through the real kernels, gather-concatenate, device merge
            from mi355x_autoscaler.parallel.elastic import shard_snapshot

            W = rng.choice([2, 4, 8])
            ds_full = eng.packer.pack(snap, eng.device)
            rows_chunks = []
            for rk in range(W):
                shard = shard_snapshot(snap, rk, W)
                ds_s = eng.packer.pack(shard, eng.device)
                segs = eng.build_segments(ds_s)
                rows_chunks.append(eng.rows_device(segs).clone())
            gathered = torch.cat(rows_chunks, dim=0)
            ds_full = eng.packer.pack(snap, eng.device)
            d_sh = eng.decide_rows_device(ds_full, gathered, snap.pool_names)
            assert d_sh is not None, ctx
            check_equal(d_sh, ref, ctx + f" sharded W={W}")
    print(f"fuzz ok: {n_iters} iterations (incl. simulated sharded merges)")


if __name__ == "__main__":
    main(int(sys.argv[1]) if len(sys.argv) > 1 else None)
