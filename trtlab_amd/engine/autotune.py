"""Build-time kernel tactic autotuning (the role of TensorRT's builder
tactic selection, which the reference gets for free inside `.engine` files).

For every unique conv/gemm shape in a plan, time the tile-config candidates
(0 = the heuristic incl. split-K, 1..4 = fixed BMxBN tiles) on the actual
GPU with dummy buffers, and bake the winner into the op dict (`tile`).
Winners persist through the plan cache (plan_io), so tuning happens once
per model/config.
"""
from __future__ import annotations

import time
from typing import Dict, Tuple

from trtlab_amd.engine.planner import EnginePlan, K_CONV, K_GEMM

_CANDIDATES = (0, 1, 2, 3, 4)          # conv: heuristic + 4 fixed tiles
_CANDIDATES_GEMM = (0, 1, 2, 3, 4, 5)  # + 256x128 (staging-bound shapes)


def _conv_key(d: dict) -> Tuple:
    return ("conv", d["dtype"], d["Nb"], d["H"], d["W"], d["C"], d["Cout"],
            d["KH"], d["sh"], d["ph"])


def _gemm_key(d: dict) -> Tuple:
    return ("gemm", d["dtype"], d["M"], d["N"], d["K"])


def autotune_plan(plan: EnginePlan, device: int = 0, reps: int = 30,
                  warmup: int = 8, verbose: bool = False) -> Dict[Tuple, int]:
    """Mutates plan.ops in place; returns {shape_key: chosen_tile}."""
    from trtlab_amd import native

    C = native()
    C.hip.set_device(device)
    mem = C.memory

    # one shared scratch pool of buffers, grown to the largest need
    bufs: Dict[str, int] = {}
    sizes: Dict[str, int] = {}

    def buf(name: str, nbytes: int) -> int:
        nbytes = max(nbytes, 256)
        if sizes.get(name, 0) < nbytes:
            if name in bufs:
                mem.device_free(bufs[name], sizes[name])
            bufs[name] = mem.device_malloc(nbytes, device)
            sizes[name] = nbytes
        return bufs[name]

    chosen: Dict[Tuple, int] = {}
    try:
        for d in plan.ops:
            if d["kind"] == K_CONV:
                key = _conv_key(d)
            elif d["kind"] == K_GEMM:
                key = _gemm_key(d)
            else:
                continue
            if key in chosen:
                d["tile"] = chosen[key]
                continue
            # element sizes: dtype 2/3 are 1-byte in AND out; dtype 4 is
            # fp8 in / fp16 OUT (transformer projections)
            in_esize = 1 if d["dtype"] in (2, 3, 4) else 2
            out_esize = 1 if d["dtype"] in (2, 3) else 2
            if d["kind"] == K_CONV:
                oh = (d["H"] + 2 * d["ph"] - d["KH"]) // d["sh"] + 1
                ow = (d["W"] + 2 * d["pw"] - d["KW"]) // d["sw"] + 1
                kk = ((d["KH"] * d["KW"] * d["C"] + 127) // 128) * 128
                a = buf("in", d["Nb"] * d["H"] * d["W"] * d["C"] * in_esize)
                w = buf("w", d["Cout"] * kk * in_esize)
                o = buf("out", d["Nb"] * oh * ow * d["Cout"] * out_esize)
            else:
                a = buf("in", d["M"] * d["K"] * in_esize)
                w = buf("w", d["N"] * d["K"] * in_esize)
                o = buf("out", d["M"] * d["N"] * out_esize)
            sc = buf("scale", max(d.get("Cout", 0), d.get("N", 0)) * 4)
            bi = buf("bias", max(d.get("Cout", 0), d.get("N", 0)) * 4)
            zp = buf("zero", 256)

            def run(tile: int, sync: bool = False):
                if d["kind"] == K_CONV:
                    C.ops.conv2d(d["dtype"], a, w, o, scale=sc, bias=bi,
                                 zero_page=zp, Nb=d["Nb"], H=d["H"],
                                 W=d["W"], C=d["C"], Cout=d["Cout"],
                                 KH=d["KH"], KW=d["KW"], sh=d["sh"],
                                 sw=d["sw"], ph=d["ph"], pw=d["pw"],
                                 epi=4, sync=sync, tile=tile)
                else:
                    C.ops.gemm_bt(d["dtype"], a, w, o, scale=sc, bias=bi,
                                  M=d["M"], N=d["N"], K=d["K"], epi=4,
                                  sync=sync, tile=tile)

            best, best_us = 0, float("inf")
            cands = (_CANDIDATES if d["kind"] == K_CONV
                     else _CANDIDATES_GEMM)
            for tile in cands:
                for _ in range(warmup):
                    run(tile)
                C.hip.device_synchronize()
                t0 = time.perf_counter()
                for _ in range(reps):
                    run(tile)
                C.hip.device_synchronize()
                us = (time.perf_counter() - t0) / reps * 1e6
                if us < best_us * 0.98:  # prefer the heuristic on ties
                    best, best_us = tile, us
            chosen[key] = best
            d["tile"] = best
            if verbose:
                print(f"  autotune {key}: tile={best} ({best_us:.1f} us)")
    finally:
        for name, p in bufs.items():
            mem.device_free(p, sizes[name])
    return chosen
