#!/usr/bin/env python3
"""Guided tour of the framework internals (reference: examples/10_Internals
— affinity / ThreadPool / MemoryStack / Pool walk-through). Runs on CPU.

  python examples/internals.py
"""
import sys
import time
from pathlib import Path

import numpy as np

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))


def section(title):
    print(f"\n=== {title} ===")


def main():
    # ---- 1. memory: transactional stacks & the arena planner ----
    section("memory: TransactionalStack (request-scoped scratch)")
    from trtlab_amd.memory import ArenaPlanner, TransactionalStack

    st = TransactionalStack(1 << 20)
    a = st.allocate(1000)
    st.begin()               # a request begins
    b = st.allocate(5000)
    print(f"offsets: persistent={a}, request-scoped={b}")
    st.commit()              # request done -> scratch rolls back
    print(f"after commit, next alloc reuses: {st.allocate(100)}")

    section("memory: liveness-planned activation arena")
    p = ArenaPlanner()
    p.add("conv1_out", 1 << 20, first_use=0, last_use=1)
    p.add("conv2_out", 1 << 20, first_use=1, last_use=2)
    p.add("conv3_out", 1 << 20, first_use=2, last_use=3)
    offsets, total = p.plan()
    print(f"3x 1MiB tensors, serial lifetimes -> arena {total >> 20} MiB "
          f"(offsets {offsets})")

    # ---- 2. core: pools as concurrency limiters ----
    section("core: Pool checkouts return themselves")
    from trtlab_amd.core import Dispatcher, Pool, ThreadPool

    pool = Pool(["ctx0", "ctx1"])
    with pool.pop() as ctx:
        print(f"checked out {ctx}; available={pool.available}")
    print(f"after release: available={pool.available}")

    section("core: ThreadPool + Dispatcher dynamic batching")
    calls = []
    d = Dispatcher(max_batch_size=4, timeout_s=0.02,
                   compute_batch_fn=lambda items: (calls.append(len(items)),
                                                   [i * 2 for i in items])[1])
    futs = [d.enqueue(i) for i in range(6)]
    print(f"results: {[f.result(5) for f in futs]}; batch sizes: {calls}")
    d.shutdown()

    # ---- 3. engine: IR -> fused plan ----
    section("engine: planner fusion on a ResNet bottleneck")
    from trtlab_amd.engine.planner import Planner
    from trtlab_amd.models import build_resnet

    g = build_resnet(50, batch=1, image=64, calibrate=False)
    plan = Planner().compile(g)
    kinds = {}
    for dd in plan.ops:
        kinds[dd["kind"]] = kinds.get(dd["kind"], 0) + 1
    print(f"175 IR nodes -> {len(plan.ops)} fused ops (by kind: {kinds})")
    print(f"activation arena: {plan.arena_bytes >> 20} MiB for "
          f"{len(plan.offsets)} tensors (liveness reuse)")

    # ---- 4. rpc: loopback echo ----
    section("rpc: in-process echo server + client")
    from trtlab_amd.rpc import (AsyncService, EchoRequest, EchoResponse,
                                Server, SyncClient)

    server = Server("127.0.0.1:0")
    svc = AsyncService("trtlab.Echo")

    async def echo(req, ctx, res):
        return EchoResponse(message=req.message.upper(), tag=req.tag)

    svc.register_unary("Echo", echo, EchoRequest, EchoResponse)
    server.register_service(svc)
    server.async_start()
    c = SyncClient(f"127.0.0.1:{server.port}")
    r = c.call("trtlab.Echo", "Echo", EchoRequest(message="hello", tag=1),
               EchoResponse, timeout=5)
    print(f"echo -> {r.message}")
    c.close()
    server.shutdown()

    # ---- 5. streaming windows ----
    section("core: cyclic windowed buffer (overlapping stream windows)")
    from trtlab_amd.core.windowed import CyclicWindowedBuffer

    buf = CyclicWindowedBuffer(window_size=8, overlap=3)
    buf.push(np.arange(18, dtype=np.float32))
    for w in buf.pop_windows():
        print(f"  window: {w[:4]} ... {w[-3:]}  (last 3 shared with next)")


if __name__ == "__main__":
    main()
