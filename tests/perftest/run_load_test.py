#!/usr/bin/env python3
"""Load test: router + N fake engines, sustained QPS, error budget check
(reference .github/workflows/router-e2e-test.yml:52-76 pattern)."""
import argparse
import asyncio
import sys
import time
import os

sys.path.insert(0, os.path.join(os.path.dirname(__file__), "..", ".."))


async def main_async(duration: float, qps: float, workers: int) -> int:
    import aiohttp
    import uvicorn
    import threading

    from production_stack_amd.router import app as app_mod
    from production_stack_amd.router.parser import parse_args
    from tests.fake_engine import FakeEngineServer

    backends = [FakeEngineServer(18900 + i, model="m1") for i in range(4)]
    for b in backends:
        b.start()
    args = parse_args(
        [
            "--static-backends", ",".join(b.url for b in backends),
            "--static-models", "m1",
            "--routing-logic", "roundrobin",
            "--port", "18890",
        ]
    )
    application = app_mod.build_app()
    app_mod.initialize_all(application, args)
    server = uvicorn.Server(
        uvicorn.Config(application, host="127.0.0.1", port=18890,
                       log_level="error")
    )
    t = threading.Thread(target=server.run, daemon=True)
    t.start()
    await asyncio.sleep(1.0)

    ok = err = 0
    sem = asyncio.Semaphore(workers)

    async def one(session):
        nonlocal ok, err
        async with sem:
            try:
                async with session.post(
                    "http://127.0.0.1:18890/v1/completions",
                    json={"model": "m1", "prompt": "x", "max_tokens": 2},
                    timeout=aiohttp.ClientTimeout(total=30),
                ) as r:
                    if r.status == 200:
                        await r.read()
                        ok += 1
                    else:
                        err += 1
            except Exception:
                err += 1

    t0 = time.time()
    tasks = []
    async with aiohttp.ClientSession() as session:
        while time.time() - t0 < duration:
            tasks.append(asyncio.create_task(one(session)))
            await asyncio.sleep(1.0 / qps)
        await asyncio.gather(*tasks)
    server.should_exit = True
    for b in backends:
        b.stop()
    total = ok + err
    print(f"requests={total} ok={ok} err={err} "
          f"rate={total/(time.time()-t0):.1f}/s")
    return 0 if err == 0 and total > 0 else 1


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--duration", type=float, default=30)
    ap.add_argument("--qps", type=float, default=10)
    ap.add_argument("--num-workers", type=int, default=32)
    a = ap.parse_args()
    sys.exit(asyncio.run(main_async(a.duration, a.qps, a.num_workers)))


if __name__ == "__main__":
    main()
