import time
import ethrex_amd
# MSM at the SP1 wrap-circuit sizes (2^21-2^23) + shard size
for lg in (20, 21, 22, 23):
    n = 1 << lg
    p = ethrex_amd.MsmPlan(n)
    p.gen_points(0)
    p.upload_scalars(ethrex_amd.gen_fr(42, n))
    for _ in range(3):
        p.run()
    t0 = time.perf_counter()
    K = 10
    for _ in range(K):
        p.run_async()
    p.sync()
    dt = (time.perf_counter() - t0) / K * 1000
    adds = 16 * (n + (1 << 17))
    print(f"msm 2^{lg}: {dt:.2f} ms  {adds/dt/1e6:.2f} G adds/s")
    p.destroy()
# NTT at the named 2^22 config + 2^26 fallback
for lg in (22, 26):
    n = 1 << lg
    q = ethrex_amd.NttPlan(n)
    q.upload(ethrex_amd.gen_fr(43, n))
    for _ in range(3):
        q.run()
    ts = []
    for _ in range(5):
        q.run()
        ts.append(q.last_times()["total_ms"])
    ms = sum(ts) / len(ts)
    print(f"ntt 2^{lg}: {ms:.2f} ms  {n/ms/1e6:.2f} G elems/s")
    q.destroy()
