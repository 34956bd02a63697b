import os, sys, time
sys.path.insert(0, "/root/repo")
sys.path.insert(0, "/root/repo/tests/assets/summer")
os.environ["KT_LOCAL_MODE"] = "true"
os.environ["KT_USERNAME"] = "mprobe"
import kubetorch_amd as kt
import summer as summer_mod
import httpx

f = kt.fn(summer_mod.summer).to(kt.Compute(gpus=1))
try:
    from kubetorch_amd.controller.app import HUB
    pod = HUB.driver.pods(f.name, "default")[0]
    t0 = time.time()
    r = httpx.get(f"http://{pod}/metrics", timeout=30)
    dt = time.time() - t0
    print("status:", r.status_code, "latency:", round(dt, 2), "s, bytes:", len(r.text))
    act = [l for l in r.text.splitlines() if "active" in l]
    print("active lines:", act)
    print("first 400:", r.text[:400].replace(chr(10), " | "))
    # second scrape timing (autoscaler uses timeout=3)
    t0 = time.time(); r2 = httpx.get(f"http://{pod}/metrics", timeout=30)
    print("second scrape latency:", round(time.time() - t0, 2))
finally:
    f.teardown()
