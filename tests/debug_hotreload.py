import os, sys, time
sys.path.insert(0, "/root/repo")
sys.path.insert(0, "/root/repo/tests/assets/summer")
os.environ["KT_LOCAL_MODE"] = "true"; os.environ["KT_USERNAME"] = "hotr"
import kubetorch_amd as kt
import summer as summer_mod

f = kt.fn(summer_mod.summer).to(kt.Compute(cpus=1))
try:
    assert f(1, 2) == 3
    times = []
    for i in range(5):
        t0 = time.time(); f.to(); times.append(time.time() - t0)
        assert f(i, i) == 2 * i
    print("warm hot-reload s:", [round(t, 3) for t in times])
finally:
    f.teardown()
