"""Autoscaled serving (BASELINE config 5): deploy a service with a KPA
concurrency target; under concurrent load the control plane scales
replicas up (Knative in-cluster; the controller's own KPA loop on the
local driver), and calls round-robin across the live pods.

Run: KT_LOCAL_MODE=true PYTHONPATH=. python examples/07_autoscale_serving.py
"""
import threading
import time

import kubetorch_amd as kt


def infer(x: float, delay: float = 8.0):
    import os
    import time as _t

    _t.sleep(delay)  # stand-in for model latency
    return {"y": x * 2, "pod": os.environ.get("POD_NAME", "?")}


def main():
    f = kt.fn(infer).to(
        kt.Compute(cpus=1).autoscale(target=1, min_scale=1, max_scale=3,
                                     scale_down_delay="10s"))
    try:
        print("warm:", f(1.0, delay=0, kt_timeout=60))
        results = []
        threads = [threading.Thread(
            target=lambda i=i: results.append(f(float(i), kt_timeout=120)))
            for i in range(3)]
        for t in threads:
            t.start()
        # watch the scale-up while the calls are in flight
        peak = 1
        for _ in range(20):
            w = f.workload() or {}
            peak = max(peak, len(w.get("pods") or []))
            if peak >= 3:
                break
            time.sleep(0.5)
        print(f"replicas under load: {peak}")
        for t in threads:
            t.join()
        # later calls round-robin across the scaled-out pods
        spread = {f(float(i), delay=0, kt_timeout=60)["pod"]
                  for i in range(6)}
        print(f"calls now spread over {len(spread)} pod(s): {sorted(spread)}")
        time.sleep(12)  # idle past the scale-down delay
        print("after idle:", f(9.0, delay=0, kt_timeout=60))
    finally:
        f.teardown()


if __name__ == "__main__":
    main()
