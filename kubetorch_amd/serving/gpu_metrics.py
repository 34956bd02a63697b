"""MI355X GPU metrics for the pod /metrics endpoint: amd-smi (amdsmi python
bindings, falling back to `rocm-smi --json`) instead of the reference's
dcgm-exporter. Scraped by the metrics store on the same schedule."""
import json
import subprocess

from prometheus_client.core import GaugeMetricFamily



class AMDGPUCollector:
    """Custom prometheus collector: utilization, VRAM, power, temperature
    per GPU. Cheap no-op on CPU-only nodes."""

    def collect(self):
        # A collector exception (or a non-numeric sample — amd-smi reports
        # "N/A" for some fields) surfaces during prometheus TEXT
        # GENERATION and turns the whole /metrics endpoint into a 500,
        # which silently starves every consumer of the core metrics (the
        # autoscaler's concurrency signal, the TTL heartbeat). Coerce to
        # float and never raise.
        try:
            stats = _read_gpu_stats()
        except Exception:
            stats = []
        if not stats:
            return
        util = GaugeMetricFamily("kt_gpu_utilization_percent",
                                 "GPU busy percent", labels=["gpu"])
        vram = GaugeMetricFamily("kt_gpu_vram_used_bytes",
                                 "VRAM used", labels=["gpu"])
        vram_total = GaugeMetricFamily("kt_gpu_vram_total_bytes",
                                       "VRAM total", labels=["gpu"])
        power = GaugeMetricFamily("kt_gpu_power_watts",
                                  "socket power", labels=["gpu"])
        temp = GaugeMetricFamily("kt_gpu_temp_celsius",
                                 "junction temperature", labels=["gpu"])
        for i, s in enumerate(stats):
            lbl = [str(i)]
            for fam, key in ((util, "util"), (vram, "vram_used"),
                             (vram_total, "vram_total"), (power, "power"),
                             (temp, "temp")):
                try:
                    val = float(s.get(key))
                except (TypeError, ValueError):
                    continue
                fam.add_metric(lbl, val)
        yield from (util, vram, vram_total, power, temp)


def _read_gpu_stats():
    try:
        import amdsmi

        amdsmi.amdsmi_init()
        try:
            out = []
            for h in amdsmi.amdsmi_get_processor_handles():
                s = {}
                try:
                    s["util"] = amdsmi.amdsmi_get_gpu_activity(h)["gfx_activity"]
                except Exception:
                    pass
                try:
                    vu = amdsmi.amdsmi_get_gpu_vram_usage(h)
                    s["vram_used"] = vu["vram_used"] * 1024 * 1024
                    s["vram_total"] = vu["vram_total"] * 1024 * 1024
                except Exception:
                    pass
                try:
                    s["power"] = amdsmi.amdsmi_get_power_info(h)["average_socket_power"]
                except Exception:
                    pass
                try:
                    s["temp"] = amdsmi.amdsmi_get_temp_metric(
                        h, amdsmi.AmdSmiTemperatureType.JUNCTION,
                        amdsmi.AmdSmiTemperatureMetric.CURRENT)
                except Exception:
                    pass
                out.append(s)
            return out
        finally:
            amdsmi.amdsmi_shut_down()
    except Exception:
        pass
    # fallback: rocm-smi JSON
    try:
        r = subprocess.run(["rocm-smi", "--showuse", "--showmemuse",
                            "--showpower", "--json"],
                           capture_output=True, text=True, timeout=10)
        data = json.loads(r.stdout)
        out = []
        for card, vals in sorted(data.items()):
            if not card.startswith("card"):
                continue
            s = {}
            for k, v in vals.items():
                kl = k.lower()
                try:
                    if "gpu use" in kl:
                        s["util"] = float(v)
                    elif "memory use" in kl and "%" not in kl:
                        s["vram_used"] = float(v)
                    elif "power" in kl:
                        s["power"] = float(v)
                except (TypeError, ValueError):
                    pass
            out.append(s)
        return out
    except Exception:
        return []


_registered = False


def register():
    global _registered
    if not _registered:
        try:
            import torch

            if torch.cuda.is_available():
                from kubetorch_amd.serving.metrics import register_custom

                register_custom(AMDGPUCollector())
        except Exception:
            pass
        _registered = True
