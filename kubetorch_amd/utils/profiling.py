"""Step profiling surfaced through the log stream.

BASELINE asks for rocprof/rocm-smi counters "surfaced through the same log
stream" as everything else. Heavy counter collection stays with rocprofv3
(offline, profiles/); this is the in-process view: a context manager that
wraps a training/serving step in torch.profiler (ROCm kernels included) and
emits a compact top-kernel table to the pod's logger — visible in
`kt logs` / the client's live stream like any other output.

    from kubetorch_amd.utils.profiling import profile_step
    with profile_step("train_step", trace_dir="gpurun_out/traces"):
        loss = model.loss(x, y); loss.backward(); engine.step()
"""
import contextlib
import os
import time


@contextlib.contextmanager
def profile_step(name="step", top=15, trace_dir=None, printer=print):
    """Profile the enclosed block; on exit print a top-`top` kernel/op
    table (self device time first, CPU ops on CPU-only hosts) and, if
    trace_dir is set, export a chrome trace alongside."""
    import torch
    from torch.profiler import ProfilerActivity, profile

    acts = [ProfilerActivity.CPU]
    on_gpu = torch.cuda.is_available()
    if on_gpu:
        acts.append(ProfilerActivity.CUDA)
    t0 = time.perf_counter()
    with profile(activities=acts, record_shapes=False) as prof:
        yield prof
    wall_ms = (time.perf_counter() - t0) * 1000
    sort = "self_cuda_time_total" if on_gpu else "self_cpu_time_total"
    try:
        table = prof.key_averages().table(sort_by=sort, row_limit=top)
    except Exception:
        table = prof.key_averages().table(row_limit=top)
    printer(f"[kt-profile] {name}: {wall_ms:.1f} ms wall\n{table}")
    if trace_dir:
        os.makedirs(trace_dir, exist_ok=True)
        path = os.path.join(trace_dir, f"{name}_{int(time.time())}.json")
        prof.export_chrome_trace(path)
        printer(f"[kt-profile] chrome trace: {path}")


def gpu_snapshot():
    """One-shot device utilization/memory snapshot (amd-smi via the pod
    metrics collector) as a dict — loggable next to profile tables."""
    try:
        from kubetorch_amd.serving.gpu_metrics import _read_gpu_stats

        return _read_gpu_stats()
    except Exception:
        return {}
