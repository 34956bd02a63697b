"""LR schedules for the training engine (plain functions — the engine takes
the LR per step: `engine.step(lr=warmup_cosine(step, ...))`)."""
import math


def warmup_cosine(step, base_lr, warmup_steps, total_steps, min_lr=0.0):
    """Linear warmup to base_lr over warmup_steps, then cosine decay to
    min_lr at total_steps (the Llama/Chinchilla default)."""
    if step < warmup_steps:
        return base_lr * (step + 1) / max(1, warmup_steps)
    if step >= total_steps:
        return min_lr
    t = (step - warmup_steps) / max(1, total_steps - warmup_steps)
    return min_lr + 0.5 * (base_lr - min_lr) * (1 + math.cos(math.pi * t))


def warmup_linear(step, base_lr, warmup_steps, total_steps, min_lr=0.0):
    if step < warmup_steps:
        return base_lr * (step + 1) / max(1, warmup_steps)
    if step >= total_steps:
        return min_lr
    t = (step - warmup_steps) / max(1, total_steps - warmup_steps)
    return base_lr + (min_lr - base_lr) * t


def constant_with_warmup(step, base_lr, warmup_steps):
    if step < warmup_steps:
        return base_lr * (step + 1) / max(1, warmup_steps)
    return base_lr
