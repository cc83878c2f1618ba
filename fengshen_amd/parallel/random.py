"""Model-parallel RNG tracker + our own activation checkpointing.

Behavioral parity: reference mpu/random.py (re-exports deepspeed.checkpointing)
and the RNG forks at mpu/layers.py:49 / transformer.py:375.  We own both:
the reference outsourced recompute + RNG fork to DeepSpeed; here it is a
self-contained torch.autograd.Function with CPU+HIP RNG state capture.
"""
from __future__ import annotations

import contextlib
from typing import Dict

import torch

_MODEL_PARALLEL_RNG_TRACKER_NAME = "model-parallel-rng"


def _get_device_rng_state():
    if torch.cuda.is_available() and torch.cuda.is_initialized():
        return torch.cuda.get_rng_state()
    return None


def _set_device_rng_state(state):
    if state is not None and torch.cuda.is_available():
        torch.cuda.set_rng_state(state)


class RNGStatesTracker:
    """Named RNG states; fork() swaps in a state so TP ranks draw *different*
    dropout masks for sharded activations but identical masks elsewhere."""

    def __init__(self):
        self.states_: Dict[str, tuple] = {}

    def reset(self):
        self.states_ = {}

    def get_states(self):
        return dict(self.states_)

    def set_states(self, states):
        self.states_ = dict(states)

    def add(self, name: str, seed: int):
        if name in self.states_:
            raise RuntimeError(f"rng state {name} already present")
        cpu_state = torch.get_rng_state()
        dev_state_orig = _get_device_rng_state()
        torch.manual_seed(seed)  # seeds CPU and device
        self.states_[name] = (torch.get_rng_state(), _get_device_rng_state())
        torch.set_rng_state(cpu_state)
        _set_device_rng_state(dev_state_orig)

    @contextlib.contextmanager
    def fork(self, name: str = _MODEL_PARALLEL_RNG_TRACKER_NAME):
        if name not in self.states_:
            # not seeded (single-process / tests): no-op fork
            yield
            return
        orig_cpu = torch.get_rng_state()
        orig_dev = _get_device_rng_state()
        cpu_state, dev_state = self.states_[name]
        torch.set_rng_state(cpu_state)
        _set_device_rng_state(dev_state)
        try:
            yield
        finally:
            self.states_[name] = (torch.get_rng_state(), _get_device_rng_state())
            torch.set_rng_state(orig_cpu)
            _set_device_rng_state(orig_dev)


_RNG_TRACKER = RNGStatesTracker()


def get_rng_tracker() -> RNGStatesTracker:
    return _RNG_TRACKER


# alias with the reference's name (mpu/random.py: get_cuda_rng_tracker)
get_cuda_rng_tracker = get_rng_tracker


def model_parallel_manual_seed(seed: int) -> None:
    """Seed data-parallel-identical / tensor-parallel-distinct RNG streams
    (reference: model_parallel_cuda_manual_seed, megatron_deepspeed.py:369)."""
    from fengshen_amd.parallel import groups

    tp_rank = groups.get_tensor_model_parallel_rank()
    # 2718 offset matches megatron lineage so seeds differ from data seeds
    model_parallel_seed = seed + 2718 + tp_rank
    torch.manual_seed(seed)
    if torch.cuda.is_available():
        torch.cuda.manual_seed(seed)
    _RNG_TRACKER.reset()
    _RNG_TRACKER.add(_MODEL_PARALLEL_RNG_TRACKER_NAME, model_parallel_seed)


class CheckpointFunction(torch.autograd.Function):
    """Activation checkpointing with full RNG restoration (CPU + device +
    model-parallel tracker)."""

    @staticmethod
    def forward(ctx, run_function, *args):
        ctx.run_function = run_function
        ctx.fwd_cpu_rng = torch.get_rng_state()
        ctx.fwd_dev_rng = _get_device_rng_state()
        ctx.fwd_tracker_states = _RNG_TRACKER.get_states()
        with torch.no_grad():
            outputs = run_function(*args)
        ctx.save_for_backward(*[a for a in args if torch.is_tensor(a)])
        ctx.input_is_tensor = [torch.is_tensor(a) for a in args]
        ctx.non_tensor_inputs = [a for a in args if not torch.is_tensor(a)]
        return outputs

    @staticmethod
    def backward(ctx, *grads):
        saved = list(ctx.saved_tensors)
        nons = list(ctx.non_tensor_inputs)
        inputs = []
        for is_t in ctx.input_is_tensor:
            inputs.append(saved.pop(0) if is_t else nons.pop(0))
        detached = [x.detach().requires_grad_(x.requires_grad) if torch.is_tensor(x) else x
                    for x in inputs]

        # restore RNG to forward-time state, recompute, then restore current
        cur_cpu = torch.get_rng_state()
        cur_dev = _get_device_rng_state()
        cur_tracker = _RNG_TRACKER.get_states()
        torch.set_rng_state(ctx.fwd_cpu_rng)
        _set_device_rng_state(ctx.fwd_dev_rng)
        _RNG_TRACKER.set_states(ctx.fwd_tracker_states)
        with torch.enable_grad():
            outputs = ctx.run_function(*detached)
        torch.set_rng_state(cur_cpu)
        _set_device_rng_state(cur_dev)
        _RNG_TRACKER.set_states(cur_tracker)

        if torch.is_tensor(outputs):
            outputs = (outputs,)
        out_tensors = [o for o in outputs if torch.is_tensor(o) and o.requires_grad]
        grad_tensors = [g for o, g in zip(outputs, grads)
                        if torch.is_tensor(o) and o.requires_grad]
        torch.autograd.backward(out_tensors, grad_tensors)
        input_grads = tuple(
            x.grad if torch.is_tensor(x) and x.requires_grad else None for x in detached)
        return (None,) + input_grads


def checkpoint(run_function, *args):
    """Recompute-on-backward activation checkpointing."""
    return CheckpointFunction.apply(run_function, *args)
