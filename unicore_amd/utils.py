"""Core tensor/seed/grad utilities.

Parity targets (fresh implementation): reference unicore/utils.py —
apply_to_sample/move_to_cuda:43-84, multi-tensor grad-norm + clip:87-135,
import_user_module:138-171, activation fns:174-195, torch_seed:219-242,
checkpoint_sequential:306-333, tensor-tree helpers:336-411,
fp32_to_bf16_sr:414-423, set_jit_fusion_options:426-433.
"""

import contextlib
import importlib
import os
import sys
from functools import partial
from typing import Callable, List

import torch
import torch.nn.functional as F


# --------------------------------------------------------------------------
# Sample-tree traversal and device movement
# --------------------------------------------------------------------------


def apply_to_sample(f, sample):
    """Apply *f* to every tensor in a nested dict/list/tuple/set sample."""
    if hasattr(sample, "__len__") and len(sample) == 0:
        return {}

    def visit(node):
        if torch.is_tensor(node):
            return f(node)
        if isinstance(node, dict):
            return {k: visit(v) for k, v in node.items()}
        if isinstance(node, list):
            return [visit(v) for v in node]
        if isinstance(node, tuple):
            return tuple(visit(v) for v in node)
        if isinstance(node, set):
            return {visit(v) for v in node}
        return node

    return visit(sample)


def move_to_cuda(sample, device=None):
    target = device if device is not None else torch.cuda.current_device()
    # non_blocking is a no-op for unpinned sources, so always request it:
    # pinned batches (the buffered loader pins) overlap H2D with compute
    return apply_to_sample(
        lambda t: t.to(device=target, non_blocking=True), sample
    )


def move_to_cpu(sample):
    def to_host(t):
        # half-precision tensors have poor CPU support; widen first
        if t.dtype in (torch.bfloat16, torch.float16):
            t = t.to(dtype=torch.float32)
        return t.cpu()

    return apply_to_sample(to_host, sample)


# --------------------------------------------------------------------------
# Gradient norm / clipping (multi-tensor L2 via our fused kernel on GPU)
# --------------------------------------------------------------------------


def multi_tensor_total_norm(grads, chunk_size=2048 * 64) -> torch.Tensor:
    """Global L2 norm over a gradient list, grouped per (device, dtype).

    Each CUDA group goes through the fused multi-tensor kernel (one launch
    per group) when the extension is loaded; other groups use per-tensor
    torch.norm.
    """
    by_device = {}
    for g in grads:
        by_device.setdefault(g.device, {}).setdefault(g.dtype, []).append(g)

    partials = []
    for device, by_dtype in by_device.items():
        for group in by_dtype.values():
            if device.type == "cuda":
                from .ops import fused_l2norm, has_kernels

                if has_kernels():
                    partials.append(
                        fused_l2norm(group, chunk_size).to(device)
                    )
                    continue
            partials += [
                torch.norm(g, p=2, dtype=torch.float32) for g in group
            ]
    return torch.norm(torch.stack(partials))


def clip_grad_norm_(params, max_norm, aggregate_norm_fn=None) -> torch.Tensor:
    """Global-norm gradient clipping; returns the pre-clip norm."""
    if isinstance(params, torch.Tensor):
        params = [params]
    params = list(params)
    grads = [
        p.grad.detach()
        for p in params
        if p is not None and getattr(p, "grad", None) is not None
    ]
    if not grads:
        return (
            params[0].new_tensor(0.0) if params else torch.tensor(0.0)
        )

    if len(grads) == 1:
        total_norm = torch.norm(grads[0], p=2, dtype=torch.float32)
    else:
        total_norm = multi_tensor_total_norm(grads)

    if aggregate_norm_fn is not None:
        total_norm = aggregate_norm_fn(total_norm)

    if max_norm > 0:
        scale = (float(max_norm) / (total_norm + 1e-6)).clamp_(max=1)
        torch._foreach_mul_(grads, scale)
    return total_norm


# --------------------------------------------------------------------------
# Plugin loading (--user-dir)
# --------------------------------------------------------------------------


def import_user_module(args):
    """Import the --user-dir package so its @register_* decorators run."""
    requested = getattr(args, "user_dir", None)
    if requested is None:
        return
    path = os.path.abspath(requested)
    if not os.path.exists(path):
        # also try a path relative to the installed package
        in_pkg = os.path.join(os.path.dirname(__file__), requested)
        if os.path.exists(in_pkg):
            path = in_pkg
    parent, name = os.path.split(path)
    if name not in sys.modules:
        sys.path.insert(0, parent)
        importlib.import_module(name)
        sys.path.pop(0)


# --------------------------------------------------------------------------
# Activations
# --------------------------------------------------------------------------

_ACTIVATION_TABLE = {
    "relu": F.relu,
    "gelu": F.gelu,
    "tanh": torch.tanh,
    "linear": lambda x: x,
}


def get_activation_fn(activation: str) -> Callable:
    """Activation callable for a name in {relu, gelu, tanh, linear}."""
    try:
        return _ACTIVATION_TABLE[activation]
    except KeyError:
        raise NotImplementedError(f"activation {activation} not supported")


# --------------------------------------------------------------------------
# Seeding
# --------------------------------------------------------------------------


@contextlib.contextmanager
def torch_seed(seed, *args):
    """Deterministically seed torch (CPU + current GPU) inside the context and
    restore RNG state after.

    Mirrors the reference contract (unicore/utils.py:219-242): the effective
    seed is a hash-stack of ``seed`` and ``*args``, so per-rank dropout and
    rank-identical SR streams can be derived from (seed, num_updates, i, rank)
    and (seed, num_updates) respectively (reference unicore/trainer.py:602-607,
    712-713).
    """
    if seed is None:
        yield
        return
    seed = int(seed)
    for arg in args:
        seed = int(hash((seed, int(arg)))) % int(1e8)
    cpu_state = torch.random.get_rng_state()
    gpu_state = None
    use_cuda = torch.cuda.is_available() and torch.cuda.is_initialized()
    if use_cuda:
        gpu_state = torch.cuda.random.get_rng_state()
    torch.manual_seed(seed)
    if use_cuda:
        torch.cuda.manual_seed(seed)
    try:
        yield
    finally:
        torch.random.set_rng_state(cpu_state)
        if gpu_state is not None:
            torch.cuda.random.set_rng_state(gpu_state)


def get_rng_state():
    snapshot = {"torch_rng_state": torch.get_rng_state()}
    if torch.cuda.is_available():
        snapshot["cuda_rng_state"] = torch.cuda.get_rng_state()
    return snapshot


def set_rng_state(state):
    torch.set_rng_state(state["torch_rng_state"])
    if torch.cuda.is_available():
        torch.cuda.set_rng_state(state["cuda_rng_state"])


class set_torch_seed:
    """RAII variant of torch_seed with a bare int seed."""

    def __init__(self, seed):
        assert isinstance(seed, int)
        self.rng_state = get_rng_state()
        torch.manual_seed(seed)
        if torch.cuda.is_available():
            torch.cuda.manual_seed(seed)

    def __enter__(self):
        return self

    def __exit__(self, *exc):
        set_rng_state(self.rng_state)


# --------------------------------------------------------------------------
# Misc environment helpers
# --------------------------------------------------------------------------


def set_jit_fusion_options():
    """No-op kept for CLI parity: the hot ops run through our own fused HIP
    kernels (+ hipGraphs), not the TorchScript fuser."""


def load_gemm_tunings(path) -> bool:
    """Arm TunableOp with an offline-tuned hipBLASLt/rocBLAS algorithm
    table (tuning itself stays OFF — results were produced by
    tools/tunableop_pershape.sh). Returns True when the file was loaded.

    Safe to call on any stack: entries whose validators (ROCm/hipBLASLt
    versions) do not match are ignored by TunableOp."""
    if not path or not os.path.exists(path) or not torch.cuda.is_available():
        return False
    import torch.cuda.tunable as tunable

    tunable.enable(True)
    tunable.tuning_enable(False)
    tunable.read_file(path)
    return True


def has_parameters(module):
    return next(module.parameters(), None) is not None


class CudaEnvironment:
    """Device capability snapshot, gathered across ranks for the startup
    banner."""

    def __init__(self):
        prop = torch.cuda.get_device_properties(
            f"cuda:{torch.cuda.current_device()}"
        )
        self.name = prop.name
        self.major, self.minor = prop.major, prop.minor
        self.total_memory_in_GB = prop.total_memory / 1024**3

    @staticmethod
    def pretty_print_cuda_env_list(cuda_env_list):
        center = f"CUDA enviroments for all {len(cuda_env_list)} workers"
        pad = "*" * (40 - len(center) // 2)
        lines = [pad + center + pad]
        for rank, env in enumerate(cuda_env_list):
            lines.append(
                f"rank {rank:3d}: "
                f"capabilities = {env.major:2d}.{env.minor:<2d} ; "
                f"total memory = {env.total_memory_in_GB:.3f} GB ; "
                f"name = {env.name:40s}"
            )
        lines.append("*" * (40 + len(center) + 40))
        return "\n".join(lines)


# --------------------------------------------------------------------------
# Activation checkpointing helper (reference unicore/utils.py:306-333)
# --------------------------------------------------------------------------


def checkpoint_sequential(functions, input, enabled=True):
    """Run *functions* in sequence, recomputing activations in backward
    when enabled (and grad is on). Each function may take/return tuples."""

    def as_tuple(v):
        return v if type(v) is tuple else (v,)

    def run(fn, packed):
        return as_tuple(fn(*packed))

    packed = as_tuple(input)
    if enabled and torch.is_grad_enabled():
        for fn in functions:
            packed = torch.utils.checkpoint.checkpoint(
                lambda *a, _fn=fn: run(_fn, a), *packed
            )
    else:
        for fn in functions:
            packed = run(fn, packed)
    return packed


# --------------------------------------------------------------------------
# Tensor-tree helpers (AlphaFold-style; reference unicore/utils.py:336-411)
# --------------------------------------------------------------------------


def permute_final_dims(tensor: torch.Tensor, inds: List[int]):
    base = -len(inds)
    keep = list(range(len(tensor.shape[:base])))
    return tensor.permute(keep + [base + i for i in inds])


def flatten_final_dims(t: torch.Tensor, num_dims: int):
    return t.reshape(t.shape[:-num_dims] + (-1,))


def masked_mean(mask, value, dim, eps=1e-10):
    mask = mask.expand(*value.shape)
    return (mask * value).sum(dim=dim) / (eps + mask.sum(dim=dim))


def dict_multimap(fn, dicts):
    head = dicts[0]
    out = {}
    for key, value in head.items():
        stacked = [d[key] for d in dicts]
        out[key] = (
            dict_multimap(fn, stacked) if type(value) is dict else fn(stacked)
        )
    return out


def one_hot(x, num_classes, dtype=torch.float32):
    out = torch.zeros(*x.shape, num_classes, dtype=dtype, device=x.device)
    out.scatter_(-1, x.long().unsqueeze(-1), 1)
    return out


def batched_gather(data, inds, dim=0, num_batch_dims=0):
    assert dim < 0 or dim - num_batch_dims >= 0
    index = []
    for i, size in enumerate(data.shape[:num_batch_dims]):
        shape = (1,) * i + (-1,) + (1,) * (len(inds.shape) - i - 1)
        index.append(torch.arange(size).view(*shape))
    tail = [slice(None)] * (len(data.shape) - num_batch_dims)
    tail[dim - num_batch_dims if dim >= 0 else dim] = inds
    index.extend(tail)
    return data[index]


def dict_map(fn, dic, leaf_type):
    return {
        key: (
            dict_map(fn, value, leaf_type)
            if type(value) is dict
            else tree_map(fn, value, leaf_type)
        )
        for key, value in dic.items()
    }


def tree_map(fn, tree, leaf_type):
    if isinstance(tree, dict):
        return dict_map(fn, tree, leaf_type)
    if isinstance(tree, list):
        return [tree_map(fn, v, leaf_type) for v in tree]
    if isinstance(tree, tuple):
        return tuple(tree_map(fn, v, leaf_type) for v in tree)
    if isinstance(tree, leaf_type):
        return fn(tree)
    raise ValueError("Not supported")


tensor_tree_map = partial(tree_map, leaf_type=torch.Tensor)


# --------------------------------------------------------------------------
# fp32 -> bf16 stochastic rounding (reference unicore/utils.py:414-423)
# --------------------------------------------------------------------------


def fp32_to_bf16_sr(t_fp32: torch.Tensor, t_bf16: torch.Tensor):
    """Write t_fp32 into t_bf16 with stochastic rounding.

    On GPU this uses our Philox SR kernel; elsewhere a pure-torch emulation
    that adds a uniform dither in the truncated mantissa bits before
    truncating to bf16.
    """
    if t_fp32.is_cuda:
        from .ops import fused_fp32_to_bf16_sr, has_kernels

        if has_kernels():
            fused_fp32_to_bf16_sr(t_fp32, t_bf16)
            return
    # Pure-torch fallback: add a random value in [0, 2^-16) ulp-scaled to the
    # fp32 bit pattern, then truncate (round-to-zero on the bit pattern).
    bits = t_fp32.view(torch.int32)
    rand = torch.randint(
        0, 1 << 16, t_fp32.shape, dtype=torch.int32, device=t_fp32.device
    )
    dithered = bits + rand
    # truncate low 16 bits -> bf16 pattern
    truncated = dithered & ~0xFFFF
    # Handle inf/nan: keep original value (bit-dither can overflow exponent)
    out = truncated.view(torch.float32)
    bad = ~torch.isfinite(t_fp32)
    out = torch.where(bad, t_fp32, out)
    t_bf16.copy_(out.bfloat16())


@contextlib.contextmanager
def validate_with_ema(trainer, ema=False):
    """Swap the trainer's wrapped model for a copy of the EMA model during
    validation (reference unicore/utils.py:436-452)."""
    if not ema:
        yield
        return
    from copy import deepcopy

    _wrapped_model = trainer._wrapped_model
    trainer._wrapped_model = deepcopy(trainer.ema.model_ema)
    if trainer.args.fp16:
        trainer._wrapped_model.half()
    elif trainer.args.bf16:
        trainer._wrapped_model.bfloat16()
    try:
        yield
    finally:
        del trainer._wrapped_model
        trainer._wrapped_model = _wrapped_model


def item(tensor):
    if hasattr(tensor, "item"):
        return tensor.item()
    if hasattr(tensor, "__getitem__"):
        return tensor[0]
    return tensor
