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


def apply_to_sample(f, sample):
    if hasattr(sample, "__len__") and len(sample) == 0:
        return {}

    def _apply(x):
        if torch.is_tensor(x):
            return f(x)
        elif isinstance(x, dict):
            return {key: _apply(value) for key, value in x.items()}
        elif isinstance(x, list):
            return [_apply(x) for x in x]
        elif isinstance(x, tuple):
            return tuple(_apply(x) for x in x)
        elif isinstance(x, set):
            return {_apply(x) for x in x}
        else:
            return x

    return _apply(sample)


def move_to_cuda(sample, device=None):
    device = device or torch.cuda.current_device()

    def _move_to_cuda(tensor):
        # non_blocking is ignored if tensor is not pinned, so we can always set
        # to True (H2D copies overlap with compute when the source is pinned;
        # the buffered loader pins its batches)
        return tensor.to(device=device, non_blocking=True)

    return apply_to_sample(_move_to_cuda, sample)


def move_to_cpu(sample):
    def _move_to_cpu(tensor):
        # PyTorch has poor support for half tensors (float16) on CPU.
        if tensor.dtype in {torch.bfloat16, torch.float16}:
            tensor = tensor.to(dtype=torch.float32)
        return tensor.cpu()

    return apply_to_sample(_move_to_cpu, sample)


# --------------------------------------------------------------------------
# Gradient norm / clipping (multi-tensor L2 via our fused kernel on GPU)
# --------------------------------------------------------------------------


def multi_tensor_total_norm(grads, chunk_size=2048 * 64) -> torch.Tensor:
    """L2 norm over a list of gradients, grouped per (device, dtype).

    On a GPU with our extension loaded this calls the fused multi-tensor
    kernel (one kernel per group); otherwise it falls back to
    torch._foreach_norm which is itself a fused multi-tensor path.
    """
    per_device_grads = {}
    for grad in grads:
        device = grad.device
        cur_device_grads = per_device_grads.setdefault(device, {})
        dtype = grad.dtype
        cur_device_grads.setdefault(dtype, []).append(grad)
    norms = []
    for device, per_dtype_grads in per_device_grads.items():
        for grads_group in per_dtype_grads.values():
            if device.type == "cuda":
                from .ops import fused_l2norm, has_kernels

                if has_kernels():
                    norms.append(fused_l2norm(grads_group, chunk_size).to(device))
                    continue
            norms += [torch.norm(g, p=2, dtype=torch.float32) for g in grads_group]
    total_norm = torch.norm(torch.stack(norms))
    return total_norm


def clip_grad_norm_(params, max_norm, aggregate_norm_fn=None) -> torch.Tensor:
    def grad_exists(p):
        return p is not None and getattr(p, "grad", None) is not None

    if isinstance(params, torch.Tensor):
        params = [params]
    params = list(params)
    grads = [p.grad.detach() for p in filter(grad_exists, params)]
    if len(grads) == 0:
        if len(params) > 0:
            return params[0].new_tensor(0.0)
        else:
            return torch.tensor(0.0)

    if len(grads) == 1:
        total_norm = torch.norm(grads[0], p=2, dtype=torch.float32)
    else:
        total_norm = multi_tensor_total_norm(grads)

    if aggregate_norm_fn is not None:
        total_norm = aggregate_norm_fn(total_norm)

    if max_norm > 0:
        max_norm = float(max_norm)
        clip_coef = (max_norm / (total_norm + 1e-6)).clamp_(max=1)
        torch._foreach_mul_(grads, clip_coef)
    return total_norm


# --------------------------------------------------------------------------
# Plugin loading (--user-dir)
# --------------------------------------------------------------------------


def import_user_module(args):
    module_path = getattr(args, "user_dir", None)
    if module_path is None:
        return
    module_path = os.path.abspath(args.user_dir)
    if not os.path.exists(module_path):
        unicore_rel_path = os.path.join(os.path.dirname(__file__), args.user_dir)
        if os.path.exists(unicore_rel_path):
            module_path = unicore_rel_path
    module_parent, module_name = os.path.split(module_path)

    if module_name not in sys.modules:
        sys.path.insert(0, module_parent)
        importlib.import_module(module_name)
        sys.path.pop(0)


# --------------------------------------------------------------------------
# Activations
# --------------------------------------------------------------------------


def get_activation_fn(activation: str) -> Callable:
    """Returns the activation function corresponding to `activation`"""
    if activation == "relu":
        return F.relu
    elif activation == "gelu":
        return F.gelu
    elif activation == "tanh":
        return torch.tanh
    elif activation == "linear":
        return lambda x: x
    else:
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
    state = torch.random.get_rng_state()
    cuda_state = None
    use_cuda = torch.cuda.is_available() and torch.cuda.is_initialized()
    if use_cuda:
        cuda_state = torch.cuda.random.get_rng_state()
    torch.manual_seed(seed)
    if use_cuda:
        torch.cuda.manual_seed(seed)
    try:
        yield
    finally:
        torch.random.set_rng_state(state)
        if use_cuda and cuda_state is not None:
            torch.cuda.random.set_rng_state(cuda_state)


# --------------------------------------------------------------------------
# Misc environment helpers
# --------------------------------------------------------------------------


def set_jit_fusion_options():
    """Set PyTorch fusion options (no-op stub kept for CLI parity)."""
    # We rely on our own fused HIP kernels + hipGraphs rather than the TorchScript
    # fuser; nothing to configure here.
    pass


def has_parameters(module):
    try:
        next(module.parameters())
        return True
    except StopIteration:
        return False


def get_rng_state():
    state = {"torch_rng_state": torch.get_rng_state()}
    if torch.cuda.is_available():
        state["cuda_rng_state"] = torch.cuda.get_rng_state()
    return state


def set_rng_state(state):
    torch.set_rng_state(state["torch_rng_state"])
    if torch.cuda.is_available():
        torch.cuda.set_rng_state(state["cuda_rng_state"])


class set_torch_seed(object):
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


class CudaEnvironment(object):
    def __init__(self):
        cur_device = torch.cuda.current_device()
        prop = torch.cuda.get_device_properties("cuda:{}".format(cur_device))
        self.name = prop.name
        self.major = prop.major
        self.minor = prop.minor
        self.total_memory_in_GB = prop.total_memory / 1024 / 1024 / 1024

    @staticmethod
    def pretty_print_cuda_env_list(cuda_env_list):
        """Given a list of CudaEnviorments, pretty print them"""
        num_workers = len(cuda_env_list)
        center = "CUDA enviroments for all {} workers".format(num_workers)
        banner_len = 40 - len(center) // 2
        first_line = "*" * banner_len + center + "*" * banner_len
        msgs = [first_line]
        for r, env in enumerate(cuda_env_list):
            msgs.append(
                "rank {:3d}: ".format(r)
                + "capabilities = {:2d}.{:<2d} ; ".format(env.major, env.minor)
                + "total memory = {:.3f} GB ; ".format(env.total_memory_in_GB)
                + "name = {:40s}".format(env.name)
            )
        msgs.append("*" * (40 + len(center) + 40))
        return "\n".join(msgs)


# --------------------------------------------------------------------------
# Activation checkpointing helper (reference unicore/utils.py:306-333)
# --------------------------------------------------------------------------


def checkpoint_sequential(
    functions,
    input,
    enabled=True,
):
    def wrap_tuple(a):
        return (a,) if type(a) is not tuple else a

    def exec(func, a):
        return wrap_tuple(func(*a))

    def get_wrap_exec(func):
        def wrap_exec(*a):
            return exec(func, a)

        return wrap_exec

    input = wrap_tuple(input)

    is_grad_enabled = torch.is_grad_enabled()

    if enabled and is_grad_enabled:
        for func in functions:
            input = torch.utils.checkpoint.checkpoint(get_wrap_exec(func), *input)
    else:
        for func in functions:
            input = exec(func, input)
    return input


# --------------------------------------------------------------------------
# Tensor-tree helpers (AlphaFold-style; reference unicore/utils.py:336-411)
# --------------------------------------------------------------------------


def permute_final_dims(tensor: torch.Tensor, inds: List[int]):
    zero_index = -1 * len(inds)
    first_inds = list(range(len(tensor.shape[:zero_index])))
    return tensor.permute(first_inds + [zero_index + i for i in inds])


def flatten_final_dims(t: torch.Tensor, num_dims: int):
    return t.reshape(t.shape[:-num_dims] + (-1,))


def masked_mean(mask, value, dim, eps=1e-10):
    mask = mask.expand(*value.shape)
    return torch.sum(mask * value, dim=dim) / (eps + torch.sum(mask, dim=dim))


def dict_multimap(fn, dicts):
    first = dicts[0]
    new_dict = {}
    for k, v in first.items():
        all_v = [d[k] for d in dicts]
        if type(v) is dict:
            new_dict[k] = dict_multimap(fn, all_v)
        else:
            new_dict[k] = fn(all_v)
    return new_dict


def one_hot(x, num_classes, dtype=torch.float32):
    x_one_hot = torch.zeros(*x.shape, num_classes, dtype=dtype, device=x.device)
    x_one_hot.scatter_(-1, x.long().unsqueeze(-1), 1)
    return x_one_hot


def batched_gather(data, inds, dim=0, num_batch_dims=0):
    assert dim < 0 or dim - num_batch_dims >= 0
    ranges = []
    for i, s in enumerate(data.shape[:num_batch_dims]):
        r = torch.arange(s)
        r = r.view(*(*((1,) * i), -1, *((1,) * (len(inds.shape) - i - 1))))
        ranges.append(r)

    remaining_dims = [slice(None) for _ in range(len(data.shape) - num_batch_dims)]
    remaining_dims[dim - num_batch_dims if dim >= 0 else dim] = inds
    ranges.extend(remaining_dims)
    return data[ranges]


def dict_map(fn, dic, leaf_type):
    new_dict = {}
    for k, v in dic.items():
        if type(v) is dict:
            new_dict[k] = dict_map(fn, v, leaf_type)
        else:
            new_dict[k] = tree_map(fn, v, leaf_type)
    return new_dict


def tree_map(fn, tree, leaf_type):
    if isinstance(tree, dict):
        return dict_map(fn, tree, leaf_type)
    elif isinstance(tree, list):
        return [tree_map(fn, x, leaf_type) for x in tree]
    elif isinstance(tree, tuple):
        return tuple([tree_map(fn, x, leaf_type) for x in tree])
    elif isinstance(tree, leaf_type):
        return fn(tree)
    else:
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
