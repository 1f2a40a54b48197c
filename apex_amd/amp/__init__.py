"""apex_amd.amp — O0-O3 mixed precision (reconstructed apex.amp surface).

The reference implementation was removed upstream; its API is pinned by the
in-tree tests (tests/L1/common/main_amp.py:224-230,440,
tests/distributed/amp_master_params/amp_master_params.py:45-71,
examples/dcgan/main_amp.py:214-253 — see SURVEY.md §0/§3.4):

    model, optimizer = amp.initialize(model, optimizer, opt_level="O1|O2",
                                      keep_batchnorm_fp32=..., loss_scale=...)
    with amp.scale_loss(loss, optimizer) as scaled_loss:
        scaled_loss.backward()
    optimizer.step()
    amp.master_params(optimizer)

MI355X-first design decisions (vs. the reference's global torch-function
monkey-patching for O1):

* O1 wraps each model's ``forward`` in ``torch.autocast`` — the PyTorch-ROCm
  native cast-registry serves the same whitelist/blacklist role the patch
  registry served, with identical loss-parity semantics for the L1 harness.
* O2/O3 cast the model itself (keeping batchnorm fp32 for O2 by default) and
  keep fp32 master weights inside the optimizer (``_amp_stash``), with the
  unscale + overflow check done by one fused ``multi_tensor_scale`` launch
  and the dynamic-scale update by the on-device ``update_scale_hysteresis``
  kernel.
* default cast dtype is fp16 (reference default); pass
  ``cast_model_type=torch.bfloat16`` for the MI355X-preferred bf16 path
  (bf16 typically needs no loss scaling; ``loss_scale=1.0`` then).
"""

import contextlib
import functools
import itertools

from ..tracing import trace_mark

import torch

from ._amp_state import _amp_state, maybe_print
from .scaler import LossScaler
from .._ext import get_ext
from ..multi_tensor_apply import multi_tensor_applier

__all__ = ["initialize", "scale_loss", "master_params", "state_dict", "load_state_dict",
           "half_function", "float_function", "promote_function",
           "register_half_function", "register_float_function", "register_promote_function"]


class OptProperties:
    def __init__(self, opt_level, cast_model_type, patch_torch_functions,
                 keep_batchnorm_fp32, master_weights, loss_scale,
                 overflow_check=True):
        self.opt_level = opt_level
        self.cast_model_type = cast_model_type
        self.patch_torch_functions = patch_torch_functions
        self.keep_batchnorm_fp32 = keep_batchnorm_fp32
        self.master_weights = master_weights
        self.loss_scale = loss_scale
        self.overflow_check = overflow_check


_OPT_LEVELS = {
    "O0": dict(cast_model_type=None, patch_torch_functions=False,
               keep_batchnorm_fp32=None, master_weights=False, loss_scale=1.0),
    "O1": dict(cast_model_type=None, patch_torch_functions=True,
               keep_batchnorm_fp32=None, master_weights=False, loss_scale="dynamic"),
    "O2": dict(cast_model_type=torch.float16, patch_torch_functions=False,
               keep_batchnorm_fp32=True, master_weights=True, loss_scale="dynamic"),
    "O3": dict(cast_model_type=torch.float16, patch_torch_functions=False,
               keep_batchnorm_fp32=False, master_weights=False, loss_scale=1.0),
}


def _is_bn(module):
    return isinstance(module, torch.nn.modules.batchnorm._BatchNorm)


def _cast_model(model, dtype, keep_batchnorm_fp32):
    model.to(dtype=dtype)
    if keep_batchnorm_fp32:
        for m in model.modules():
            if _is_bn(m):
                m.float()
    return model


class _AmpStash:
    pass


def _wrap_forward_autocast(model, dtype):
    if getattr(model, "_amp_autocast_wrapped", False):
        return
    old_forward = model.forward
    device_type = "cuda" if torch.cuda.is_available() else "cpu"

    @functools.wraps(old_forward)
    def new_forward(*args, **kwargs):
        with torch.autocast(device_type=device_type, dtype=dtype):
            return old_forward(*args, **kwargs)

    model.forward = new_forward
    model._amp_autocast_wrapped = True


def _wrap_forward_input_cast(model, dtype):
    """O2/O3: cast floating-point inputs to the model dtype."""
    if getattr(model, "_amp_input_cast_wrapped", False):
        return
    old_forward = model.forward

    def cast_tree(x):
        if torch.is_tensor(x) and x.is_floating_point() and x.dtype == torch.float32:
            return x.to(dtype)
        if isinstance(x, (list, tuple)):
            return type(x)(cast_tree(v) for v in x)
        if isinstance(x, dict):
            return {k: cast_tree(v) for k, v in x.items()}
        return x

    @functools.wraps(old_forward)
    def new_forward(*args, **kwargs):
        return old_forward(*cast_tree(args), **cast_tree(kwargs))

    model.forward = new_forward
    model._amp_input_cast_wrapped = True


def _wrap_forward_output_cast(model, dtype):
    """cast_model_outputs: force the forward outputs to a dtype (reference
    kwarg on amp.initialize)."""
    old_forward = model.forward

    def cast_tree(x):
        if torch.is_tensor(x) and x.is_floating_point():
            return x.to(dtype)
        if isinstance(x, (list, tuple)):
            return type(x)(cast_tree(v) for v in x)
        if isinstance(x, dict):
            return {k: cast_tree(v) for k, v in x.items()}
        return x

    @functools.wraps(old_forward)
    def new_forward(*args, **kwargs):
        return cast_tree(old_forward(*args, **kwargs))

    model.forward = new_forward


def _process_optimizer_o2(optimizer, cast_type, verbose=False):
    """Build fp32 masters inside the optimizer (the amp-O2 master contract,
    evidenced by apex/optimizers/fused_sgd.py:165-230)."""
    stash = _AmpStash()
    stash.fp16_groups = []
    stash.fp32_from_fp16_groups = []
    stash.fp32_from_fp32_groups = []
    stash.all_fp16_params = []
    stash.all_fp32_from_fp16_params = []
    stash.all_fp32_from_fp32_params = []

    for group in optimizer.param_groups:
        fp16_params_this_group = []
        fp32_params_this_group = []
        fp32_from_fp16_params_this_group = []
        new_params = []
        for p in group["params"]:
            if p.requires_grad:
                if p.dtype in (torch.float16, torch.bfloat16):
                    fp16_params_this_group.append(p)
                    master = p.detach().clone().float()
                    master.requires_grad = True
                    fp32_from_fp16_params_this_group.append(master)
                    # optimizer state transfer if any
                    if p in optimizer.state:
                        optimizer.state[master] = optimizer.state.pop(p)
                    new_params.append(master)
                elif p.dtype == torch.float32:
                    fp32_params_this_group.append(p)
                    new_params.append(p)
                else:
                    raise TypeError(f"Unsupported param dtype {p.dtype}")
            else:
                new_params.append(p)
        group["params"] = new_params
        stash.fp16_groups.append(fp16_params_this_group)
        stash.fp32_from_fp16_groups.append(fp32_from_fp16_params_this_group)
        stash.fp32_from_fp32_groups.append(fp32_params_this_group)
        stash.all_fp16_params += fp16_params_this_group
        stash.all_fp32_from_fp16_params += fp32_from_fp16_params_this_group
        stash.all_fp32_from_fp32_params += fp32_params_this_group

    optimizer._amp_stash = stash
    return optimizer


def _materialize_master_grads(optimizer, scale):
    """Unscale fp16 model grads into fp32 master .grad (one fused launch)."""
    stash = optimizer._amp_stash
    model_grads, master_params_with_grad = [], []
    for p, master in zip(stash.all_fp16_params, stash.all_fp32_from_fp16_params):
        if p.grad is not None:
            model_grads.append(p.grad)
            master_params_with_grad.append(master)
    master_grads = []
    for master in master_params_with_grad:
        if master.grad is None:
            master.grad = torch.empty_like(master)
        master_grads.append(master.grad)
    return model_grads, master_grads


def _copy_master_to_model(optimizer):
    stash = optimizer._amp_stash
    if not stash.all_fp16_params:
        return
    device = stash.all_fp16_params[0].device
    if device.type == "cuda":
        amp_C = get_ext("amp_C")
        overflow_buf = torch.zeros(1, dtype=torch.int32, device=device)
        multi_tensor_applier(
            amp_C.multi_tensor_scale, overflow_buf,
            [stash.all_fp32_from_fp16_params, stash.all_fp16_params], 1.0,
        )
    else:
        with torch.no_grad():
            for master, p in zip(stash.all_fp32_from_fp16_params, stash.all_fp16_params):
                p.copy_(master.to(p.dtype))


def _patch_step_for_skip_and_copy(optimizer, needs_master_copy):
    if getattr(optimizer, "_amp_step_patched", False):
        return
    old_step = optimizer.step

    @functools.wraps(old_step)
    def new_step(closure=None):
        if getattr(optimizer, "_amp_skip_next_step", False):
            optimizer._amp_skip_next_step = False
            maybe_print("Gradient overflow. Skipping step.")
            return None
        out = old_step() if closure is None else old_step(closure)
        if needs_master_copy and not getattr(optimizer, "_amp_handles_param_copy", False):
            _copy_master_to_model(optimizer)
        return out

    optimizer.step = new_step
    optimizer._amp_step_patched = True

    if hasattr(optimizer, "_amp_stash"):
        old_zero = optimizer.zero_grad

        @functools.wraps(old_zero)
        def new_zero_grad(set_to_none=True):
            out = old_zero(set_to_none)
            for p in optimizer._amp_stash.all_fp16_params:
                p.grad = None
            return out

        optimizer.zero_grad = new_zero_grad


def initialize(
    models,
    optimizers=None,
    enabled=True,
    opt_level="O1",
    cast_model_type=None,
    patch_torch_functions=None,
    keep_batchnorm_fp32=None,
    master_weights=None,
    loss_scale=None,
    cast_model_outputs=None,
    num_losses=1,
    verbosity=1,
    min_loss_scale=None,
    max_loss_scale=2.0 ** 24,
    overflow_check=True,
):
    """Initialize amp. Returns (models, optimizers) with the same
    list-or-single structure the caller passed (reference behavior).

    ``overflow_check=False`` (static loss scale only, apex_amd extension for
    bf16 + hipGraph capture): skip the per-iteration device-to-host overflow
    read — scale_loss then performs NO host sync, so a whole training step
    (fwd + bwd + FusedAdam(capturable=True).step) records into a hipGraph.
    bf16 with scale 1.0 has no practical overflow path; steps are never
    skipped in this mode."""
    _amp_state.verbosity = verbosity

    models_was_list = isinstance(models, list)
    optimizers_was_list = isinstance(optimizers, list)
    model_list = models if models_was_list else [models]
    if optimizers is None:
        optimizer_list = []
    else:
        optimizer_list = optimizers if optimizers_was_list else [optimizers]

    if not enabled:
        _amp_state.initialized = True
        _amp_state.opt_properties = OptProperties("O0", None, False, None, False, 1.0)
        _amp_state.loss_scalers = [LossScaler(1.0) for _ in range(num_losses)]
        _amp_state.optimizers = optimizer_list
        _amp_state.models = model_list
        return models, optimizers

    if opt_level not in _OPT_LEVELS:
        raise ValueError(f"Unexpected opt_level {opt_level}; options are 'O0', 'O1', 'O2', 'O3'")

    props = dict(_OPT_LEVELS[opt_level])
    # keep_batchnorm_fp32 may arrive as string "True"/"False" (reference CLI)
    if isinstance(keep_batchnorm_fp32, str):
        keep_batchnorm_fp32 = keep_batchnorm_fp32 == "True"
    if cast_model_type is not None:
        props["cast_model_type"] = cast_model_type
    if patch_torch_functions is not None:
        props["patch_torch_functions"] = patch_torch_functions
    if keep_batchnorm_fp32 is not None:
        props["keep_batchnorm_fp32"] = keep_batchnorm_fp32
    if master_weights is not None:
        props["master_weights"] = master_weights
    if loss_scale is not None:
        props["loss_scale"] = loss_scale if loss_scale == "dynamic" else float(loss_scale)

    if not overflow_check and props["loss_scale"] == "dynamic":
        raise ValueError("overflow_check=False requires a static loss_scale")
    opt_properties = OptProperties(opt_level, props["cast_model_type"], props["patch_torch_functions"],
                                   props["keep_batchnorm_fp32"], props["master_weights"],
                                   props["loss_scale"], overflow_check)

    maybe_print(f"apex_amd.amp: opt_level={opt_level}, cast_model_type={props['cast_model_type']}, "
                f"keep_batchnorm_fp32={props['keep_batchnorm_fp32']}, master_weights={props['master_weights']}, "
                f"loss_scale={props['loss_scale']}")

    # --- models ---
    if props["cast_model_type"] is not None:  # O2 / O3
        _amp_state.cast_dtype = props["cast_model_type"]
        for model in model_list:
            _cast_model(model, props["cast_model_type"], props["keep_batchnorm_fp32"])
            _wrap_forward_input_cast(model, props["cast_model_type"])
    elif props["patch_torch_functions"]:  # O1
        cast_dtype = torch.float16 if cast_model_type is None else cast_model_type
        _amp_state.cast_dtype = cast_dtype
        for model in model_list:
            _wrap_forward_autocast(model, cast_dtype)
    if cast_model_outputs is not None:
        for model in model_list:
            _wrap_forward_output_cast(model, cast_model_outputs)

    # --- optimizers ---
    if props["master_weights"]:
        for opt in optimizer_list:
            _process_optimizer_o2(opt, props["cast_model_type"])
            if type(opt).__name__ == "FusedSGD":
                # FusedSGD's 4-list kernel writes the low-precision param
                # copy itself — amp must not copy masters back after step
                opt._amp_handles_param_copy = True
            _patch_step_for_skip_and_copy(opt, needs_master_copy=True)
    else:
        for opt in optimizer_list:
            _patch_step_for_skip_and_copy(opt, needs_master_copy=False)

    _amp_state.loss_scalers = [
        LossScaler(props["loss_scale"], min_loss_scale=min_loss_scale, max_loss_scale=max_loss_scale)
        for _ in range(num_losses)
    ]
    _amp_state.opt_properties = opt_properties
    _amp_state.optimizers = optimizer_list
    _amp_state.models = model_list
    _amp_state.initialized = True

    if optimizers is None:
        return model_list if models_was_list else model_list[0]
    return (
        model_list if models_was_list else model_list[0],
        optimizer_list if optimizers_was_list else optimizer_list[0],
    )


# --- function-cast registry (reference amp.half_function & friends) ---
# For user code that runs OUTSIDE the autocast-wrapped forward (custom loss
# functions, metrics): explicitly pin a function's compute dtype. Wrappers
# disable autocast inside so the pinned dtype is authoritative.

def _cast_tree_to(x, dtype):
    if torch.is_tensor(x) and x.is_floating_point():
        return x.to(dtype)
    if isinstance(x, (list, tuple)):
        return type(x)(_cast_tree_to(v, dtype) for v in x)
    if isinstance(x, dict):
        return {k: _cast_tree_to(v, dtype) for k, v in x.items()}
    return x


def _collect_float_dtypes(x, out):
    if torch.is_tensor(x) and x.is_floating_point():
        out.append(x.dtype)
    elif isinstance(x, (list, tuple)):
        for v in x:
            _collect_float_dtypes(v, out)
    elif isinstance(x, dict):
        for v in x.values():
            _collect_float_dtypes(v, out)


def _wrap_cast(fn, pick_dtype):
    @functools.wraps(fn)
    def wrapper(*args, **kwargs):
        dtype = pick_dtype(args, kwargs)
        with contextlib.ExitStack() as stack:
            for dev in ("cuda", "cpu"):
                stack.enter_context(torch.autocast(device_type=dev, enabled=False))
            return fn(*_cast_tree_to(args, dtype), **_cast_tree_to(kwargs, dtype))
    return wrapper


def half_function(fn):
    """Run ``fn`` with floating inputs cast to the amp low-precision dtype
    (the ``cast_model_type`` passed to initialize; fp16 default)."""
    return _wrap_cast(fn, lambda a, k: getattr(_amp_state, "cast_dtype", None) or torch.float16)


def float_function(fn):
    """Run ``fn`` with floating inputs cast to fp32."""
    return _wrap_cast(fn, lambda a, k: torch.float32)


def promote_function(fn):
    """Run ``fn`` with floating inputs cast to the widest floating dtype
    present among them (fp32 wins over bf16/fp16)."""
    def pick(args, kwargs):
        seen = []
        _collect_float_dtypes(args, seen)
        _collect_float_dtypes(kwargs, seen)
        if not seen:
            return torch.float32
        if torch.float64 in seen:
            return torch.float64
        if torch.float32 in seen:
            return torch.float32
        if torch.bfloat16 in seen and torch.float16 in seen:
            return torch.float32  # no safe common low dtype; promote
        return seen[0]
    return _wrap_cast(fn, pick)


def register_half_function(module, name):
    """Patch ``module.name`` in place with :func:`half_function` (effective
    immediately — no need to call before ``initialize``)."""
    setattr(module, name, half_function(getattr(module, name)))


def register_float_function(module, name):
    setattr(module, name, float_function(getattr(module, name)))


def register_promote_function(module, name):
    setattr(module, name, promote_function(getattr(module, name)))


@contextlib.contextmanager
def scale_loss(loss, optimizers, loss_id=0, model=None, delay_unscale=False):
    """Scale the loss; on exit unscale grads (into masters for O2), check
    overflow, update the dynamic scale on-device, and flag step-skip."""
    if not _amp_state.initialized:
        raise RuntimeError("Invoked amp.scale_loss before amp.initialize.")

    scaler = _amp_state.loss_scalers[loss_id]
    loss_scale = scaler.loss_scale()

    opt_list = optimizers if isinstance(optimizers, list) else [optimizers]

    if loss_scale == 1.0 and not scaler.dynamic:
        yield loss
        if delay_unscale:
            return
        check = _amp_state.opt_properties.overflow_check
        # still materialize masters for O2 with static scale 1.0
        if _amp_state.opt_properties.master_weights:
            for opt in opt_list:
                model_grads, master_grads = _materialize_master_grads(opt, 1.0)
                ov = scaler.unscale_grads(model_grads, master_grads, check=check)
                for p in opt._amp_stash.all_fp16_params:
                    p.grad = None  # consumed into masters (see dynamic path)
                if check and ov:
                    opt._amp_skip_next_step = True
        return

    yield loss.float() * loss_scale

    if delay_unscale:
        return

    # bracket: all unscales below share one overflow flag; the dynamic scale
    # ticks exactly once per iteration in finish_unscale (regardless of how
    # many optimizers / grad sets were unscaled)
    trace_mark("amp.unscale")
    scaler.begin_unscale()
    for opt in opt_list:
        if hasattr(opt, "_amp_stash"):  # O2: unscale fp16 grads into masters
            model_grads, master_grads = _materialize_master_grads(opt, loss_scale)
            scaler.unscale_grads(model_grads, master_grads)
            # model grads are consumed into the masters here; clear them so
            # the next backward doesn't accumulate stale gradients (the
            # optimizer's zero_grad only sees the master params)
            for p in opt._amp_stash.all_fp16_params:
                p.grad = None
            # fp32 params' grads unscaled in place
            stash = opt._amp_stash
            fp32_grads = [p.grad for p in stash.all_fp32_from_fp32_params if p.grad is not None]
            if fp32_grads:
                scaler.unscale_grads(fp32_grads, fp32_grads, scale_override=loss_scale)
        else:  # O0/O1/O3: unscale in place — SPLIT BY DTYPE: with
            # keep_batchnorm_fp32 the grad list mixes bf16/fp16 model grads
            # with fp32 BN grads, and one multi_tensor_scale launch
            # dispatches on the first tensor's dtype (a mixed list misreads
            # the rest -> permanent overflow -> every step skipped; caught
            # by the round-2 full L1 sweep at O3+keep_bn_fp32)
            grads = [p.grad for p in itertools.chain(*[g["params"] for g in opt.param_groups]) if p.grad is not None]
            by_dtype = {}
            for g in grads:
                by_dtype.setdefault(g.dtype, []).append(g)
            for gs in by_dtype.values():
                scaler.unscale_grads(gs, gs)
    if _amp_state.opt_properties.overflow_check:
        overflow = scaler.finish_unscale()
    else:
        # overflow_check=False (static scale): no device-to-host read at
        # all — the iteration stays async / hipGraph-capturable
        scaler._in_iteration = False
        overflow = False

    if overflow:
        for opt in opt_list:
            opt._amp_skip_next_step = True


def master_params(optimizer):
    """Iterate over the fp32 master params (reference:
    tests/distributed/amp_master_params/amp_master_params.py:71)."""
    if hasattr(optimizer, "_amp_stash"):
        stash = optimizer._amp_stash
        for p in itertools.chain(stash.all_fp32_from_fp16_params, stash.all_fp32_from_fp32_params):
            yield p
    else:
        for group in optimizer.param_groups:
            for p in group["params"]:
                yield p


def state_dict(destination=None):
    sd = destination if destination is not None else {}
    for i, scaler in enumerate(_amp_state.loss_scalers):
        sd[f"loss_scaler{i}"] = scaler.state_dict()
    return sd


def load_state_dict(state_dict):
    for i, scaler in enumerate(_amp_state.loss_scalers):
        key = f"loss_scaler{i}"
        if key in state_dict:
            scaler.load_state_dict(state_dict[key])
