"""ASP — Automatic SParsity (2:4 structured sparsity workflow).

API parity with the reference ``apex.contrib.sparsity.ASP``
(apex/contrib/sparsity/asp.py:39-311 and README "2 lines of Python"):

    ASP.prune_trained_model(model, optimizer)

computes N:M masks for whitelisted layers (Linear/Conv) and monkey-patches
``optimizer.step`` so masks are re-applied after every step. With
``allow_permutation=True`` the accuracy-preserving channel permutation
search (permutation_lib: greedy channel-swap + exhaustive stripe-pair, GPU
kernel in csrc/permutation_search.hip) reorders input channels before the
masks are computed, with FX-graph propagation to keep the model function
unchanged.
"""

import types

import torch


def _is_whitelisted(module, whitelist):
    return isinstance(module, tuple(whitelist))


class ASP:
    __model = None
    __optimizer = None
    __sparse_parameters = []  # (module_name, module, p_name, param, mask)
    __calculate_mask = None
    __allow_recompute = False

    @classmethod
    def init_model_for_pruning(
        cls,
        model,
        mask_calculator="m4n2_1d",
        verbosity=2,
        whitelist=(torch.nn.Linear, torch.nn.Conv1d, torch.nn.Conv2d),
        allowed_layer_names=None,
        disallowed_layer_names=(),
        allow_recompute_mask=False,
        custom_layer_dict=None,
        allow_permutation=True,
    ):
        assert cls.__model is None, "ASP has been initialized already"
        cls.__model = model
        cls.__allow_recompute = allow_recompute_mask
        cls.__allow_permutation = allow_permutation

        if isinstance(mask_calculator, str):
            from .sparse_masklib import create_mask

            pattern = mask_calculator
            cls.__calculate_mask = lambda t: create_mask(t, pattern)
        else:
            cls.__calculate_mask = mask_calculator

        sparse_param_names = {"weight"}
        for name, mod in model.named_modules():
            if not _is_whitelisted(mod, whitelist):
                continue
            if allowed_layer_names is not None and name not in allowed_layer_names:
                continue
            if name in disallowed_layer_names:
                continue
            for p_name, p in mod.named_parameters(recurse=False):
                if p_name not in sparse_param_names:
                    continue
                if p.dim() < 2 or p.shape[-1] % 4 != 0:
                    continue  # cannot form 4-wide groups
                mask = torch.ones_like(p, dtype=torch.bool)
                buf_name = p_name.split(".")[-1] + "_mma_mask"
                mod.register_buffer(buf_name, mask)
                cls.__sparse_parameters.append((name, mod, p_name, p, buf_name))

    @classmethod
    def init_optimizer_for_pruning(cls, optimizer):
        assert cls.__optimizer is None, "ASP optimizer has been initialized already"
        cls.__optimizer = optimizer
        old_step = optimizer.step

        def patched_step(self_opt, *args, **kwargs):
            out = old_step(*args, **kwargs)
            with torch.no_grad():
                for _, mod, p_name, p, buf_name in cls.__sparse_parameters:
                    p.mul_(getattr(mod, buf_name).to(p.dtype))
            return out

        optimizer.step = types.MethodType(patched_step, optimizer)

    @classmethod
    def compute_sparse_masks(cls):
        if getattr(cls, "_ASP__allow_permutation", False):
            # re-parameterize first: permute channel groups so the 2:4 masks
            # keep more magnitude (model function unchanged)
            from .permutation_lib import permute_model_for_sparsity

            permute_model_for_sparsity(cls.__model)
        with torch.no_grad():
            for _, mod, p_name, p, buf_name in cls.__sparse_parameters:
                mask = cls.__calculate_mask(p)
                getattr(mod, buf_name).copy_(mask)
                p.mul_(mask.to(p.dtype))

    @classmethod
    def restore_pruned_weights(cls):
        with torch.no_grad():
            for _, mod, p_name, p, buf_name in cls.__sparse_parameters:
                getattr(mod, buf_name).fill_(True)

    @classmethod
    def is_sparsity_enabled(cls):
        return len(cls.__sparse_parameters) > 0

    @classmethod
    def prune_trained_model(cls, model, optimizer):
        cls.init_model_for_pruning(model, mask_calculator="m4n2_1d", verbosity=2,
                                   whitelist=(torch.nn.Linear, torch.nn.Conv2d),
                                   allow_recompute_mask=False, allow_permutation=False)
        cls.init_optimizer_for_pruning(optimizer)
        cls.compute_sparse_masks()

    @classmethod
    def _reset(cls):
        """test helper — ASP keeps class-level state like the reference"""
        cls.__model = None
        cls.__optimizer = None
        cls.__sparse_parameters = []
        cls.__calculate_mask = None
        cls.__allow_permutation = False
