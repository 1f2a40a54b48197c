"""Network-wide channel-permutation propagation for 2:4 sparsity.

Counterpart of the reference ``apex.contrib.sparsity.permutation_lib``
(Permutation class): a permutation that improves a pruned layer's 2:4 kept
magnitude must be applied to the layer's INPUT channels and, consistently,
to the OUTPUT channels of every producer that feeds those channels — through
any permutation-equivariant ops in between — so the model function is
unchanged. The reference walks its own JSON graph dump; here we trace with
``torch.fx`` and walk the node graph directly.

Scope (conservative by construction — a group is only permuted when its
whole closure is understood):

* consumers: ``nn.Linear`` (input dim 1), 1x1 ``nn.Conv2d`` (input dim 1)
* producers: ``nn.Linear`` / 1x1 ``nn.Conv2d`` (output dim 0 + bias),
  ``nn.Embedding`` (dim 1)
* transparent ops: elementwise activations, dropout, ``+`` (residual),
  ``nn.LayerNorm`` / FusedLayerNorm / FusedRMSNorm (their affine params are
  permuted along with the channel dim)
* a group is rejected if any node in its closure has a user outside the
  closure (the permuted values would leak into un-permuted consumers), or
  if the walk reaches a model input or an op not listed above.
"""

import operator

import torch
import torch.fx

from .permutation_search import exhaustive_search, search_for_good_permutation

_ACT_FUNCTIONS = {
    torch.relu, torch.tanh, torch.sigmoid, torch.nn.functional.relu,
    torch.nn.functional.gelu, torch.nn.functional.silu,
    torch.nn.functional.tanh, torch.nn.functional.sigmoid,
    torch.nn.functional.dropout,
}
_ADD_FUNCTIONS = {operator.add, torch.add}
_ACT_METHODS = {"relu", "tanh", "sigmoid"}


def _norm_types():
    types = [torch.nn.LayerNorm]
    try:
        from ...normalization import FusedLayerNorm, FusedRMSNorm

        types += [FusedLayerNorm, FusedRMSNorm]
    except Exception:
        pass
    return tuple(types)


def _is_transparent_module(mod):
    return isinstance(mod, (torch.nn.ReLU, torch.nn.GELU, torch.nn.SiLU,
                            torch.nn.Tanh, torch.nn.Sigmoid, torch.nn.Dropout,
                            torch.nn.Identity))


def _is_consumer(mod):
    if isinstance(mod, torch.nn.Linear):
        return mod.weight.shape[1] % 4 == 0
    if isinstance(mod, torch.nn.Conv2d) and mod.kernel_size == (1, 1) and mod.groups == 1:
        return mod.weight.shape[1] % 4 == 0
    return False


def _is_producer(mod):
    return isinstance(mod, (torch.nn.Linear, torch.nn.Embedding)) or (
        isinstance(mod, torch.nn.Conv2d) and mod.kernel_size == (1, 1) and mod.groups == 1)


class _Group:
    def __init__(self):
        self.consumers = []   # modules: weight[:, perm]
        self.producers = []   # modules: weight[perm, :] (+ bias) or Embedding dim 1
        self.norms = []       # modules: weight/bias[perm]
        self.nodes = set()    # fx nodes inside the closure (producer outputs
                              # up to, not including, the consumer call)
        self.consumer_nodes = set()
        self.ok = True


def _walk(node, gm, group, visiting):
    """Backward walk from a consumer's input; records producers/norms and the
    transparent closure. Sets group.ok=False on anything not understood."""
    if node in visiting:
        return
    visiting.add(node)
    if node.op == "call_module":
        mod = gm.get_submodule(node.target)
        if _is_producer(mod):
            group.producers.append(mod)
            group.nodes.add(node)
            return
        if isinstance(mod, _norm_types()):
            if getattr(mod, "weight", None) is not None:
                group.norms.append(mod)
            group.nodes.add(node)
            _walk(node.args[0], gm, group, visiting)
            return
        if _is_transparent_module(mod):
            group.nodes.add(node)
            _walk(node.args[0], gm, group, visiting)
            return
        group.ok = False
        return
    if node.op == "call_function":
        if node.target in _ADD_FUNCTIONS:
            group.nodes.add(node)
            for a in node.args:
                if isinstance(a, torch.fx.Node):
                    _walk(a, gm, group, visiting)
            return
        if node.target in _ACT_FUNCTIONS:
            group.nodes.add(node)
            _walk(node.args[0], gm, group, visiting)
            return
        group.ok = False
        return
    if node.op == "call_method" and node.target in _ACT_METHODS:
        group.nodes.add(node)
        _walk(node.args[0], gm, group, visiting)
        return
    # placeholder (model input), get_attr, output, anything else
    group.ok = False


def find_permutation_groups(model, example_inputs=None):
    """Trace ``model`` and return a list of closed, permutable channel
    groups. Groups sharing a producer are merged; groups whose closure
    leaks to an outside user are dropped."""
    gm = torch.fx.symbolic_trace(model)
    groups = []
    for node in gm.graph.nodes:
        if node.op != "call_module":
            continue
        mod = gm.get_submodule(node.target)
        if not _is_consumer(mod):
            continue
        g = _Group()
        g.consumers.append(mod)
        g.consumer_nodes.add(node)
        _walk(node.args[0], gm, g, set())
        if g.ok and g.producers:
            groups.append(g)

    # merge groups that share any producer or closure node (residual streams)
    merged = []
    for g in groups:
        target = None
        for m in merged:
            if (set(map(id, g.producers)) & set(map(id, m.producers))) or (g.nodes & m.nodes):
                target = m
                break
        if target is None:
            merged.append(g)
        else:
            target.consumers.extend(c for c in g.consumers
                                    if id(c) not in set(map(id, target.consumers)))
            target.producers.extend(p for p in g.producers
                                    if id(p) not in set(map(id, target.producers)))
            target.norms.extend(n for n in g.norms
                                if id(n) not in set(map(id, target.norms)))
            target.nodes |= g.nodes
            target.consumer_nodes |= g.consumer_nodes

    # closure check: every user of every closure node must stay inside the
    # closure or be one of the group's consumers
    out = []
    for g in merged:
        closed = True
        for n in g.nodes:
            for user in n.users:
                if user not in g.nodes and user not in g.consumer_nodes:
                    closed = False
                    break
            if not closed:
                break
        if closed:
            out.append(g)
    return out


def _consumer_weight2d(mod):
    w = mod.weight.detach()
    if w.dim() == 4:  # 1x1 conv
        w = w.reshape(w.shape[0], w.shape[1])
    return w


@torch.no_grad()
def _apply_group_permutation(g, perm):
    for mod in g.consumers:
        if mod.weight.dim() == 4:
            mod.weight.copy_(mod.weight[:, perm, :, :])
        else:
            mod.weight.copy_(mod.weight[:, perm])
    for mod in g.producers:
        if isinstance(mod, torch.nn.Embedding):
            mod.weight.copy_(mod.weight[:, perm])
        else:
            if mod.weight.dim() == 4:
                mod.weight.copy_(mod.weight[perm, :, :, :])
            else:
                mod.weight.copy_(mod.weight[perm, :])
            if mod.bias is not None:
                mod.bias.copy_(mod.bias[perm])
    for mod in g.norms:
        if getattr(mod, "weight", None) is not None:
            mod.weight.copy_(mod.weight[perm])
        if getattr(mod, "bias", None) is not None:
            mod.bias.copy_(mod.bias[perm])


def permute_model_for_sparsity(model, strategy="exhaustive", verbose=False):
    """Find permutable channel groups and permute each to maximize the 2:4
    kept magnitude summed over the group's consumer weights. The model
    function is unchanged (pure re-parameterization). Returns the number of
    groups permuted."""
    try:
        groups = find_permutation_groups(model)
    except Exception as e:  # symbolic_trace can fail on dynamic control flow
        if verbose:
            print(f"permutation search skipped (trace failed: {e})")
        return 0
    applied = 0
    for g in groups:
        wcat = torch.cat([_consumer_weight2d(m).float() for m in g.consumers], dim=0)
        if strategy == "exhaustive":
            perm = exhaustive_search(wcat)
        else:
            perm = search_for_good_permutation(wcat, strategy=strategy)
        if torch.equal(perm, torch.arange(len(perm))):
            continue
        _apply_group_permutation(g, perm.to(g.consumers[0].weight.device))
        applied += 1
        if verbose:
            print(f"permuted group: {len(g.consumers)} consumer(s), "
                  f"{len(g.producers)} producer(s), {len(g.norms)} norm(s)")
    return applied
