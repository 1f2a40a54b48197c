"""apex_amd.mlp.MLP — an N-layer perceptron executed as one fused call.

API parity with the reference ``apex.mlp`` (apex/mlp/mlp.py:11-111):
``MlpFunction`` takes flat ``(input, w0..wn, b0..bn)``; the device call
returns the output plus one flat reserved buffer holding every intermediate
activation; backward replays from the reserved buffer. Activations:
none / relu / sigmoid.

Device path (csrc/mlp.hip): hipBLASLt GEMM with BIAS / RELU_AUX_BIAS
epilogues per layer, bias+activation gradients fused into the dgrad GEMM
epilogues on the way back. CPU path: reference torch math.
"""

import torch

from .._ext import get_ext

_ACT = {"none": 0, "relu": 1, "sigmoid": 2}


class MlpFunction(torch.autograd.Function):
    @staticmethod
    def forward(ctx, bias, activation, *args):
        input = args[0]
        if input.is_cuda:
            ext = get_ext("mlp")
            output, reserved = ext.forward(bias, activation, list(args))
            ctx.save_for_backward(*args, reserved)
        else:
            nlayers = (len(args) - 1) // 2 if bias else len(args) - 1
            weights = args[1:nlayers + 1]
            biases = args[nlayers + 1:] if bias else [None] * nlayers
            x = input
            inters = []
            for w, b in zip(weights, biases):
                x = torch.nn.functional.linear(x, w, b)
                # activation applies to every layer incl. the last (reference
                # semantics, cf. tests/L0/run_mlp/test_mlp.py ref Sequential)
                if activation == 1:
                    x = torch.relu(x)
                elif activation == 2:
                    x = torch.sigmoid(x)
                inters.append(x)
            output = inters[-1]
            reserved = torch.cat([t.reshape(-1) for t in inters]) if inters else input.new_empty(0)
            ctx.save_for_backward(*args, reserved)
        ctx.bias = bias
        ctx.activation = activation
        return output

    @staticmethod
    def backward(ctx, grad_o):
        saved = ctx.saved_tensors
        args, reserved = saved[:-1], saved[-1]
        input = args[0]
        if input.is_cuda:
            ext = get_ext("mlp")
            grads = ext.backward(ctx.bias, ctx.activation, grad_o.contiguous(), reserved, list(args))
            return (None, None, *grads)
        # CPU reference backward
        bias, activation = ctx.bias, ctx.activation
        nlayers = (len(args) - 1) // 2 if bias else len(args) - 1
        weights = list(args[1:nlayers + 1])
        x = input
        acts = [x]
        offset = 0
        for i in range(nlayers):
            numel = x.shape[0] * weights[i].shape[0]
            y = reserved[offset:offset + numel].view(x.shape[0], weights[i].shape[0])
            offset += numel
            acts.append(y)
            x = y
        dy = grad_o
        wgrads = [None] * nlayers
        bgrads = [None] * nlayers
        for i in reversed(range(nlayers)):
            y = acts[i + 1]
            if activation == 1:
                dy = dy * (y > 0).to(dy.dtype)
            elif activation == 2:
                dy = dy * y * (1 - y)
            wgrads[i] = dy.t() @ acts[i]
            if bias:
                bgrads[i] = dy.sum(0)
            dy = dy @ weights[i]
        grad_input = dy
        out = [grad_input] + wgrads + (bgrads if bias else [])
        return (None, None, *out)


def mlp_function(bias, activation, *args):
    return MlpFunction.apply(bias, activation, *args)


class MLP(torch.nn.Module):
    """Launch an MLP in one fused call (reference: apex/mlp/mlp.py:33-111).

    mlp_sizes: [input_features, hidden1, ..., output_features]
    """

    def __init__(self, mlp_sizes, bias=True, relu=True, activation=None):
        super().__init__()
        self.num_layers = len(mlp_sizes) - 1
        self.mlp_sizes = list(mlp_sizes)
        self.bias = 1 if bias else 0
        if activation is None:
            activation = "relu" if relu else "none"
        if activation not in _ACT:
            raise TypeError(f"activation must be relu or none or sigmoid, got {activation}")
        self.activation = _ACT[activation]

        self.weights = []
        self.biases = []
        for i in range(self.num_layers):
            w = torch.nn.Parameter(torch.empty(mlp_sizes[i + 1], mlp_sizes[i]))
            self.weights.append(w)
            setattr(self, f"weight_{i}", w)
            if self.bias:
                b = torch.nn.Parameter(torch.empty(mlp_sizes[i + 1]))
                self.biases.append(b)
                setattr(self, f"bias_{i}", b)
        self.reset_parameters()

    def reset_parameters(self):
        for weight in self.weights:
            dimsum = weight.size(0) + weight.size(1)
            std = (2.0 / dimsum) ** 0.5
            torch.nn.init.normal_(weight, 0.0, std)
        for bias in self.biases:
            std = (2.0 / bias.size(0)) ** 0.5
            torch.nn.init.normal_(bias, 0.0, std)

    def forward(self, input):
        return mlp_function(self.bias, self.activation, input, *self.weights, *self.biases)

    def extra_repr(self):
        return f"MLP sizes: {self.mlp_sizes}, Bias={self.bias}, activation={self.activation}"
