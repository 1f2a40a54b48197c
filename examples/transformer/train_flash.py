"""Minimal transformer training loop on the apex_amd fused stack.

Shows the round-2 fast path end to end (reference analogue: the Megatron-era
kernel consumers):

* flash attention (hand-written MFMA kernels; strided QKV views, fused
  philox attention dropout when ``--dropout`` > 0),
* amp O2 bf16 with ``overflow_check=False`` (static scale: no host sync in
  the step),
* ``FusedAdam(capturable=True)`` and whole-step hipGraph capture/replay,
* fused add+LayerNorm residual stream and fused cross-entropy.

Run:  python examples/transformer/train_flash.py [--steps 50] [--graph 0]
CPU smoke (tiny, eager): python examples/transformer/train_flash.py --cpu
"""

import argparse
import os
import sys
import time

sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)), "..", ".."))

import torch


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--steps", type=int, default=50)
    ap.add_argument("--batch", type=int, default=32)
    ap.add_argument("--seq-len", type=int, default=512)
    ap.add_argument("--dropout", type=float, default=0.0,
                    help="fused attention dropout (eager attention path if "
                         "combined with --graph: the philox seed would "
                         "freeze inside a captured graph)")
    ap.add_argument("--graph", type=int, default=1)
    ap.add_argument("--cpu", action="store_true")
    args = ap.parse_args()

    from apex_amd import amp
    from apex_amd.models.transformer import BertModel, TransformerLMConfig
    from apex_amd.optimizers import FusedAdam
    from apex_amd.contrib.xentropy import SoftmaxCrossEntropyLoss

    device = "cpu" if args.cpu else "cuda"
    if args.cpu:
        cfg = TransformerLMConfig(vocab_size=512, hidden=128, layers=2, heads=2,
                                  seq_len=64)
        args.batch, args.steps, args.graph = 2, 3, 0
    else:
        cfg = TransformerLMConfig(vocab_size=30528, hidden=768, layers=12,
                                  heads=12, seq_len=args.seq_len)

    torch.manual_seed(0)
    model = BertModel(cfg).to(device)
    use_graph = bool(args.graph) and not args.cpu and args.dropout == 0.0
    opt = FusedAdam(model.parameters(), lr=1e-4, weight_decay=0.01,
                    capturable=use_graph)
    model, opt = amp.initialize(model, opt, opt_level="O2",
                                cast_model_type=torch.bfloat16, loss_scale=1.0,
                                keep_batchnorm_fp32=False, verbosity=0,
                                overflow_check=not use_graph)

    tokens = torch.randint(0, cfg.vocab_size, (args.batch, cfg.seq_len), device=device)

    def step():
        opt.zero_grad()
        logits = model(tokens)
        loss = SoftmaxCrossEntropyLoss.apply(
            logits.reshape(-1, cfg.vocab_size).contiguous(),
            tokens.reshape(-1), 0.0, -1, True).mean()
        with amp.scale_loss(loss, opt) as scaled:
            scaled.backward()
        opt.step()
        return loss

    for _ in range(3):
        last = step()
    run = step
    if use_graph:
        torch.cuda.synchronize()
        g = torch.cuda.CUDAGraph()
        with torch.cuda.graph(g):
            step()
        g.replay()
        run = g.replay
        print("# whole-step hipGraph captured")

    if not args.cpu:
        torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(args.steps):
        run()
    if not args.cpu:
        torch.cuda.synchronize()
    dt = time.perf_counter() - t0
    toks = args.batch * cfg.seq_len * args.steps / dt
    print(f"loss(before timing)={float(last):.4f}  {toks:,.0f} tokens/s "
          f"({dt / args.steps * 1e3:.1f} ms/step, graph={use_graph})")


if __name__ == "__main__":
    main()
