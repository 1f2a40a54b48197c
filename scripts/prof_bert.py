"""rocprofv3 target: a few BERT-base training steps (kernel breakdown for the
transformer path: fused_dense GEMMs, softmax, layernorm, adam)."""

import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch


def main():
    from apex_amd import amp
    from apex_amd.models.transformer import BertModel, bert_base_config
    from apex_amd.optimizers import FusedAdam
    from apex_amd.contrib.xentropy import SoftmaxCrossEntropyLoss

    device = "cuda"
    torch.manual_seed(0)
    cfg = bert_base_config(seq_len=512)
    model = BertModel(cfg).to(device)
    opt = FusedAdam(model.parameters(), lr=1e-4, weight_decay=0.01)
    model, opt = amp.initialize(model, opt, opt_level="O2", cast_model_type=torch.bfloat16,
                                loss_scale=1.0, keep_batchnorm_fp32=False, verbosity=0)
    batch = 32
    tokens = torch.randint(0, cfg.vocab_size, (batch, cfg.seq_len), device=device)
    mask = torch.zeros(batch, 1, cfg.seq_len, cfg.seq_len, dtype=torch.bool, device=device)

    def step():
        opt.zero_grad()
        logits = model(tokens, mask)
        losses = SoftmaxCrossEntropyLoss.apply(
            logits.reshape(-1, cfg.vocab_size).contiguous(), tokens.reshape(-1), 0.0, -1, True
        )
        loss = losses.mean()
        with amp.scale_loss(loss, opt) as scaled:
            scaled.backward()
        opt.step()

    import time
    nwarm = int(os.environ.get("PROF_WARMUP", "3"))
    nsteps = int(os.environ.get("PROF_STEPS", "5"))
    for _ in range(nwarm):
        step()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(nsteps):
        step()
    torch.cuda.synchronize()
    print(f"bert step: {(time.perf_counter() - t0) / nsteps * 1000:.1f} ms")


if __name__ == "__main__":
    main()
