"""Localize the NaN in COBRA's pure-bf16 GraphedTrainStep path.

Run on GPU: python -u tools/repro_cobra_nan.py
Runs the fused step EAGERLY (use_graph=False), sweeps loss/grads/masters
for the first non-finite value, then re-runs the offending step with
forward hooks to name the first non-finite activation.
"""

import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch


def main():
    from genrec_amd.data.cobra_synthetic import (SyntheticCobraDataset,
                                                 cobra_collate_fn)
    from genrec_amd.models.cobra import Cobra
    from genrec_amd.parallel.graph_runner import GraphedTrainStep

    torch.manual_seed(42)
    dev = "cuda:0"
    ds = SyntheticCobraDataset(num_users=500, num_items=2000, split="train")
    ehd = int(os.environ.get("NAN_EHD", "384"))
    ehds = int(os.environ.get("NAN_EHEADS", "6"))
    m = Cobra(encoder_n_layers=2, encoder_hidden_dim=ehd,
              encoder_num_heads=ehds, encoder_vocab_size=32128,
              id_vocab_size=256, n_codebooks=3, d_model=384,
              decoder_n_layers=4, decoder_num_heads=6,
              decoder_dropout=0.1).to(dev)
    print(f"encoder {ehd}/{ehds} heads", flush=True)
    if os.environ.get("NAN_NODROP", "0") == "1":
        import torch.nn as nn
        for mod in m.modules():
            if isinstance(mod, nn.Dropout):
                mod.p = 0.0
            if isinstance(mod, nn.MultiheadAttention):
                mod.dropout = 0.0
        for layer in m.decoder.layers:
            layer.dropout_p = 0.0
        print("all dropout off", flush=True)
    m.static_infonce = True
    fixed = ds.max_items_per_seq + 1
    bs = 32

    def mk(i0):
        b = cobra_collate_fn([ds[(i0 + j) % len(ds)] for j in range(bs)],
                             m.pad_id, 3, train=True, fixed_items=fixed)
        return {"input_ids": b["input_ids"].to(dev),
                "encoder_input_ids": b["encoder_input_ids"].to(dev)}

    if os.environ.get("NAN_SDPA_MATH", "0") == "1":
        torch.backends.cuda.enable_flash_sdp(False)
        torch.backends.cuda.enable_mem_efficient_sdp(False)
        print("sdpa: math only", flush=True)
    loss_mode = os.environ.get("NAN_LOSS", "both")  # both|sparse|dense

    def get_loss(o):
        if loss_mode == "sparse":
            return o.loss_sparse
        if loss_mode == "dense":
            return o.loss_dense
        return o.loss_sparse + o.loss_dense

    use_graph = os.environ.get("NAN_GRAPH", "0") == "1"
    runner = GraphedTrainStep(
        m, mk(0), loss_getter=get_loss,
        lr=3e-4, weight_decay=0.01, clip_norm=1.0, world=1,
        use_graph=use_graph)
    print("captured:", runner.captured, flush=True)
    if os.environ.get("NAN_LR0", "0") == "1":
        runner.set_lr(0.0)
        print("lr pinned to 0 (params frozen)", flush=True)

    def bad(t):
        return not torch.isfinite(t.float()).all()

    off = []
    o = 0
    for p in runner.params:
        off.append((o, o + p.numel()))
        o += p.numel()
    names = [n for n, p in m.named_parameters() if p.requires_grad]

    n_steps = int(os.environ.get("NAN_STEPS", "30"))
    for step in range(n_steps):
        batch = mk(step * bs)
        loss = runner.step(batch)
        lf = float(loss.detach().float())
        gbad = bad(runner.flat_grads)
        mbad = bad(runner.flat_master)
        print(f"step {step:3d} loss {lf:9.4f} grads_bad={gbad} "
              f"master_bad={mbad}", flush=True)
        if mbad or gbad:
            for nm, t in (("master", runner.flat_master),
                          ("params", runner.flat_params.float()),
                          ("m", runner.m), ("v", runner.v),
                          ("grads", runner.flat_grads.float())):
                nf = ~torch.isfinite(t)
                cnt = int(nf.sum())
                if cnt:
                    idx = nf.nonzero().flatten()
                    print(f"  {nm}: {cnt} nonfinite, first={int(idx[0])} "
                          f"last={int(idx[-1])} of {t.numel()}", flush=True)
                else:
                    print(f"  {nm}: clean", flush=True)
        if lf != lf or gbad or mbad:
            fg = runner.flat_grads.float()
            fm = runner.flat_master
            for (a, b), n in zip(off, names):
                g = fg[a:b]
                mm = fm[a:b]
                if not torch.isfinite(g).all() or not torch.isfinite(mm).all():
                    print(f"  BAD {n}: grad finite={torch.isfinite(g).all()}"
                          f" absmax={g.abs().max():.3e} "
                          f"master finite={torch.isfinite(mm).all()}",
                          flush=True)
            # name the first non-finite activation
            hooks = []

            def mkhook(name):
                def h(mod, inp, out):
                    ts = out if isinstance(out, (tuple, list)) else [out]
                    for t in ts:
                        if torch.is_tensor(t) and t.is_floating_point() \
                                and not torch.isfinite(t.float()).all():
                            print(f"  ACT-NAN {name} "
                                  f"({type(mod).__name__})", flush=True)
                            break
                return h

            for n, mod in m.named_modules():
                hooks.append(mod.register_forward_hook(mkhook(n)))
            out = m(**batch)
            print(f"  refwd loss_sparse={float(out.loss_sparse):.4f} "
                  f"loss_dense={float(out.loss_dense):.4f}", flush=True)
            for h in hooks:
                h.remove()
            break
    print("done", flush=True)


if __name__ == "__main__":
    main()
