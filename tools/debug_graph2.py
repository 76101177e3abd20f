"""Replicate bench.py's graph path exactly, with env toggles to bisect.

Toggles (1 = keep bench behavior):
  V_INIT_DIST, V_WD, V_CPU_BATCHES, V_PRE_BWD_STATIC, V_WARMUP3
"""

import os
import sys

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def flag(name, default="1"):
    return os.environ.get(name, default) == "1"


def main():
    import bench as benchmod
    from genrec_amd.models.tiger import Tiger

    if flag("V_INIT_DIST"):
        from genrec_amd.parallel import init_distributed

        ctx = init_distributed()
        device = ctx.device
    else:
        device = torch.device("cuda:0")
    torch.manual_seed(1234)

    cfg = dict(embedding_dim=128, attn_dim=384, dropout=0.1, num_heads=6,
               n_layers=8, num_item_embeddings=256, num_user_embeddings=10000,
               sem_id_dim=3)
    model = Tiger(**cfg).to(device)

    if flag("V_CPU_BATCHES"):
        batches = [benchmod.build_batch(256, 20, 3, 256, 10000, device,
                                        seed=100 + i) for i in range(8)]
    else:
        B, L = 256, 60
        batches = []
        for i in range(8):
            batches.append(dict(
                user_input_ids=torch.randint(0, 10000, (B, 1), device=device),
                item_input_ids=torch.randint(0, 256, (B, L), device=device),
                token_type_ids=(torch.arange(L, device=device) % 3)
                .unsqueeze(0).expand(B, -1).contiguous(),
                target_input_ids=torch.randint(0, 256, (B, 3), device=device),
                target_token_type_ids=torch.arange(3, device=device)
                .unsqueeze(0).expand(B, -1).contiguous(),
                seq_mask=torch.ones(B, L, dtype=torch.long, device=device),
            ))

    amp = torch.autocast(device_type="cuda", dtype=torch.bfloat16, cache_enabled=False)
    wd = 0.035 if flag("V_WD") else 0.0
    opt = torch.optim.AdamW(model.parameters(), lr=1e-4, weight_decay=wd,
                            capturable=True, foreach=True)
    model.train()
    params = [p for p in model.parameters() if p.requires_grad]
    static = {k: v.clone() for k, v in batches[0].items()}

    if flag("V_PRE_BWD_STATIC"):
        with amp:
            model(**static).loss.backward()
    flat_grads = torch.zeros(sum(p.numel() for p in params), device=device)
    off = 0
    for p in params:
        p.grad = flat_grads[off:off + p.numel()].view_as(p)
        off += p.numel()

    def inner_step():
        flat_grads.zero_()
        with amp:
            out = model(**static)
        out.loss.backward()
        norm = flat_grads.norm()
        flat_grads.mul_(torch.clamp(1.0 / (norm + 1e-6), max=1.0))
        opt.step()
        return out.loss

    n_warm = 3 if flag("V_WARMUP3") else 5
    s = torch.cuda.Stream()
    s.wait_stream(torch.cuda.current_stream())
    with torch.cuda.stream(s):
        for _ in range(n_warm):
            inner_step()
    torch.cuda.current_stream().wait_stream(s)
    print("# warmup ok", flush=True)
    graph = torch.cuda.CUDAGraph()
    with torch.cuda.graph(graph):
        loss = inner_step()
    print("# capture ok", flush=True)
    for i in range(6):
        b = batches[i % len(batches)]
        for key in static:
            static[key].copy_(b[key], non_blocking=True)
        graph.replay()
        torch.cuda.synchronize()
        print(f"# replay {i} ok loss={loss.item():.4f}", flush=True)
    print("# ALL OK", flush=True)


if __name__ == "__main__":
    main()
