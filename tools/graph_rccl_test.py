import os, sys
sys.path.insert(0, ".")
import torch
import torch.distributed as dist

os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
os.environ.setdefault("MASTER_PORT", "29781")
dist.init_process_group("nccl", rank=0, world_size=1)
try:
    from shifu_amd.models.wide_deep import WideDeep
    from shifu_amd.ops.flat import FlatParams, split_params, bind_mirrors
    from shifu_amd.ops.loss import weighted_loss
    from shifu_amd.ops.optim import FusedOptimizer
    from shifu_amd.parallel.dist import GradAggregator
    from shifu_amd.train.graph import GraphedStep

    torch.manual_seed(0)
    model = WideDeep(32, [3000] * 4, 16, [64], ["relu"], seed=1,
                     sharded_embeddings="table", world=1, rank=0,
                     unified=True).cuda()
    for p in model.parameters():
        if getattr(p, "_is_embedding_arena", False):
            p.data = p.data.to(torch.bfloat16)
    dps, eps_ = split_params(model)
    flat = FlatParams(dps, mirror_bf16=True)
    bind_mirrors(model, flat)
    agg = GradAggregator(flat, eps_, bucket_mb=8)
    opt = FusedOptimizer(flat, eps_, optimizer="adam", lr=1e-3,
                         emb_optimizer="adagrad", emb_lr=0.05)
    B = 256
    dense = torch.randn(B, 32, device="cuda").to(torch.bfloat16)
    cats = torch.randint(0, 3000, (B, 4), device="cuda")
    y = (torch.rand(B, device="cuda") > 0.5).float()
    w = torch.ones(B, device="cuda")

    def body():
        loss = weighted_loss(model(dense, cats), y, w, "sigmoid_ce")
        loss.backward()
        agg.finish()
        opt.step()
        opt.zero_grad()
        return loss

    stepper = GraphedStep(body, warmup=3)
    stepper.capture()
    l1 = float(stepper.run())
    l2 = float(stepper.run())
    torch.cuda.synchronize()
    print(f"GRAPH_RCCL_OK loss1={l1:.4f} loss2={l2:.4f}")
finally:
    dist.destroy_process_group()
