import os, sys, functools, torch
sys.path.insert(0, ".")
import adanet_amd
from adanet_amd.distributed import RoundRobinStrategy
from adanet_amd.head import MultiClassHead
from adanet_amd.models import simple_dnn
from adanet_amd.ops.optim import FusedSGD

dev = torch.device("cuda:0")
D, C, B = 3072, 10, 2048
torch.manual_seed(1234)
teacher = torch.randn(D, C)
pool = []
for i in range(8):
    x = torch.randn(B, D)
    y = (x @ teacher).argmax(1)
    x = x.to(dev).to(torch.bfloat16); y = y.to(dev)
    x.adanet_cache_key = ("t", i)
    pool.append((x, y))

def input_fn():
    def gen():
        i = 0
        while True:
            yield pool[i % 8]; i += 1
    return gen()

gen = simple_dnn.Generator(
    optimizer_fn=functools.partial(FusedSGD, lr=0.05, momentum=0.9),
    mixture_optimizer_fn=functools.partial(FusedSGD, lr=0.005),
    layer_size=2048, initial_num_layers=1, learn_mixture_weights=True,
    dropout=0.1, seed=77)
est = adanet_amd.Estimator(
    head=MultiClassHead(C), subnetwork_generator=gen,
    max_iteration_steps=150, force_grow=True, adanet_lambda=1e-4,
    model_dir="/tmp/nan_inspect",
    config=adanet_amd.RunConfig(tf_random_seed=42, device="cuda:0",
                                log_step_count_steps=10**9))
for nsteps in (4, 8, 16, 24, 40):
    est.train(input_fn, steps=(nsteps - est.global_step))
    it = est._current_iteration
    it.flush_losses()
    print("== step", est.global_step, "graph", it._graph is not None)
    for spec in it.subnetwork_specs:
        ol = spec.out_logits
        print("  sub", spec.name, "loss", round(spec.last_loss, 4),
              "out_nan", int(torch.isnan(ol.float()).sum()) if ol is not None else "-")
    for spec, cand in zip(it.ensemble_specs, it.candidates):
        w = getattr(spec.ensemble, "mixture_weights", None) if spec.ensemble is not None else None
        print("  ens", spec.name, "ema", cand.adanet_loss,
              "w", w.detach().float().tolist() if w is not None else None,
              "wgrad_nan", int(torch.isnan(w.grad.float()).sum()) if (w is not None and w.grad is not None) else "-")
