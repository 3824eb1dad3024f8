"""bf16 training-parity curve: ~60 steps of the headline
config on the sodkit CDNA4 kernel stack vs the stock-PyTorch twin model
(tools/bench_torch_native.MINetTorch — MIOpen/rocBLAS eager) with
IDENTICAL initial weights, data sequence and SGD hyperparameters.

The two loss curves answer "correct AND fast": if the hand-written
bf16 kernels (conv fwd/dgrad/wgrad, fused BN, bilinear, fused loss,
fused SGD) implement the same math, the trajectories track each other
within bf16 accumulation noise.

Run on the GPU box: python tools/train_parity.py [steps]
Writes gpurun_out/train_parity.json.
"""
import importlib.util
import json
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch  # noqa: E402

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def load_twin_module():
    spec = importlib.util.spec_from_file_location(
        "btn", os.path.join(REPO, "tools", "bench_torch_native.py"))
    btn = importlib.util.module_from_spec(spec)
    spec.loader.exec_module(btn)
    return btn


def copy_weights_to_twin(sod_model, twin):
    """Ordered 1:1 state-dict copy (verified 532/532 aligned): sodkit conv
    weights are KRSC [Cout,R,S,Cin]; torch wants [Cout,Cin,R,S]."""
    ssd = sod_model.state_dict()
    tsd = twin.state_dict()
    tkeys = [k for k in tsd if "num_batches" not in k]
    skeys = list(ssd.keys())
    assert len(tkeys) == len(skeys), (len(tkeys), len(skeys))
    new = {}
    for tk, sk in zip(tkeys, skeys):
        v = ssd[sk]
        if v.dim() == 4:
            v = v.permute(0, 3, 1, 2).contiguous()
        assert v.shape == tsd[tk].shape, (tk, sk, v.shape, tsd[tk].shape)
        new[tk] = v
    twin.load_state_dict(new, strict=False)


def make_batches(steps, batch, size, seed=1):
    g = torch.Generator().manual_seed(seed)
    out = []
    for _ in range(steps):
        x = torch.rand(batch, 3, size, size, generator=g)
        t = (torch.rand(batch, 1, size, size, generator=g) > 0.5).float()
        out.append((x, t))
    return out


def run_sodkit(batches, lr, wd, mu):
    from sodkit.models import build_model, convert_precision
    from sodkit.ops import SodSGD, split_decay_groups, hybrid_loss
    from sodkit.parallel import GradReducer
    torch.manual_seed(0)
    m = build_model("minet_res50")
    base = {k: v.clone() for k, v in m.state_dict().items()}
    m = m.cuda()
    convert_precision(m, torch.bfloat16)
    red = GradReducer(m)
    decay, no_decay = split_decay_groups(m)
    opt = SodSGD([{"params": decay, "weight_decay": wd},
                  {"params": no_decay, "weight_decay": 0.0}],
                 momentum=mu)
    losses = []
    for x, t in batches:
        xg = x.cuda().to(torch.bfloat16).contiguous(
            memory_format=torch.channels_last)
        tg = t.cuda()
        red.zero_grad()
        loss = hybrid_loss(m(xg), tg)
        loss.backward()
        red.finalize()
        opt.step(lr=lr, grad_scale=1.0)
        losses.append(float(loss.item()))
    return losses, base


def run_twin(batches, base_state, lr, wd, mu):
    btn = load_twin_module()
    from sodkit.ops import reference
    from sodkit.ops.sgd import split_decay_groups
    import torch.nn as nn
    torch.manual_seed(0)
    m = btn.MINetTorch()
    # identical initial weights (from the sodkit model's init)
    class _Shim:
        def state_dict(self):
            return base_state
    copy_weights_to_twin(_Shim(), m)
    m = m.cuda().to(memory_format=torch.channels_last).to(torch.bfloat16)
    for mod in m.modules():
        if isinstance(mod, nn.BatchNorm2d):
            mod.float()
    decay, no_decay = split_decay_groups(m)
    opt = torch.optim.SGD(
        [{"params": decay, "weight_decay": wd},
         {"params": no_decay, "weight_decay": 0.0}],
        lr=lr, momentum=mu, foreach=True)
    losses = []
    for x, t in batches:
        xg = x.cuda().to(torch.bfloat16).contiguous(
            memory_format=torch.channels_last)
        tg = t.cuda()
        opt.zero_grad(set_to_none=True)
        loss = reference.hybrid_loss(m(xg), tg)
        loss.backward()
        opt.step()
        losses.append(float(loss.item()))
    return losses


def main():
    steps = int(sys.argv[1]) if len(sys.argv) > 1 else 60
    batch, size = 8, 192   # headline arch; smaller batch/res to fit the
    #                        twin + sodkit runs in one short GPU window
    lr, wd, mu = 0.01, 5e-4, 0.9
    batches = make_batches(steps, batch, size)
    sod_losses, base = run_sodkit(batches, lr, wd, mu)
    twin_losses = run_twin(batches, base, lr, wd, mu)
    rel = [abs(a - b) / max(abs(b), 1e-6)
           for a, b in zip(sod_losses, twin_losses)]
    out = {
        "config": {"model": "minet_res50", "batch": batch, "size": size,
                   "steps": steps, "lr": lr, "wd": wd, "momentum": mu,
                   "dtype": "bf16"},
        "sodkit_losses": sod_losses,
        "torch_losses": twin_losses,
        "max_rel_diff": max(rel),
        "final_rel_diff": rel[-1],
        "sod_first_last": [sod_losses[0], sod_losses[-1]],
        "twin_first_last": [twin_losses[0], twin_losses[-1]],
        "both_decreased": sod_losses[-1] < sod_losses[0] * 0.7
                          and twin_losses[-1] < twin_losses[0] * 0.7,
    }
    os.makedirs(os.path.join(REPO, "gpurun_out"), exist_ok=True)
    with open(os.path.join(REPO, "gpurun_out", "train_parity.json"), "w") as f:
        json.dump(out, f, indent=1)
    print("TRAIN_PARITY", json.dumps({k: out[k] for k in
          ("max_rel_diff", "final_rel_diff", "sod_first_last",
           "twin_first_last", "both_decreased")}))


if __name__ == "__main__":
    main()
