"""Stock-PyTorch (MIOpen/rocBLAS eager) equivalent of the bench step, for an
apples-to-apples comparison against the sodkit CDNA4 kernels: same MINet-R50
architecture built from torch.nn primitives, channels_last bf16, same hybrid
loss math, torch SGD.  Run: python tools/bench_torch_native.py [steps]"""
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch
import torch.nn as nn
import torch.nn.functional as F

from sodkit.ops import reference


class ConvBNReLU(nn.Sequential):
    def __init__(self, cin, cout, k, s=1, p=0, d=1, relu=True):
        layers = [nn.Conv2d(cin, cout, k, s, p, d, bias=False),
                  nn.BatchNorm2d(cout)]
        if relu:
            layers.append(nn.ReLU(inplace=True))
        super().__init__(*layers)


class Bottleneck(nn.Module):
    def __init__(self, cin, planes, stride=1):
        super().__init__()
        cout = planes * 4
        self.c1 = ConvBNReLU(cin, planes, 1)
        self.c2 = ConvBNReLU(planes, planes, 3, stride, 1)
        self.c3 = nn.Conv2d(planes, cout, 1, bias=False)
        self.bn3 = nn.BatchNorm2d(cout)
        self.down = (nn.Sequential(nn.Conv2d(cin, cout, 1, stride, bias=False),
                                   nn.BatchNorm2d(cout))
                     if stride != 1 or cin != cout else None)

    def forward(self, x):
        idn = self.down(x) if self.down is not None else x
        return F.relu(self.bn3(self.c3(self.c2(self.c1(x)))) + idn)


class ResNet50(nn.Module):
    out_channels = [64, 256, 512, 1024, 2048]

    def __init__(self):
        super().__init__()
        self.stem = ConvBNReLU(3, 64, 7, 2, 3)
        cin = 64
        stages = []
        for planes, n, st in zip([64, 128, 256, 512], [3, 4, 6, 3], [1, 2, 2, 2]):
            blocks = [Bottleneck(cin, planes, st)]
            cin = planes * 4
            blocks += [Bottleneck(cin, planes) for _ in range(n - 1)]
            stages.append(nn.Sequential(*blocks))
        self.layers = nn.ModuleList(stages)

    def forward(self, x):
        c1 = self.stem(x)
        f = F.max_pool2d(c1, 3, 2, 1)
        out = [c1]
        for l in self.layers:
            f = l(f)
            out.append(f)
        return out


def up(x, ref):
    return F.interpolate(x, size=ref.shape[2:], mode="bilinear",
                         align_corners=False)


class AIM(nn.Module):
    def __init__(self, ch, lo, hi):
        super().__init__()
        self.m = ConvBNReLU(ch, ch, 3, 1, 1)
        self.l = ConvBNReLU(ch, ch, 3, 2, 1) if lo else None
        self.h = ConvBNReLU(ch, ch, 3, 1, 1) if hi else None
        self.fc = nn.Conv2d(ch, ch, 3, 1, 1, bias=False)
        self.fb = nn.BatchNorm2d(ch)

    def forward(self, lo, mid, hi):
        y = self.m(mid)
        if self.l is not None and lo is not None:
            y = y + self.l(lo)
        if self.h is not None and hi is not None:
            y = y + up(self.h(hi), mid)
        return F.relu(self.fb(self.fc(y)) + mid)


class SIM(nn.Module):
    def __init__(self, ch=64, mid=32):
        super().__init__()
        self.h0 = ConvBNReLU(ch, mid, 3, 1, 1)
        self.l0 = ConvBNReLU(ch, mid, 3, 2, 1)
        self.h1 = ConvBNReLU(mid, mid, 3, 1, 1)
        self.l1 = ConvBNReLU(mid, mid, 3, 1, 1)
        self.oc = nn.Conv2d(mid, ch, 3, 1, 1, bias=False)
        self.ob = nn.BatchNorm2d(ch)

    def forward(self, x):
        h = self.h0(x)
        l = self.l0(x)
        ih = self.h1(up(l, h) + h)
        il = self.l1(F.interpolate(h, size=l.shape[2:], mode="bilinear",
                                   align_corners=False) + l)
        return F.relu(self.ob(self.oc(up(il, ih) + ih)) + x)


class MINetTorch(nn.Module):
    def __init__(self, ch=64):
        super().__init__()
        self.backbone = ResNet50()
        bc = self.backbone.out_channels
        self.trans = nn.ModuleList([ConvBNReLU(c, ch, 1) for c in bc])
        self.aims = nn.ModuleList([AIM(ch, i > 0, i < 4) for i in range(5)])
        self.sims = nn.ModuleList([SIM(ch) for _ in range(5)])
        self.ups = nn.ModuleList([ConvBNReLU(ch, ch, 3, 1, 1) for _ in range(4)])
        self.head = ConvBNReLU(ch, 32, 3, 1, 1)
        self.cls = nn.Conv2d(32, 1, 1)

    def forward(self, x):
        f = self.backbone(x)
        t = [tr(fi) for tr, fi in zip(self.trans, f)]
        a = [self.aims[i](t[i - 1] if i > 0 else None, t[i],
                          t[i + 1] if i < 4 else None) for i in range(5)]
        d = self.sims[4](a[4])
        for i in range(3, -1, -1):
            d = self.sims[i](self.ups[i](up(d, a[i]) + a[i]))
        return up(self.cls(self.head(d)), x)


def main():
    steps = int(sys.argv[1]) if len(sys.argv) > 1 else 20
    warm = 5
    torch.manual_seed(0)
    dev = torch.device("cuda")
    m = MINetTorch().to(dev).to(memory_format=torch.channels_last)
    m = m.to(torch.bfloat16)
    # BN in fp32 (same policy as sodkit)
    for mod in m.modules():
        if isinstance(mod, nn.BatchNorm2d):
            mod.float()
    opt = torch.optim.SGD(m.parameters(), lr=0.05, momentum=0.9,
                          weight_decay=5e-4, foreach=True)
    g = torch.Generator().manual_seed(1)
    x = torch.rand(32, 3, 352, 352, generator=g).to(dev).to(torch.bfloat16) \
        .contiguous(memory_format=torch.channels_last)
    t = (torch.rand(32, 1, 352, 352, generator=g) > 0.5).float().to(dev)

    def step():
        opt.zero_grad(set_to_none=True)
        z = m(x)
        loss = reference.hybrid_loss(z, t)
        loss.backward()
        opt.step()
        return loss

    for _ in range(warm):
        step()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(steps):
        step()
    torch.cuda.synchronize()
    dt = time.perf_counter() - t0
    print(f"torch-native MINet-R50 352 bf16 b32: "
          f"{32 * steps / dt:.1f} imgs/s  ({dt / steps * 1e3:.1f} ms/step)")


if __name__ == "__main__":
    main()
