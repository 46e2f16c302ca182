"""Microbenchmark: per-node loop vs vmap(grad) vs manual grouped-conv for
the batched CNN update (B nodes x bs images), on whatever device is
available. Finds where the batched path's time goes on ROCm.
"""

import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

from benchmarks.onoszko_bench import cifar10net_factory


def timed(fn, iters=10, warmup=3, sync=True):
    for _ in range(warmup):
        fn()
    if sync and torch.cuda.is_available():
        torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    if sync and torch.cuda.is_available():
        torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters * 1000


def main(B=33, bs=32):
    dev = torch.device("cuda:0" if torch.cuda.is_available() else "cpu")
    module = cifar10net_factory().to(dev)
    names = [n for n, _ in module.named_parameters()]
    shapes = [tuple(p.shape) for p in module.parameters()]
    numels = [p.numel() for p in module.parameters()]
    D = sum(numels)
    torch.manual_seed(0)
    rows = (torch.randn(B, D, device=dev) * 0.05)
    x = torch.randn(B, bs, 3, 32, 32, device=dev)
    y = torch.randint(0, 10, (B, bs), device=dev)

    def stacked_params(r):
        out, off = {}, 0
        for n_, s_, c_ in zip(names, shapes, numels):
            out[n_] = r[:, off : off + c_].view(len(r), *s_)
            off += c_
        return out

    # --- per-node loop
    def loop():
        crit = torch.nn.CrossEntropyLoss()
        for i in range(B):
            off = 0
            with torch.no_grad():
                for p, c_ in zip(module.parameters(), numels):
                    p.copy_(rows[i, off : off + c_].view(p.shape))
                    off += c_
            opt = torch.optim.SGD(module.parameters(), lr=0.1)
            opt.zero_grad()
            crit(module(x[i]), y[i]).backward()
            opt.step()

    print(f"loop      B={B}: {timed(loop):8.2f} ms")

    # --- vmap(grad)
    import torch.func as tfunc

    def loss_fn(pd, xb, yb):
        out = tfunc.functional_call(module, pd, (xb,))
        return torch.nn.functional.cross_entropy(out, yb)

    gfn = tfunc.vmap(tfunc.grad(loss_fn))

    def vmapped():
        st = stacked_params(rows.clone())
        g = gfn(st, x, y)
        with torch.no_grad():
            for n_ in names:
                st[n_].add_(g[n_], alpha=-0.1)

    print(f"vmap-grad B={B}: {timed(vmapped):8.2f} ms")

    # vmap forward only
    def fwd(pd):
        return tfunc.functional_call(module, pd, (x.reshape(-1, 3, 32, 32)[:bs],))

    def vf():
        with torch.no_grad():
            tfunc.vmap(fwd)(stacked_params(rows))

    print(f"vmap-fwd  B={B}: {timed(vf):8.2f} ms")

    # --- manual grouped conv (wide module), standard autograd
    import torch.nn.functional as F

    def grouped():
        st = stacked_params(rows.clone())
        for v in st.values():
            v.requires_grad_(True)
        # x: [B, bs, 3, 32, 32] -> [bs, B*3, 32, 32]
        xg = x.permute(1, 0, 2, 3, 4).reshape(bs, B * 3, 32, 32)
        w1 = st["conv1.weight"].reshape(B * 32, 3, 3, 3)
        b1 = st["conv1.bias"].reshape(-1)
        h = F.relu(F.conv2d(xg, w1, b1, groups=B))
        h = F.max_pool2d(h, 2)
        w2 = st["conv2.weight"].reshape(B * 64, 32, 3, 3)
        h = F.relu(F.conv2d(h, w2, st["conv2.bias"].reshape(-1), groups=B))
        h = F.max_pool2d(h, 2)
        w3 = st["conv3.weight"].reshape(B * 64, 64, 3, 3)
        h = F.relu(F.conv2d(h, w3, st["conv3.bias"].reshape(-1), groups=B))
        h = F.max_pool2d(h, 2)  # [bs, B*64, 2, 2]
        h = h.reshape(bs, B, 256).permute(1, 0, 2)  # [B, bs, 256]
        h = F.relu(torch.baddbmm(
            st["fc1.bias"].unsqueeze(1), h, st["fc1.weight"].transpose(1, 2)
        ))
        out = torch.baddbmm(
            st["fc2.bias"].unsqueeze(1), h, st["fc2.weight"].transpose(1, 2)
        )  # [B, bs, 10]
        loss = F.cross_entropy(out.reshape(B * bs, 10), y.reshape(-1))
        loss.backward()

    print(f"grouped   B={B}: {timed(grouped):8.2f} ms")


if __name__ == "__main__":
    main(B=33)
    main(B=100)
    main(B=1)
