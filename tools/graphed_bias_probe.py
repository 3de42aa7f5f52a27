"""Minimal repro probe: bias gradients under make_graphed_callables.

Hypothesis from the update-engine investigation: the bias-grad reduction
(sum over M rows) returns garbage on replay, possibly only when another
captured graph's replays interleave (pool/workspace aliasing).

    PYTHONPATH=. python tools/graphed_bias_probe.py
"""
import torch

torch.manual_seed(0)
dev = "cuda"
M, K, N = 26624, 2048, 2048
lin1 = torch.nn.Linear(K, N).to(dev)
lin2 = torch.nn.Linear(N, 256).to(dev)
x = torch.randn(M, K, device=dev)
x2 = torch.randn(4096, K, device=dev)


class Net(torch.nn.Module):
    def __init__(self):
        super().__init__()
        self.l1, self.l2 = lin1, lin2

    def forward(self, inp):
        with torch.autocast("cuda", dtype=torch.bfloat16,
                            cache_enabled=False):
            y = self.l2(torch.relu(self.l1(inp)))
        return y.float().square().mean()


net = Net()
params = list(net.parameters())


def eager_grads():
    loss = net(x)
    return torch.autograd.grad(loss, params)


g_ref = [g.clone() for g in eager_grads()]

# side no-grad graph over the same weights (mimics the engine's FRONT)
gF = torch.cuda.CUDAGraph()
s = torch.cuda.Stream()
s.wait_stream(torch.cuda.current_stream())
with torch.cuda.stream(s):
    for _ in range(3):
        with torch.no_grad():
            net(x2)
torch.cuda.current_stream().wait_stream(s)
with torch.cuda.graph(gF):
    with torch.no_grad():
        net(x2)

graphed = torch.cuda.make_graphed_callables(net, (x,))


def report(tag):
    rows = []
    for (n, p), r in zip(net.named_parameters(), g_ref):
        cos = torch.nn.functional.cosine_similarity(
            p.grad.float().flatten(), r.float().flatten(), dim=0).item()
        rows.append(f"{n} cos={cos:+.4f} |cap|={p.grad.norm():.3e} "
                    f"|ref|={r.norm():.3e}")
    print(tag, " | ".join(rows), flush=True)


for trial in range(4):
    if trial >= 2:
        gF.replay()      # interleave the no-grad graph from trial 2 on
    loss = graphed(x)
    for p in params:
        p.grad = None
    loss.backward()
    report(f"trial{trial} (interleaved={trial >= 2}):")
