"""Pinpoint the step-2 divergence in the e2e ResNet50 train test: print the
max-|grad| per layer each step and the first activation explosion."""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch  # noqa: E402

from mpi_operator_amd import models  # noqa: E402
from mpi_operator_amd.optim import FusedSGD  # noqa: E402


def main():
    torch.manual_seed(50)
    m = models.to_mi355x(models.resnet50(num_classes=100), "cuda")
    m.train()
    x = torch.randn(4, 3, 64, 64, device="cuda", dtype=torch.bfloat16) \
        .contiguous(memory_format=torch.channels_last)
    y = torch.randint(0, 100, (4,), device="cuda")
    opt = FusedSGD(m.parameters(), lr=0.02, momentum=0.9)

    acts = {}

    def hook(name):
        def h(_mod, _in, out):
            o = out[0] if isinstance(out, tuple) else out
            if torch.is_tensor(o):
                acts[name] = o.float().abs().max().item()
        return h

    for name, mod in m.named_modules():
        if len(list(mod.children())) == 0:
            mod.register_forward_hook(hook(name))

    for step in range(4):
        opt.zero_grad()
        loss = m.loss(m(x), y)
        loss.backward()
        bad_act = [(n, v) for n, v in acts.items() if v > 1e4 or v != v]
        gmax = sorted(((p.grad.float().abs().max().item(), n)
                       for n, p in m.named_parameters() if p.grad is not None),
                      reverse=True)
        pmax = sorted(((p.float().abs().max().item(), n)
                       for n, p in m.named_parameters()), reverse=True)
        print(f"step {step}: loss={float(loss):.4f} "
              f"top grads={[(n.split('.')[-2:], round(v, 3)) for v, n in gmax[:4]]} "
              f"top params={[(n.split('.')[-2:], round(v, 3)) for v, n in pmax[:3]]}")
        if bad_act:
            print("  BAD ACTS:", bad_act[:8])
        opt.step()
        if float(loss) != float(loss) or float(loss) > 1e6:
            # rerun forward to find first exploding activation
            acts.clear()
            with torch.no_grad():
                m(x)
            first_bad = None
            for n, v in acts.items():
                if v > 1e4 or v != v:
                    first_bad = (n, v)
                    break
            print("  first bad act after step:", first_bad)
            break


if __name__ == "__main__":
    main()
