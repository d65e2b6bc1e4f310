"""Cross-box reproducibility: sha256 of the split kernel's outputs for a
fixed seeded input. Run on two different gpurun boxes; hashes must match
(atomic-free design + fixed reduction order = bitwise reproducibility
across devices of the same arch)."""
import hashlib
import os
import sys

sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)),
                                ".."))
import torch  # noqa: E402

import mi355ml  # noqa: E402,F401
from mi355ml.ops import glm_value_grad, kmeans_assign  # noqa: E402


def h(t):
    return hashlib.sha256(t.cpu().numpy().tobytes()).hexdigest()[:16]


def main():
    g = torch.Generator(device="cuda").manual_seed(777)
    x = torch.randn(2_000_000, 100, generator=g, device="cuda")
    c = torch.randn(100, 100, generator=g, device="cuda")
    lab, sums, cnt, inertia = kmeans_assign(x, c)
    y = (torch.rand(2_000_000, generator=g, device="cuda") < 0.5).float()
    w = torch.randn(100, generator=g, device="cuda") * 0.05
    loss, gw, gb = glm_value_grad(x, y, w, torch.tensor(0.1), "logistic")
    print("labels ", h(lab))
    print("sums   ", h(sums))
    print("counts ", h(cnt))
    print("inertia", float(inertia))
    print("gw     ", h(gw))
    print("loss   ", float(loss))


if __name__ == "__main__":
    main()
