import sys; sys.path.insert(0, "/root/repo")
import os, torch
from tensorlink_amd import ops
from tensorlink_amd.module import DistributedModel

def main():
    rank = int(os.environ["RANK"])
    m = DistributedModel("tiny", training=True, world_size=2, mode="torchrun",
                         lr=1e-3)
    if rank != 0:
        m.serve_worker()
        return
    opt = m.create_optimizer(lr=1e-3)
    torch.manual_seed(0)
    ids = torch.randint(0, 1024, (4, 12))
    losses = []
    for _ in range(3):
        logits = m(ids)
        loss = ops.causal_lm_loss(logits, ids)
        loss.backward()
        opt.step(); opt.zero_grad()
        losses.append(float(loss))
    out = m.generate(ids, max_new_tokens=4)
    assert out.shape == (4, 16)
    assert losses[-1] < losses[0], losses
    print("TORCHRUN-DM-OK", [round(l,3) for l in losses])
    m._bcast(("shutdown",))

if __name__ == "__main__":
    main()
