"""RL weight sync: a trainer pushes weights, a generator fleet pulls them.

With GPUs, the direct path does ONE-SIDED reads straight out of the
trainer's live parameter memory over xGMI (the generator sees optimizer
updates after each `push`, ~15 ms for a full Llama-3-8B on MI355X).
On CPU this example uses the buffered store path with the same API.

Run:  python example/rl_weight_sync.py
"""

import asyncio
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

import torchstore_amd as ts


def make_model():
    torch.manual_seed(0)
    return torch.nn.Sequential(
        torch.nn.Linear(256, 512), torch.nn.ReLU(), torch.nn.Linear(512, 256)
    )


async def main():
    await ts.initialize(num_storage_volumes=1)
    # the one-sided direct path needs trainer and generator in DIFFERENT
    # processes (a process cannot open its own IPC handles) — see
    # bench.py's GeneratorActor for the real two-process topology.  This
    # single-process example uses the buffered store path.
    direct = False

    trainer = make_model()
    if direct:
        trainer.cuda()
    opt = torch.optim.SGD(trainer.parameters(), lr=0.1)

    generator = make_model()
    if direct:
        generator.cuda()
    with torch.no_grad():
        for p in generator.parameters():
            p.zero_()

    for step in range(3):
        # trainer: one optimizer step, then publish
        x = torch.randn(32, 256, device="cuda" if direct else "cpu")
        loss = trainer(x).square().mean()
        opt.zero_grad()
        loss.backward()
        opt.step()
        await ts.put_state_dict(trainer.state_dict(), "policy", direct=direct)

        # generator: sync weights
        sd = generator.state_dict()
        await ts.get_state_dict("policy", sd, direct=direct)
        generator.load_state_dict(sd)

        drift = max(
            (pt - pg).abs().max().item()
            for pt, pg in zip(trainer.parameters(), generator.parameters())
        )
        print(f"step {step}: loss={loss.item():.4f} max weight drift={drift:.2e}")
        assert drift == 0.0

    await ts.shutdown()
    print("ok")


if __name__ == "__main__":
    asyncio.run(main())
