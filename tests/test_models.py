"""Real-model state_dict exchange (transformers, random init — no network).

Reference: tests/test_models.py pushes/pulls Qwen3-1.7B (HF_TOKEN-gated);
here a small random-init Llama-architecture model exercises the same path
ungated, plus the synthetic Llama-3-8B-shape generator used by the bench.
"""

import pytest
import torch

import torchstore_amd as ts
from torchstore_amd.strategy import SingletonStrategy


async def _with_store(body):
    await ts.initialize(
        num_storage_volumes=1,
        strategy=SingletonStrategy(),
        storage_device="cpu",
    )
    try:
        await body()
    finally:
        await ts.shutdown()


def _tiny_llama():
    transformers = pytest.importorskip("transformers")
    cfg = transformers.LlamaConfig(
        hidden_size=64,
        intermediate_size=128,
        num_attention_heads=4,
        num_key_value_heads=2,
        num_hidden_layers=2,
        vocab_size=256,
    )
    torch.manual_seed(0)
    return transformers.LlamaForCausalLM(cfg)


async def test_transformers_model_roundtrip():
    async def body():
        model = _tiny_llama()
        sd = model.state_dict()
        await ts.put_state_dict(sd, "model")
        dest_model = _tiny_llama()
        with torch.no_grad():
            for p in dest_model.parameters():
                p.zero_()
        out = await ts.get_state_dict("model", dest_model.state_dict())
        dest_model.load_state_dict(out)
        for (n1, p1), (n2, p2) in zip(
            model.state_dict().items(), dest_model.state_dict().items()
        ):
            assert n1 == n2
            assert torch.equal(p1, p2), n1

        # inference parity after the roundtrip
        ids = torch.randint(0, 255, (1, 16))
        with torch.no_grad():
            ref = model(ids).logits
            got = dest_model(ids).logits
        assert torch.equal(ref, got)

    await _with_store(body)


async def test_llama8b_shapes_layer_subset():
    """The bench's synthetic 8B generator roundtrips (1 layer on CPU)."""
    from torchstore_amd.models import llama

    async def body():
        sd = llama.make_sharded_state_dict(
            None, llama.fsdp_placement, device="cpu", layers=1, seed=3, scale=8
        )
        await ts.put_state_dict(sd, "m8b")
        out = await ts.get_state_dict("m8b")
        assert len(out) == len(sd)
        for k in sd:
            assert torch.equal(out[k], sd[k]), k

    await _with_store(body)


def test_llama8b_total_size():
    from torchstore_amd.models import llama

    shapes = llama.llama3_8b_shapes()
    n_params = sum(
        int(torch.tensor(s).prod()) for s in shapes.values()
    )
    # Meta-Llama-3-8B has 8.03B parameters
    assert 8.0e9 < n_params < 8.1e9, n_params


# ---------------------------------------------------------------------------
# FSDP2 (fully_shard) real-model exchange across independent worlds —
# the reference's workhorse (tests/test_state_dict.py:82-161, test_models.py)
# ---------------------------------------------------------------------------

import asyncio  # noqa: E402
import os  # noqa: E402
import tempfile  # noqa: E402
import uuid  # noqa: E402

from torchstore_amd.runtime import (  # noqa: E402
    Actor,
    actor_context,
    close_connections,
    endpoint,
    spawn_actors,
)


class FsdpWorker(Actor):
    """One rank of a fully_shard world sharing a tiny transformers model."""

    def __init__(self, world, pg_file, controller, seed):
        import torch.distributed as dist

        self.rank = actor_context().rank
        self.world = world
        os.environ["RANK"] = str(self.rank)
        dist.init_process_group(
            "gloo", init_method=f"file://{pg_file}",
            rank=self.rank, world_size=world,
        )
        from torchstore_amd import api
        from torchstore_amd.strategy import LocalRankStrategy

        api.attach(controller, LocalRankStrategy())
        from torch.distributed.device_mesh import init_device_mesh

        self.mesh = init_device_mesh("cpu", (world,))
        self.seed = seed

    def _model(self):
        import transformers

        cfg = transformers.LlamaConfig(
            hidden_size=64, intermediate_size=128, num_attention_heads=4,
            num_key_value_heads=2, num_hidden_layers=2, vocab_size=256,
        )
        torch.manual_seed(self.seed)
        model = transformers.LlamaForCausalLM(cfg)
        from torch.distributed.fsdp import fully_shard

        for layer in model.model.layers:
            fully_shard(layer, mesh=self.mesh)
        fully_shard(model, mesh=self.mesh)
        return model

    @endpoint
    async def push(self):
        model = self._model()
        await ts.put_state_dict(model.state_dict(), "fsdp")
        return "ok"

    @endpoint
    async def pull_and_check(self):
        model = self._model()  # same seed -> same reference weights
        sd = model.state_dict()
        with torch.no_grad():
            for v in sd.values():
                local = v.to_local() if hasattr(v, "to_local") else v
                local.zero_()
        out = await ts.get_state_dict("fsdp", sd)
        ref = self._model().state_dict()
        for k, v in out.items():
            got = v.to_local() if hasattr(v, "to_local") else v
            want = ref[k].to_local() if hasattr(ref[k], "to_local") else ref[k]
            if not torch.equal(got, want):
                return f"mismatch at {k}"
        return "ok"


@pytest.mark.parametrize("put_world,get_world", [(2, 2), (2, 3)])
async def test_fully_shard_model_across_worlds(put_world, get_world):
    """fully_shard (FSDP2) model pushed by one world, pulled — with
    resharding — by an independent world of a different size."""
    pytest.importorskip("transformers")
    controller = await ts.initialize(
        num_storage_volumes=put_world,
        strategy=None,
        storage_device="cpu",
    )
    put_mesh = get_mesh = None
    try:
        from torchstore_amd.strategy import LocalRankStrategy  # noqa: F401

        pg1 = tempfile.mktemp(prefix=f"fsdp-pg-{uuid.uuid4().hex[:6]}")
        put_mesh = await asyncio.to_thread(
            spawn_actors, put_world, FsdpWorker, "fsdp-put",
            put_world, pg1, controller, 123,
        )
        res = await put_mesh.push.call()
        assert res == ["ok"] * put_world
        pg2 = tempfile.mktemp(prefix=f"fsdp-pg-{uuid.uuid4().hex[:6]}")
        get_mesh = await asyncio.to_thread(
            spawn_actors, get_world, FsdpWorker, "fsdp-get",
            get_world, pg2, controller, 123,
        )
        res = await get_mesh.pull_and_check.call()
        assert res == ["ok"] * get_world, res
    finally:
        for m in (put_mesh, get_mesh):
            if m is not None:
                await m.stop()
        await ts.shutdown()
        await close_connections()


@pytest.mark.gpu
@pytest.mark.skipif(not torch.cuda.is_available(), reason="needs a GPU")
async def test_qwen3_1_7b_state_dict_roundtrip_gpu():
    """The reference's real-model workload (tests/test_models.py pushes
    Qwen3-1.7B, HF_TOKEN-gated): same architecture at full size, random
    init from the config — no network, no gating.  ~3.4 GB of bf16
    through the store with per-tensor bit-exact verification."""
    transformers = pytest.importorskip("transformers")
    import torchstore_amd as ts

    cfg = transformers.Qwen3Config(
        hidden_size=2048,
        intermediate_size=6144,
        num_hidden_layers=28,
        num_attention_heads=16,
        num_key_value_heads=8,
        head_dim=128,
        vocab_size=151936,
        tie_word_embeddings=True,
    )
    torch.manual_seed(11)
    with torch.device("cuda"):
        model = transformers.Qwen3ForCausalLM(cfg).to(torch.bfloat16)
    n_params = sum(p.numel() for p in model.parameters())
    assert 1.5e9 < n_params < 2.1e9, n_params

    await ts.initialize(
        num_storage_volumes=1,
        strategy=SingletonStrategy(),
        storage_device="auto",
    )
    try:
        sd = model.state_dict()
        await ts.put_state_dict(sd, "qwen3")
        dest = {k: torch.zeros_like(v) for k, v in sd.items()}
        out = await ts.get_state_dict("qwen3", dest)
        torch.cuda.synchronize()
        for k, v in sd.items():
            assert torch.equal(out[k], v), k
    finally:
        await ts.shutdown()
