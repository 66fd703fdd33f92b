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
