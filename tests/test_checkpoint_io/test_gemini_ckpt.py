"""Chunk-sharded Gemini checkpoint round-trip on CPU/gloo world 2:
save model (unsharded + HF-sharded) and optimizer from a trained state,
reload into a fresh setup, verify weights and training continuity."""

import os

import torch
import torch.distributed as dist

import colossalai_amd
from colossalai_amd import Booster
from colossalai_amd.booster.plugin import GeminiPlugin
from colossalai_amd.models import LlamaConfig, LlamaForCausalLM
from colossalai_amd.nn import FusedAdam
from colossalai_amd.testing import assert_close_loose, rerun_if_address_is_in_use, spawn


def _cfg():
    return LlamaConfig(vocab_size=128, hidden_size=64, intermediate_size=128, num_hidden_layers=2,
                       num_attention_heads=4, num_key_value_heads=2, max_position_embeddings=64)


def _boost():
    torch.manual_seed(0)
    model = LlamaForCausalLM(_cfg())
    booster = Booster(plugin=GeminiPlugin(shard_param_frac=1.0, precision="fp32", min_chunk_size_m=1))
    opt = FusedAdam(model.parameters(), lr=1e-3)
    model_b, opt_b, *_ = booster.boost(model, opt)
    return booster, model_b, opt_b


def _run(rank, world_size, port, tmp_path):
    colossalai_amd.launch(rank, world_size, "127.0.0.1", port, backend="gloo", verbose=False)
    booster, model_b, opt_b = _boost()
    x = torch.randint(0, 128, (4, 16))
    for _ in range(2):
        out = model_b(input_ids=x, labels=x)
        opt_b.backward(out["loss"])
        opt_b.step()
        opt_b.zero_grad()

    mpath = os.path.join(tmp_path, "model.pt")
    spath = os.path.join(tmp_path, "sharded")
    opath = os.path.join(tmp_path, "optim.pt")
    booster.save_model(model_b, mpath)
    booster.save_model(model_b, spath, shard=True, size_per_shard=1)
    booster.save_optimizer(opt_b, opath)
    dist.barrier()

    want = model_b.state_dict()
    want_loss = model_b(input_ids=x, labels=x)["loss"]

    # fresh setup, load unsharded
    booster2, model2, opt2 = _boost()
    booster2.load_model(model2, mpath)
    booster2.load_optimizer(opt2, opath)
    got = model2.state_dict()
    for k, v in want.items():
        assert_close_loose(got[k], v, rtol=1e-6, atol=1e-7)
    assert_close_loose(model2(input_ids=x, labels=x)["loss"], want_loss, rtol=1e-5, atol=1e-6)

    # training continuity: one more identical step on both must agree
    for mb, ob in ((model_b, opt_b), (model2, opt2)):
        out = mb(input_ids=x, labels=x)
        ob.backward(out["loss"])
        ob.step()
        ob.zero_grad()
    for k, v in model_b.state_dict().items():
        assert_close_loose(model2.state_dict()[k], v, rtol=1e-5, atol=1e-6)

    # sharded (HF-style index) load
    booster3, model3, _ = _boost()
    booster3.load_model(model3, spath)
    got3 = model3.state_dict()
    for k, v in want.items():
        assert_close_loose(got3[k], v, rtol=1e-6, atol=1e-7)
    dist.destroy_process_group()


@rerun_if_address_is_in_use()
def test_gemini_ckpt_roundtrip(tmp_path):
    spawn(_run, 2, tmp_path=str(tmp_path))
