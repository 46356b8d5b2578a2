"""Eval harness: perplexity matches hand-computed CE; multiple-choice picks
the continuation the model actually prefers (train a preference in)."""

import math

import torch
import torch.nn.functional as F

from applications.eval import evaluate_multiple_choice, evaluate_perplexity, sequence_loglikelihood
from colossalai_amd.models import LlamaConfig, LlamaForCausalLM


def _tiny():
    return LlamaConfig(vocab_size=64, hidden_size=32, intermediate_size=64, num_hidden_layers=2,
                       num_attention_heads=2, num_key_value_heads=2, max_position_embeddings=64)


def test_perplexity_matches_manual():
    torch.manual_seed(0)
    model = LlamaForCausalLM(_tiny()).eval()
    seq = list(range(20))
    res = evaluate_perplexity(model, [seq], max_len=64)
    ids = torch.tensor([seq])
    with torch.no_grad():
        logits = model(input_ids=ids)["logits"][0].float()
    nll = F.cross_entropy(logits[:-1], ids[0, 1:]).item()
    assert abs(res["nll"] - nll) < 1e-5
    assert abs(res["ppl"] - math.exp(nll)) < 1e-3
    assert res["tokens"] == 19


def test_perplexity_sliding_window_covers_all_tokens():
    torch.manual_seed(0)
    model = LlamaForCausalLM(_tiny()).eval()
    seq = list(range(50))
    res = evaluate_perplexity(model, [seq], max_len=32, stride=16)
    assert res["tokens"] == 49  # every target token scored exactly once


def test_multiple_choice_prefers_trained_continuation():
    torch.manual_seed(0)
    model = LlamaForCausalLM(_tiny())
    prompt = [1, 2, 3]
    good, bad = [10, 11, 12], [40, 41, 42]
    opt = torch.optim.AdamW(model.parameters(), lr=5e-3)
    x = torch.tensor([prompt + good])
    for _ in range(30):
        out = model(input_ids=x, labels=x)
        opt.zero_grad(); out["loss"].backward(); opt.step()
    model.eval()
    assert sequence_loglikelihood(model, prompt, good) > sequence_loglikelihood(model, prompt, bad)
    res = evaluate_multiple_choice(model, [{"prompt": prompt, "choices": [bad, good], "answer": 1}])
    assert res["accuracy"] == 1.0


def test_metrics():
    from applications.eval import exact_match, extract_numeric_answer, f1_score, first_choice, numeric_match

    assert exact_match("The Answer!", "answer") == 1.0
    assert 0.0 < f1_score("paris is the capital", "the capital is paris city") < 1.0
    assert first_choice("I think (B) is right") == "B"
    assert extract_numeric_answer("so 3+4 = 7 dollars #### 7") == "7"
    assert numeric_match("the total is 1,234.0", "#### 1234") == 1.0
    assert numeric_match("no idea", "#### 5") == 0.0


def test_dataset_loaders_and_generation(tmp_path):
    import json

    import torch

    from applications.eval import (
        build_choice_examples,
        build_generation_examples,
        evaluate_generation,
        evaluate_multiple_choice,
        greedy_generate,
        load_choice_csv,
        load_qa_jsonl,
        numeric_match,
    )
    from colossalai_amd.models import LlamaConfig, LlamaForCausalLM

    csv_p = tmp_path / "mmlu.csv"
    csv_p.write_text('what is 2+2?,three,four,five,six,B\npick A,yes,no,maybe,never,A\n')
    rows = load_choice_csv(csv_p)
    assert len(rows) == 2 and rows[0]["answer"] == 1 and len(rows[0]["choices"]) == 4

    jl = tmp_path / "gsm.jsonl"
    jl.write_text(json.dumps({"question": "2+2?", "answer": "#### 4"}) + "\n")
    qa = load_qa_jsonl(jl)
    assert qa[0]["answer"] == "#### 4"

    tokenize = lambda s: [min(ord(c), 127) for c in s][:32]
    detok = lambda ids: "".join(chr(i % 128) for i in ids)
    mc = build_choice_examples(rows, tokenize)
    gen = build_generation_examples(qa, tokenize)

    torch.manual_seed(0)
    model = LlamaForCausalLM(LlamaConfig(
        vocab_size=128, hidden_size=64, intermediate_size=128, num_hidden_layers=2,
        num_attention_heads=4, num_key_value_heads=2, max_position_embeddings=128))
    res = evaluate_multiple_choice(model, mc)
    assert 0.0 <= res["accuracy"] <= 1.0 and res["n"] == 2
    out = greedy_generate(model, gen[0]["prompt"], max_new_tokens=4)
    assert len(out) == 4
    score = evaluate_generation(model, gen, detok, numeric_match, max_new_tokens=4)
    assert score["n"] == 1


def _run_eval_tp(rank, world_size, port):
    """Loglikelihood eval on a TP-sharded model matches the unsharded one."""
    import copy

    import torch
    import torch.distributed as dist

    import colossalai_amd
    from applications.eval import evaluate_multiple_choice
    from colossalai_amd.models import LlamaConfig, LlamaForCausalLM
    from colossalai_amd.shardformer import ShardConfig, ShardFormer

    colossalai_amd.launch(rank, world_size, "127.0.0.1", port, backend="gloo", verbose=False)
    torch.manual_seed(0)
    cfg = LlamaConfig(vocab_size=128, hidden_size=64, intermediate_size=128, num_hidden_layers=2,
                      num_attention_heads=4, num_key_value_heads=2, max_position_embeddings=64)
    ref = LlamaForCausalLM(cfg).eval()
    sharded, _ = ShardFormer(ShardConfig(tensor_parallel_process_group=dist.group.WORLD,
                                         parallel_output=False)).optimize(copy.deepcopy(ref))
    examples = [
        {"prompt": [5, 9, 11], "choices": [[3, 4], [7], [2, 2, 2]], "answer": 1},
        {"prompt": [1, 2], "choices": [[8, 8], [9]], "answer": 0},
    ]
    a = evaluate_multiple_choice(ref, examples)
    b = evaluate_multiple_choice(sharded.eval(), examples)
    assert a == b
    dist.destroy_process_group()


def test_eval_tp2_matches():
    from colossalai_amd.testing import rerun_if_address_is_in_use, spawn

    rerun_if_address_is_in_use()(lambda: spawn(_run_eval_tp, 2))()
