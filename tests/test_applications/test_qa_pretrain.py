"""Retrieval QA pipeline + continued-pretraining trainer (CPU)."""

import os
import sys

import torch

sys.path.insert(0, os.path.join(os.path.dirname(__file__), "..", ".."))


def test_tfidf_retrieval_qa():
    from applications.qa import RetrievalQA, TfidfRetriever

    docs = [
        "The MI355X accelerator has 288 GB of HBM3E memory.",
        "Paris is the capital of France.",
        "Flash attention tiles K and V through the LDS.",
    ]
    retriever = TfidfRetriever(docs)
    hits = retriever.retrieve("How much HBM memory does the MI355X have?", k=2)
    assert hits and "288 GB" in hits[0][0]

    qa = RetrievalQA(retriever, generate_fn=lambda prompt: prompt.splitlines()[-2], k=1)
    prompt = qa.build_prompt("What is the capital of France?")
    assert "Paris" in prompt and "Question:" in prompt


def test_pretrain_trainer_resume(tmp_path):
    """Two steps, save, fresh trainer resumes, loss finite and step count kept."""
    import colossalai_amd  # noqa: F401
    from applications.pretrain import ContinuedPretrainTrainer
    from colossalai_amd import Booster
    from colossalai_amd.booster.plugin import TorchDDPPlugin
    from colossalai_amd.models import LlamaConfig, LlamaForCausalLM

    if not torch.distributed.is_initialized():
        colossalai_amd.launch(0, 1, "127.0.0.1", 29517, backend="gloo", verbose=False)

    cfg = LlamaConfig(vocab_size=128, hidden_size=64, intermediate_size=128, num_hidden_layers=2,
                      num_attention_heads=4, num_key_value_heads=2, max_position_embeddings=64)

    def make(save_dir=None):
        torch.manual_seed(0)
        model = LlamaForCausalLM(cfg).float()
        opt = torch.optim.AdamW(model.parameters(), lr=1e-3)
        return ContinuedPretrainTrainer(model, opt, Booster(plugin=TorchDDPPlugin()),
                                        save_dir=save_dir, save_interval=10)

    t1 = make()
    torch.manual_seed(3)
    x = torch.randint(0, 128, (2, 16))
    for _ in range(2):
        t1.train_step({"input_ids": x, "labels": x.clone()})
    t1.save(str(tmp_path))

    t2 = make()
    t2.load(str(tmp_path))
    assert t2.step_count == 2 and t2.tokens_seen == 64
    loss = t2.train_step({"input_ids": x, "labels": x.clone()})
    assert loss == loss  # finite
