"""HTTP serving front-end (reference: colossalai/inference/server/
api_server.py — FastAPI over the continuous-batching engine).

POST /generate accepts token ids (or text when a HF tokenizer is
attached) and returns the completed sequence; requests entering while
others are mid-decode join the running batch at the next engine step.
Run with ``uvicorn`` via ``python -m colossalai_amd.inference.server``.
"""

from typing import List, Optional

import torch

from .config import GenerationConfig, InferenceConfig
from .paged_engine import ContinuousBatchEngine

__all__ = ["create_app"]


def create_app(engine: ContinuousBatchEngine, tokenizer=None):
    from fastapi import FastAPI, HTTPException
    from pydantic import BaseModel

    class GenerateRequest(BaseModel):
        prompt_ids: Optional[List[int]] = None
        prompt: Optional[str] = None
        max_new_tokens: int = 64
        do_sample: bool = False
        temperature: float = 1.0
        top_k: int = 0
        top_p: float = 1.0

    class GenerateResponse(BaseModel):
        output_ids: List[int]
        output: Optional[str] = None

    app = FastAPI(title="colossalai_amd inference")

    @app.get("/health")
    def health():
        return {"status": "ok", "running": len(engine.rm.running), "waiting": len(engine.rm.waiting),
                "free_kv_blocks": engine.kv.free_blocks}

    class CompletionRequest(BaseModel):
        model: str = "colossalai_amd"
        prompt: Optional[str] = None
        prompt_ids: Optional[List[int]] = None
        max_tokens: int = 64
        temperature: float = 1.0
        top_p: float = 1.0
        n: int = 1

    @app.post("/v1/completions")
    def completions(req: CompletionRequest):
        """OpenAI-compatible completions endpoint (token ids accepted via
        the prompt_ids extension when no tokenizer is attached)."""
        if req.prompt_ids is None:
            if req.prompt is None or tokenizer is None:
                raise HTTPException(400, "provide prompt_ids, or prompt with a tokenizer attached")
            ids = tokenizer(req.prompt)["input_ids"]
        else:
            ids = req.prompt_ids
        gen = GenerationConfig(max_new_tokens=req.max_tokens, do_sample=req.temperature > 0,
                               temperature=max(req.temperature, 1e-5), top_p=req.top_p)
        choices = []
        for i in range(req.n):
            out = engine.generate([ids], gen)[0]
            new = out[len(ids):]
            choices.append({
                "index": i,
                "text": tokenizer.decode(new) if tokenizer is not None else None,
                "token_ids": new,
                "finish_reason": "length" if len(new) >= req.max_tokens else "stop",
            })
        return {"id": f"cmpl-{id(choices) & 0xFFFFFF:x}", "object": "text_completion",
                "model": req.model, "choices": choices,
                "usage": {"prompt_tokens": len(ids),
                          "completion_tokens": sum(len(c["token_ids"]) for c in choices),
                          "total_tokens": len(ids) + sum(len(c["token_ids"]) for c in choices)}}

    class ChatMessage(BaseModel):
        role: str
        content: str

    class ChatRequest(BaseModel):
        model: str = "colossalai_amd"
        messages: List[ChatMessage]
        max_tokens: int = 64
        temperature: float = 1.0
        top_p: float = 1.0

    @app.post("/v1/chat/completions")
    def chat_completions(req: ChatRequest):
        """OpenAI-compatible chat endpoint. Uses the tokenizer's chat
        template when available, else a plain role-tagged transcript."""
        if tokenizer is None:
            raise HTTPException(400, "chat completions need a tokenizer attached")
        msgs = [{"role": m.role, "content": m.content} for m in req.messages]
        if hasattr(tokenizer, "apply_chat_template"):
            ids = tokenizer.apply_chat_template(msgs, add_generation_prompt=True)
        else:
            text = "".join(f"<|{m['role']}|>\n{m['content']}\n" for m in msgs) + "<|assistant|>\n"
            ids = tokenizer(text)["input_ids"]
        gen = GenerationConfig(max_new_tokens=req.max_tokens, do_sample=req.temperature > 0,
                               temperature=max(req.temperature, 1e-5), top_p=req.top_p)
        out = engine.generate([list(ids)], gen)[0]
        new = out[len(ids):]
        return {"id": f"chatcmpl-{id(out) & 0xFFFFFF:x}", "object": "chat.completion",
                "model": req.model,
                "choices": [{"index": 0,
                             "message": {"role": "assistant", "content": tokenizer.decode(new)},
                             "finish_reason": "length" if len(new) >= req.max_tokens else "stop"}],
                "usage": {"prompt_tokens": len(ids), "completion_tokens": len(new),
                          "total_tokens": len(ids) + len(new)}}

    @app.post("/generate", response_model=GenerateResponse)
    def generate(req: GenerateRequest):
        if req.prompt_ids is None:
            if req.prompt is None or tokenizer is None:
                raise HTTPException(400, "provide prompt_ids, or prompt with a tokenizer attached")
            ids = tokenizer(req.prompt)["input_ids"]
        else:
            ids = req.prompt_ids
        gen = GenerationConfig(max_new_tokens=req.max_new_tokens, do_sample=req.do_sample,
                               temperature=req.temperature, top_k=req.top_k, top_p=req.top_p)
        out = engine.generate([ids], gen)[0]
        text = tokenizer.decode(out[len(ids):]) if tokenizer is not None else None
        return GenerateResponse(output_ids=out, output=text)

    return app


def main():
    import argparse

    import uvicorn

    from ..models.llama import LLAMA_CONFIGS, LlamaForCausalLM

    ap = argparse.ArgumentParser()
    ap.add_argument("--model", default="llama-7b", choices=list(LLAMA_CONFIGS))
    ap.add_argument("--host", default="127.0.0.1")
    ap.add_argument("--port", type=int, default=8000)
    ap.add_argument("--max-batch-size", type=int, default=32)
    args = ap.parse_args()

    model = LlamaForCausalLM(LLAMA_CONFIGS[args.model])
    if torch.cuda.is_available():
        model = model.to("cuda").bfloat16()
    engine = ContinuousBatchEngine(model, InferenceConfig(max_batch_size=args.max_batch_size))
    uvicorn.run(create_app(engine), host=args.host, port=args.port)


if __name__ == "__main__":
    main()
