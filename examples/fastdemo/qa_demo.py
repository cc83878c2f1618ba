"""Browser QA demo (FastDemo parity, FastAPI instead of streamlit).

Behavioral parity: reference examples/FastDemo/YuyuanQA.py — a small web
demo that serves a QA model with an input box and answer display.
Streamlit is not part of this stack; the same UX ships as a single-file
FastAPI app with an inline HTML page (stdlib only beyond fastapi/uvicorn,
both in the image).

Run:  python qa_demo.py [--model_path ... --port 8000]
With no --model_path, a tiny random-weight GPT2 answers (smoke mode).
"""
from __future__ import annotations

import os
import sys

sys.path.insert(0, os.path.abspath(os.path.join(
    os.path.dirname(__file__), "..", "..")))


import argparse

import torch

_PAGE = """<!doctype html><html><head><meta charset="utf-8">
<title>Fengshen-AMD QA demo</title></head><body>
<h2>QA demo (MI355X serving path)</h2>
<form action="/qa" method="get">
  <input name="q" size="60" placeholder="question...">
  <button>Ask</button>
</form>
{answer}
</body></html>"""


def build_app(model, tokenizer):
    from fastapi import FastAPI
    from fastapi.responses import HTMLResponse

    app = FastAPI()

    @app.get("/", response_class=HTMLResponse)
    def index():
        return _PAGE.format(answer="")

    @app.get("/qa", response_class=HTMLResponse)
    def qa(q: str = ""):
        ids = torch.tensor([tokenizer.encode(f"问题：{q} 答案：")])
        device = next(model.parameters()).device
        with torch.no_grad():
            out = model.generate(ids.to(device), max_new_tokens=32,
                                 do_sample=False)
        text = tokenizer.decode(out[0][ids.shape[1]:].tolist())
        return _PAGE.format(answer=f"<p><b>A:</b> {text}</p>")

    return app


def main():
    parser = argparse.ArgumentParser()
    parser.add_argument("--model_path", default=None)
    parser.add_argument("--port", default=8000, type=int)
    parser.add_argument("--smoke", action="store_true",
                        help="build the app, hit / once, exit")
    args = parser.parse_args()

    if args.model_path:
        from transformers import AutoTokenizer

        from fengshen_amd.models.gpt2.modeling_gpt2 import GPT2LMHeadModel
        tokenizer = AutoTokenizer.from_pretrained(args.model_path)
        model = GPT2LMHeadModel.from_pretrained(args.model_path)
    else:
        from fengshen_amd.models.gpt2.configuration_gpt2 import (
            gpt2_tiny_config,
        )
        from fengshen_amd.models.gpt2.modeling_gpt2 import GPT2LMHeadModel
        from fengshen_amd.tokenizer import SimpleCharTokenizer
        tokenizer = SimpleCharTokenizer()
        model = GPT2LMHeadModel(gpt2_tiny_config(torch_dtype="float32")).float()
    model.eval()
    if torch.cuda.is_available():
        model.cuda()

    app = build_app(model, tokenizer)
    if args.smoke:
        from fastapi.testclient import TestClient
        client = TestClient(app)
        assert client.get("/").status_code == 200
        r = client.get("/qa", params={"q": "你好"})
        assert r.status_code == 200
        print("smoke ok:", r.text[:80].replace("\n", " "))
        return
    import uvicorn
    uvicorn.run(app, host="0.0.0.0", port=args.port)


if __name__ == "__main__":
    main()
